"""Batched generation engine for the native model families
(reference: colossalai/inference/core/llm_engine.py — lean MI355X rewrite).

Prefill runs the flash-attention kernel over the right-padded batch (causal
masking makes pad garbage self-contained); decode steps run the dedicated
single-token decode kernel against contiguous per-sequence KV caches sized
for 288 GB HBM3E. Greedy and temperature/top-k/top-p sampling.
"""

from typing import List, Optional, Union

import torch

from ..models.llama import LlamaForCausalLM
from .config import GenerationConfig, InferenceConfig

__all__ = ["LLMEngine"]


class LLMEngine:
    def __init__(self, model: LlamaForCausalLM, config: Optional[InferenceConfig] = None):
        self.config = config or InferenceConfig()
        self.model = model.eval()
        self.device = next(model.parameters()).device
        self.dtype = next(model.parameters()).dtype
        self._caches = None
        self._cache_bs = 0
        self._graphs = {}  # batch size -> (hipGraph, input buffers, output)

    def _ensure_caches(self, batch_size: int):
        Smax = self.config.max_seq_len
        if self._caches is not None and self._cache_bs >= batch_size:
            return
        # size caches from the LAYER attributes, not the config: under TP
        # (ShardFormer-sharded model, gather_output LM head) each rank holds
        # num_kv_heads / tp local heads and caches only those
        attn0 = self.model.model.layers[0].self_attn
        self._caches = [
            (
                torch.zeros(batch_size, Smax, attn0.num_kv_heads, attn0.head_dim,
                            device=self.device, dtype=self.dtype),
                torch.zeros(batch_size, Smax, attn0.num_kv_heads, attn0.head_dim,
                            device=self.device, dtype=self.dtype),
            )
            for _ in range(len(self.model.model.layers))
        ]
        self._cache_bs = batch_size

    def _forward(self, input_ids: torch.Tensor, positions: torch.Tensor, seq_lens: torch.Tensor, prefill: bool):
        """Run the decoder with caches; returns logits for the LAST position of
        each sequence ([B, V])."""
        from ..ops import fused_add_rms_norm, rms_norm

        m = self.model.model
        table = m.rope_table(self.device)
        residual = m.embed_tokens(input_ids)
        hidden = rms_norm(residual, m.layers[0].input_layernorm_weight, m.eps)
        n = len(m.layers)
        for i, layer in enumerate(m.layers):
            kc, vc = self._caches[i]
            B = input_ids.shape[0]
            attn_out = layer.self_attn.forward_with_cache(
                hidden, table, kc[:B], vc[:B], positions, seq_lens, prefill
            )
            hidden, residual = fused_add_rms_norm(attn_out, residual, layer.post_attention_layernorm_weight, m.eps)
            mlp_out = layer.mlp(hidden)
            next_w = m.layers[i + 1].input_layernorm_weight if i + 1 < n else m.norm_weight
            hidden, residual = fused_add_rms_norm(mlp_out, residual, next_w, m.eps)
        if prefill:
            B = input_ids.shape[0]
            gather = (seq_lens - 1).long().view(B, 1, 1).expand(B, 1, hidden.shape[-1])
            hidden = hidden.gather(1, gather)
        logits = self.model.lm_head(hidden[:, -1])
        return logits.float()

    def _decode_graphed(self, step_ids, positions, seq_lens):
        """Decode step via hipGraph replay (reference idea:
        colossalai/inference/core/llm_engine.py:213 CUDA-graph capture).
        The whole 32-layer decode is launch-bound on MI355X (hundreds of
        ~µs kernels); one graph per batch size removes the gaps. Inputs
        are copied into capture-time buffers, then the graph replays onto
        the same KV caches."""
        B = step_ids.shape[0]
        if B not in self._graphs:
            buf = (step_ids.clone(), positions.clone(), seq_lens.clone())
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):  # warmup outside capture (torch requirement)
                    self._forward(buf[0], buf[1], buf[2], prefill=False)
            torch.cuda.current_stream().wait_stream(s)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                out = self._forward(buf[0], buf[1], buf[2], prefill=False)
            self._graphs[B] = (graph, buf, out)
        graph, buf, out = self._graphs[B]
        buf[0].copy_(step_ids)
        buf[1].copy_(positions)
        buf[2].copy_(seq_lens)
        graph.replay()
        return out

    @staticmethod
    def _sample(logits: torch.Tensor, gen: GenerationConfig) -> torch.Tensor:
        if not gen.do_sample:
            return logits.argmax(dim=-1)
        logits = logits / max(gen.temperature, 1e-5)
        if gen.top_k > 0:
            kth = logits.topk(gen.top_k, dim=-1).values[..., -1, None]
            logits = logits.masked_fill(logits < kth, float("-inf"))
        probs = torch.softmax(logits, dim=-1)
        if gen.top_p < 1.0:
            sorted_probs, sorted_idx = probs.sort(dim=-1, descending=True)
            cum = sorted_probs.cumsum(-1)
            mask = cum - sorted_probs > gen.top_p
            sorted_probs = sorted_probs.masked_fill(mask, 0.0)
            sorted_probs /= sorted_probs.sum(-1, keepdim=True)
            pick = torch.multinomial(sorted_probs, 1).squeeze(-1)
            return sorted_idx.gather(-1, pick.unsqueeze(-1)).squeeze(-1)
        return torch.multinomial(probs, 1).squeeze(-1)

    @torch.inference_mode()
    def generate(
        self,
        input_ids: Union[torch.Tensor, List[List[int]]],
        generation_config: Optional[GenerationConfig] = None,
    ) -> List[List[int]]:
        """input_ids: [B, S] tensor or ragged list of token lists.
        Returns full sequences (prompt + generated) per batch element."""
        gen = generation_config or GenerationConfig()
        cfg = self.config
        if isinstance(input_ids, torch.Tensor):
            prompts = [row.tolist() for row in input_ids]
        else:
            prompts = [list(p) for p in input_ids]
        B = len(prompts)
        assert B <= cfg.max_batch_size, f"batch {B} > max_batch_size {cfg.max_batch_size}"
        prompt_lens = torch.tensor([len(p) for p in prompts], device=self.device, dtype=torch.int32)
        S = int(prompt_lens.max())
        padded = torch.full((B, S), cfg.pad_token_id, dtype=torch.long, device=self.device)
        for i, p in enumerate(prompts):
            padded[i, : len(p)] = torch.tensor(p, device=self.device)
        self._ensure_caches(B)

        positions = (torch.arange(S, device=self.device).unsqueeze(0).expand(B, S)).reshape(-1).int()
        logits = self._forward(padded, positions, prompt_lens, prefill=True)
        tokens = self._sample(logits, gen)

        sequences = [list(p) for p in prompts]
        seq_lens = prompt_lens.clone()
        finished = torch.zeros(B, dtype=torch.bool, device=self.device)
        for i in range(B):
            sequences[i].append(int(tokens[i]))
        seq_lens += 1

        for _ in range(gen.max_new_tokens - 1):
            if bool(finished.all()):
                break
            step_ids = tokens.view(B, 1)
            positions = (seq_lens - 1).int()
            if cfg.use_hip_graph and self.device.type == "cuda":
                logits = self._decode_graphed(step_ids, positions, seq_lens.int())
            else:
                logits = self._forward(step_ids, positions, seq_lens, prefill=False)
            tokens = self._sample(logits, gen)
            for i in range(B):
                if not bool(finished[i]):
                    sequences[i].append(int(tokens[i]))
            seq_lens = seq_lens + (~finished).int()
            if cfg.eos_token_id is not None:
                finished |= tokens == cfg.eos_token_id
            if int(seq_lens.max()) >= cfg.max_seq_len:
                break
        return sequences
