"""Continuous-batching engine over the paged KV cache
(reference: colossalai/inference/core/llm_engine.py + request_handler.py,
rebuilt around the MI355X paged decode kernel).

Each ``step()``: admit waiting requests (prefill them as one right-padded
flash batch, sample their first tokens), then decode every running
sequence one token with ``decode_attention_paged`` over the block pools.
Short requests retire and free their blocks immediately; long prompts
stream in the moment blocks are available — throughput is bounded by the
8 TB/s KV stream, not by the longest member of a static batch.
"""

from typing import Dict, List, Optional, Union

import torch

from ..models.llama import LlamaForCausalLM
from .config import GenerationConfig, InferenceConfig
from .engine import LLMEngine
from .kv_cache import KVCacheManager
from .request_manager import Request, RequestManager

__all__ = ["ContinuousBatchEngine"]


class ContinuousBatchEngine:
    def __init__(
        self,
        model: LlamaForCausalLM,
        config: Optional[InferenceConfig] = None,
        num_blocks: Optional[int] = None,
        block_size: int = 16,
    ):
        self.config = config or InferenceConfig()
        self.model = model.eval()
        self.device = next(model.parameters()).device
        self.dtype = next(model.parameters()).dtype
        cfg = model.config
        if num_blocks is None:
            # worst case: every slot runs a max-length sequence
            num_blocks = self.config.max_batch_size * (
                (self.config.max_seq_len + block_size - 1) // block_size
            )
        # per-rank layer attributes, not the config: under TP each rank
        # caches only its local KV heads (same rule as LLMEngine)
        attn0 = model.model.layers[0].self_attn
        self.kv = KVCacheManager(cfg.num_hidden_layers, attn0.num_kv_heads, attn0.head_dim,
                                 num_blocks=num_blocks, block_size=block_size,
                                 device=self.device, dtype=self.dtype)
        self.rm = RequestManager(self.kv, max_batch_size=self.config.max_batch_size)
        self._gen = GenerationConfig()

    # -------------------------------------------------------------- requests
    def add_request(self, prompt: List[int], max_new_tokens: Optional[int] = None) -> int:
        return self.rm.add_request(list(prompt), max_new_tokens or self._gen.max_new_tokens)

    # --------------------------------------------------------------- forward
    def _forward(self, input_ids, positions, seq_ids, block_tables, seq_lens, prefill):
        from ..ops import fused_add_rms_norm, rms_norm

        m = self.model.model
        table = m.rope_table(self.device)
        slot_rows = None
        if not prefill:
            # pool row of each sequence's current token, shared by all layers
            bs = self.kv.block_size
            rows = [self.kv.table(s)[(int(seq_lens[i]) - 1) // bs] * bs + (int(seq_lens[i]) - 1) % bs
                    for i, s in enumerate(seq_ids)]
            slot_rows = torch.tensor(rows, dtype=torch.long, device=self.device)
        residual = m.embed_tokens(input_ids)
        hidden = rms_norm(residual, m.layers[0].input_layernorm_weight, m.eps)
        n = len(m.layers)
        for i, layer in enumerate(m.layers):
            attn_out = layer.self_attn.forward_with_paged_cache(
                hidden, table, self.kv, i, seq_ids, block_tables, positions, seq_lens, prefill,
                slot_rows=slot_rows,
            )
            hidden, residual = fused_add_rms_norm(attn_out, residual,
                                                  layer.post_attention_layernorm_weight, m.eps)
            mlp_out = layer.mlp(hidden)
            next_w = m.layers[i + 1].input_layernorm_weight if i + 1 < n else m.norm_weight
            hidden, residual = fused_add_rms_norm(mlp_out, residual, next_w, m.eps)
        if prefill:
            B = input_ids.shape[0]
            gather = (seq_lens - 1).long().view(B, 1, 1).expand(B, 1, hidden.shape[-1])
            hidden = hidden.gather(1, gather)
        return self.model.lm_head(hidden[:, -1]).float()

    # ------------------------------------------------------------------ step
    @torch.inference_mode()
    def step(self) -> Dict[int, List[int]]:
        """One engine iteration; returns {request_id: full tokens} for
        requests that finished during this step."""
        gen, cfg = self._gen, self.config
        finished: Dict[int, List[int]] = {}

        admitted = self.rm.schedule()
        if admitted:
            prompts = [r.prompt for r in admitted]
            lens = torch.tensor([len(p) for p in prompts], device=self.device, dtype=torch.int32)
            S = int(lens.max())
            padded = torch.full((len(prompts), S), cfg.pad_token_id, dtype=torch.long, device=self.device)
            for i, p in enumerate(prompts):
                padded[i, : len(p)] = torch.tensor(p, device=self.device)
            positions = torch.arange(S, device=self.device).unsqueeze(0).expand(len(prompts), S).reshape(-1).int()
            logits = self._forward(padded, positions, [r.request_id for r in admitted], None, lens, True)
            tokens = LLMEngine._sample(logits, gen)
            for i, r in enumerate(admitted):
                self.rm.append_token(r, int(tokens[i]), cfg.eos_token_id, cfg.max_seq_len)
                if r.status.value == "finished":
                    finished[r.request_id] = r.tokens

        running = list(self.rm.running)
        if running:
            seq_ids = [r.request_id for r in running]
            step_ids = torch.tensor([[r.tokens[-1]] for r in running], dtype=torch.long, device=self.device)
            seq_lens = torch.tensor([r.seq_len for r in running], device=self.device, dtype=torch.int32)
            positions = (seq_lens - 1).int()
            bt = self.kv.block_tables_tensor(seq_ids)
            logits = self._forward(step_ids, positions, seq_ids, bt, seq_lens, False)
            tokens = LLMEngine._sample(logits, gen)
            for i, r in enumerate(running):
                self.rm.append_token(r, int(tokens[i]), cfg.eos_token_id, cfg.max_seq_len)
                if r.status.value == "finished":
                    finished[r.request_id] = r.tokens
        return finished

    # ----------------------------------------------------------- convenience
    @torch.inference_mode()
    def generate(
        self,
        input_ids: Union[torch.Tensor, List[List[int]]],
        generation_config: Optional[GenerationConfig] = None,
    ) -> List[List[int]]:
        self._gen = generation_config or GenerationConfig()
        if isinstance(input_ids, torch.Tensor):
            prompts = [row.tolist() for row in input_ids]
        else:
            prompts = [list(p) for p in input_ids]
        ids = [self.add_request(p, self._gen.max_new_tokens) for p in prompts]
        results: Dict[int, List[int]] = {}
        while self.rm.has_work:
            results.update(self.step())
        return [results[i] for i in ids]
