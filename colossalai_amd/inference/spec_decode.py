"""Speculative decoding (reference: colossalai/inference/core/llm_engine.py
:301-388 speculative + GLIDE path — greedy draft-verify rebuilt on the
native cache kernels).

A small draft model proposes ``gamma`` tokens autoregressively; the target
model scores all of them in ONE chunk forward against its KV cache —
attention of the chunk = LSE-merge of (non-causal chunk×prefix) and
(causal chunk×chunk), the same block merge the ring attention uses, so no
special kernel is needed. Greedy acceptance: keep draft tokens while they
match the target argmax; the first mismatch is replaced by the target's
token (one guaranteed token per round, up to gamma+1).

Cache rollback is free: sequence length bookkeeping moves back and stale
cache rows are overwritten by position on the next write.
"""

from typing import List, Optional, Tuple

import torch

from ..models.llama import LlamaForCausalLM
from .config import GenerationConfig, InferenceConfig

__all__ = ["SpeculativeEngine", "BatchedSpeculativeEngine"]


def _chunk_attend(attn, hidden, table, kcache, vcache, prefix_len: int, positions):
    """Chunk of new tokens vs cache prefix + itself (causal)."""
    from ..ops import has_kernels
    from ..shardformer.layer.ring_attn import _block_fwd, _merge

    B, S, _ = hidden.shape
    Hq, Hkv, D = attn.num_heads, attn.num_kv_heads, attn.head_dim
    qkv = attn.qkv_proj(hidden)
    q = qkv[:, :, : Hq * D].view(B, S, Hq, D)
    k = qkv[:, :, Hq * D : (Hq + Hkv) * D].view(B, S, Hkv, D)
    v = qkv[:, :, (Hq + Hkv) * D :].view(B, S, Hkv, D)
    if hidden.is_cuda and has_kernels():
        from ..ops import kernels

        kernels().rope_inplace(q, k, table, positions, False)
    else:
        from ..ops.rope import apply_rope_ref

        q, k = apply_rope_ref(q, k, table, positions.long(), S, False)
    kcache[:, prefix_len : prefix_len + S] = k
    vcache[:, prefix_len : prefix_len + S] = v
    out, lse = _block_fwd(q.contiguous(), k.contiguous(), v.contiguous(), True, attn.scale)
    if prefix_len > 0:
        o_pre, l_pre = _block_fwd(q.contiguous(), kcache[:, :prefix_len].contiguous(),
                                  vcache[:, :prefix_len].contiguous(), False, attn.scale)
        out, lse = _merge(out, lse.float(), o_pre, l_pre.float())
    return attn.o_proj(out.reshape(B, S, Hq * D))


class _CachedModel:
    """Target/draft wrapper: KV caches + chunk forward returning all-position logits."""

    def __init__(self, model: LlamaForCausalLM, max_seq_len: int):
        self.model = model.eval()
        self.device = next(model.parameters()).device
        self.dtype = next(model.parameters()).dtype
        cfg = model.config
        attn0 = model.model.layers[0].self_attn  # per-rank heads under TP
        self.caches = [
            (torch.zeros(1, max_seq_len, attn0.num_kv_heads, attn0.head_dim,
                         device=self.device, dtype=self.dtype),
             torch.zeros(1, max_seq_len, attn0.num_kv_heads, attn0.head_dim,
                         device=self.device, dtype=self.dtype))
            for _ in range(cfg.num_hidden_layers)
        ]
        self.len = 0

    @torch.inference_mode()
    def forward_chunk(self, token_ids: List[int]) -> torch.Tensor:
        """Append token_ids at the current length; -> logits [S, V]."""
        from ..ops import fused_add_rms_norm, rms_norm

        m = self.model.model
        ids = torch.tensor([token_ids], device=self.device)
        S = ids.shape[1]
        positions = (torch.arange(S, device=self.device) + self.len).int()
        table = m.rope_table(self.device)
        residual = m.embed_tokens(ids)
        hidden = rms_norm(residual, m.layers[0].input_layernorm_weight, m.eps)
        n = len(m.layers)
        for i, layer in enumerate(m.layers):
            kc, vc = self.caches[i]
            attn_out = _chunk_attend(layer.self_attn, hidden, table, kc, vc, self.len, positions)
            hidden, residual = fused_add_rms_norm(attn_out, residual,
                                                  layer.post_attention_layernorm_weight, m.eps)
            mlp_out = layer.mlp(hidden)
            next_w = m.layers[i + 1].input_layernorm_weight if i + 1 < n else m.norm_weight
            hidden, residual = fused_add_rms_norm(mlp_out, residual, next_w, m.eps)
        self.len += S
        return self.model.lm_head(hidden[0]).float()

    def rollback(self, new_len: int):
        self.len = new_len


class SpeculativeEngine:
    """Greedy speculative decoding, batch size 1."""

    def __init__(self, target: LlamaForCausalLM, draft: LlamaForCausalLM,
                 config: Optional[InferenceConfig] = None, gamma: int = 4):
        self.config = config or InferenceConfig()
        self.gamma = gamma
        self.target = _CachedModel(target, self.config.max_seq_len)
        self.draft = _CachedModel(draft, self.config.max_seq_len)
        self.accepted = 0
        self.proposed = 0

    @torch.inference_mode()
    def generate(self, prompt: List[int], generation_config: Optional[GenerationConfig] = None) -> List[int]:
        gen = generation_config or GenerationConfig()
        self.target.rollback(0)
        self.draft.rollback(0)
        seq = list(prompt)
        t_logits = self.target.forward_chunk(seq)
        self.draft.forward_chunk(seq)
        next_tok = int(t_logits[-1].argmax())
        seq.append(next_tok)

        while len(seq) - len(prompt) < gen.max_new_tokens and len(seq) < self.config.max_seq_len - self.gamma - 1:
            # ---- draft proposes gamma tokens from the last accepted token
            proposal = []
            q_probs = []  # draft prob of each proposed token (sampling mode)
            tok = next_tok
            for _ in range(self.gamma):
                d_logits = self.draft.forward_chunk([tok])
                if gen.do_sample:
                    q = torch.softmax(d_logits[-1] / max(gen.temperature, 1e-5), dim=-1)
                    tok = int(torch.multinomial(q, 1))
                    q_probs.append(q)
                else:
                    tok = int(d_logits[-1].argmax())
                proposal.append(tok)
            # ---- target verifies [next_tok, proposal...] in one chunk
            t_logits = self.target.forward_chunk([next_tok] + proposal)
            self.proposed += len(proposal)
            if gen.do_sample:
                # Leviathan et al. acceptance: keep x_j with prob
                # min(1, p(x_j)/q(x_j)); on rejection resample from
                # max(0, p - q) normalized
                n_acc = 0
                correction = None
                for j, x in enumerate(proposal):
                    p = torch.softmax(t_logits[j] / max(gen.temperature, 1e-5), dim=-1)
                    q = q_probs[j]
                    ratio = float(p[x] / q[x].clamp_min(1e-20))
                    if float(torch.rand(())) < ratio:
                        n_acc += 1
                        continue
                    resid = (p - q).clamp_min(0)
                    s = resid.sum()
                    correction = int(torch.multinomial(resid / s if s > 0 else p, 1))
                    break
                if correction is None:  # all accepted: bonus from the target
                    p = torch.softmax(t_logits[self.gamma] / max(gen.temperature, 1e-5), dim=-1)
                    correction = int(torch.multinomial(p, 1))
            else:
                n_acc = 0
                for j, x in enumerate(proposal):
                    if x == int(t_logits[j].argmax()):
                        n_acc += 1
                    else:
                        break
                correction = int(t_logits[n_acc].argmax())
            self.accepted += n_acc
            accepted_tokens = proposal[:n_acc]
            seq.extend(accepted_tokens)
            seq.append(correction)
            next_tok = correction
            # ---- resync both caches to cover exactly seq[:-1]
            needed = len(seq) - 1
            self.target.rollback(needed)
            if self.draft.len > needed:
                self.draft.rollback(needed)
            elif self.draft.len < needed:
                # every draft token accepted: the last proposal was never
                # fed to the draft — replay the gap
                self.draft.forward_chunk(seq[self.draft.len : needed])
            if self.config.eos_token_id is not None and correction == self.config.eos_token_id:
                break
        return seq[: len(prompt) + gen.max_new_tokens]

    @property
    def acceptance_rate(self) -> float:
        return self.accepted / max(self.proposed, 1)


class _BatchedCachedModel:
    """Batch-of-sequences cache wrapper with per-sequence lengths.

    Single-token chunks run the batched decode path
    (``forward_with_cache``, native decode kernel on GPU); multi-token
    chunks (prefill / verification) run causal flash over the chunk and
    LSE-merge a per-sequence non-causal block against the cache prefix."""

    def __init__(self, model: LlamaForCausalLM, batch: int, max_seq_len: int):
        self.model = model.eval()
        self.device = next(model.parameters()).device
        self.dtype = next(model.parameters()).dtype
        cfg = model.config
        attn0 = model.model.layers[0].self_attn  # per-rank heads under TP
        self.caches = [
            (torch.zeros(batch, max_seq_len, attn0.num_kv_heads, attn0.head_dim,
                         device=self.device, dtype=self.dtype),
             torch.zeros(batch, max_seq_len, attn0.num_kv_heads, attn0.head_dim,
                         device=self.device, dtype=self.dtype))
            for _ in range(cfg.num_hidden_layers)
        ]
        self.lens = torch.zeros(batch, dtype=torch.int32, device=self.device)

    @torch.inference_mode()
    def forward_chunk(self, ids: torch.Tensor, valid: Optional[torch.Tensor] = None) -> torch.Tensor:
        """ids [B, n] appended at per-seq self.lens; valid [B] caps how many
        of the n tokens are real (prefill right-padding). -> logits
        [B, n, V]. Advances lens by ``valid`` (or n)."""
        from ..ops import fused_add_rms_norm, rms_norm
        from ..shardformer.layer.ring_attn import _block_fwd, _merge

        B, n = ids.shape
        m = self.model.model
        table = m.rope_table(self.device)
        n_valid = valid if valid is not None else torch.full((B,), n, dtype=torch.int32,
                                                            device=self.device)
        if n == 1:
            new_lens = (self.lens + 1).int()
            positions = (new_lens - 1).int()
            residual = m.embed_tokens(ids)
            hidden = rms_norm(residual, m.layers[0].input_layernorm_weight, m.eps)
            nl = len(m.layers)
            for i, layer in enumerate(m.layers):
                kc, vc = self.caches[i]
                attn = layer.self_attn.forward_with_cache(hidden, table, kc, vc, positions,
                                                          new_lens, prefill=False)
                hidden, residual = fused_add_rms_norm(attn, residual,
                                                      layer.post_attention_layernorm_weight, m.eps)
                mlp_out = layer.mlp(hidden)
                next_w = m.layers[i + 1].input_layernorm_weight if i + 1 < nl else m.norm_weight
                hidden, residual = fused_add_rms_norm(mlp_out, residual, next_w, m.eps)
            self.lens = new_lens
            return self.model.lm_head(hidden).float()

        # multi-token chunk with per-seq prefix lengths
        positions = (self.lens[:, None] + torch.arange(n, device=self.device)[None]).reshape(-1).int()
        residual = m.embed_tokens(ids)
        hidden = rms_norm(residual, m.layers[0].input_layernorm_weight, m.eps)
        nl = len(m.layers)
        for i, layer in enumerate(m.layers):
            attn_mod = layer.self_attn
            Hq, Hkv, D = attn_mod.num_heads, attn_mod.num_kv_heads, attn_mod.head_dim
            qkv = attn_mod.qkv_proj(hidden)
            q = qkv[:, :, : Hq * D].view(B, n, Hq, D)
            k = qkv[:, :, Hq * D : (Hq + Hkv) * D].view(B, n, Hkv, D)
            v = qkv[:, :, (Hq + Hkv) * D :].view(B, n, Hkv, D)
            from ..ops import has_kernels

            if hidden.is_cuda and has_kernels():
                from ..ops import kernels

                kernels().rope_inplace(q, k, table, positions, False)
            else:
                from ..ops.rope import apply_rope_ref

                q, k = apply_rope_ref(q, k, table, positions.long(), n, False)
            kc, vc = self.caches[i]
            for b in range(B):
                lo = int(self.lens[b])
                nb = int(n_valid[b])
                kc[b, lo : lo + nb] = k[b, :nb]
                vc[b, lo : lo + nb] = v[b, :nb]
            out, lse = _block_fwd(q.contiguous(), k.contiguous(), v.contiguous(), True,
                                  attn_mod.scale)
            lse = lse.float()
            merged = []
            for b in range(B):
                lo = int(self.lens[b])
                if lo == 0:
                    merged.append(out[b : b + 1])
                    continue
                o_pre, l_pre = _block_fwd(q[b : b + 1].contiguous(), kc[b : b + 1, :lo].contiguous(),
                                          vc[b : b + 1, :lo].contiguous(), False, attn_mod.scale)
                o_m, _ = _merge(out[b : b + 1], lse[b : b + 1], o_pre, l_pre.float())
                merged.append(o_m)
            attn = torch.cat(merged, dim=0).reshape(B, n, Hq * D)
            attn = attn_mod.o_proj(attn)
            hidden, residual = fused_add_rms_norm(attn, residual,
                                                  layer.post_attention_layernorm_weight, m.eps)
            mlp_out = layer.mlp(hidden)
            next_w = m.layers[i + 1].input_layernorm_weight if i + 1 < nl else m.norm_weight
            hidden, residual = fused_add_rms_norm(mlp_out, residual, next_w, m.eps)
        self.lens = (self.lens + n_valid).int()
        return self.model.lm_head(hidden).float()

    def rollback(self, new_lens: torch.Tensor):
        self.lens = new_lens.int().to(self.device)


class BatchedSpeculativeEngine:
    """Greedy speculative decoding over a batch of sequences: the draft
    proposes gamma tokens for every sequence; one batched chunk forward
    verifies all of them; acceptance/rollback are per-sequence length
    bookkeeping (cache rows overwrite by position)."""

    def __init__(self, target: LlamaForCausalLM, draft: LlamaForCausalLM,
                 config: Optional[InferenceConfig] = None, gamma: int = 4):
        self.config = config or InferenceConfig()
        self.gamma = gamma
        self._target_model = target
        self._draft_model = draft
        self.accepted = 0
        self.proposed = 0

    @torch.inference_mode()
    def generate(self, prompts, generation_config: Optional[GenerationConfig] = None):
        gen = generation_config or GenerationConfig()
        B = len(prompts)
        dev = next(self._target_model.parameters()).device
        target = _BatchedCachedModel(self._target_model, B, self.config.max_seq_len)
        draft = _BatchedCachedModel(self._draft_model, B, self.config.max_seq_len)

        plens = torch.tensor([len(p) for p in prompts], dtype=torch.int32, device=dev)
        S = int(plens.max())
        padded = torch.full((B, S), self.config.pad_token_id, dtype=torch.long, device=dev)
        for b, p in enumerate(prompts):
            padded[b, : len(p)] = torch.tensor(p, device=dev)
        t_logits = target.forward_chunk(padded, valid=plens)
        draft.forward_chunk(padded, valid=plens)
        gather = (plens.long() - 1).view(B, 1, 1).expand(B, 1, t_logits.shape[-1])
        next_tok = t_logits.gather(1, gather)[:, 0].argmax(-1)  # [B]

        seqs = [list(p) + [int(next_tok[b])] for b, p in enumerate(prompts)]
        done = [False] * B
        G = self.gamma
        while not all(done):
            if max(len(s) for s in seqs) + G + 2 > self.config.max_seq_len:
                break
            # ---- draft proposes G tokens for every sequence (batched)
            proposals = torch.zeros(B, G, dtype=torch.long, device=dev)
            tok = next_tok.view(B, 1)
            for j in range(G):
                d_logits = draft.forward_chunk(tok)
                tok = d_logits[:, -1].argmax(-1, keepdim=True)
                proposals[:, j] = tok[:, 0]
            draft.forward_chunk(tok)  # feed the last proposal: keeps draft cache uniform
            # ---- verify [next_tok | proposals] in one batched chunk
            chunk = torch.cat([next_tok.view(B, 1), proposals], dim=1)
            t_logits = target.forward_chunk(chunk)
            want = t_logits.argmax(-1)  # [B, G+1]: target's next token after each prefix
            n_acc = torch.zeros(B, dtype=torch.long)
            for b in range(B):
                j = 0
                while j < G and int(proposals[b, j]) == int(want[b, j]):
                    j += 1
                n_acc[b] = j
            corrections = want[torch.arange(B), n_acc]  # first mismatch (or bonus)
            for b in range(B):
                if done[b]:
                    continue
                self.proposed += G
                self.accepted += int(n_acc[b])
                budget = gen.max_new_tokens - (len(seqs[b]) - len(prompts[b]))
                new = [int(x) for x in proposals[b, : int(n_acc[b])]] + [int(corrections[b])]
                seqs[b].extend(new[:budget])
                if len(seqs[b]) - len(prompts[b]) >= gen.max_new_tokens:
                    done[b] = True
            next_tok = torch.tensor([s[-1] for s in seqs], device=dev)
            new_lens = torch.tensor([len(s) - 1 for s in seqs], dtype=torch.int32, device=dev)
            target.rollback(new_lens)
            draft.rollback(new_lens)
        return [s[: len(p) + gen.max_new_tokens] for s, p in zip(seqs, prompts)]

    @property
    def acceptance_rate(self) -> float:
        return self.accepted / max(self.proposed, 1)


class GlideCrossAttention(torch.nn.Module):
    """GLIDE's glance: the drafter cross-attends to the TARGET model's KV
    cache so a tiny draft stays anchored to the big model's context
    (reference: colossalai/inference/modeling/models/glide_llama.py).
    Output projection is zero-initialized — an untrained glance is a
    no-op, training moves it off the identity."""

    def __init__(self, d_draft: int, t_kv_heads: int, t_head_dim: int):
        super().__init__()
        import torch.nn as nn

        self.t_kv_heads = t_kv_heads
        self.t_head_dim = t_head_dim
        inner = t_kv_heads * t_head_dim
        self.q_proj = nn.Linear(d_draft, inner, bias=False)
        self.o_proj = nn.Linear(inner, d_draft, bias=False)
        nn.init.zeros_(self.o_proj.weight)
        self.scale = t_head_dim ** -0.5

    def forward(self, hidden, kcache, vcache, prefix_len: int):
        """hidden [1, S, d]; kcache/vcache [1, max, Hkv, D] target cache.
        The cache holds only verified prefix rows, so attending to ALL of
        it is causally safe for every drafted position."""
        if prefix_len == 0:
            return hidden
        B, S, _ = hidden.shape
        H, D = self.t_kv_heads, self.t_head_dim
        q = self.q_proj(hidden).view(B, S, H, D).permute(0, 2, 1, 3)
        k = kcache[:, :prefix_len].permute(0, 2, 1, 3).to(q.dtype)
        v = vcache[:, :prefix_len].permute(0, 2, 1, 3).to(q.dtype)
        attn = torch.softmax((q @ k.transpose(-1, -2)) * self.scale, dim=-1) @ v
        out = self.o_proj(attn.permute(0, 2, 1, 3).reshape(B, S, H * D))
        return hidden + out


class _GlideCachedModel(_CachedModel):
    """Draft wrapper that glances at the target's last-layer KV before the
    LM head."""

    def __init__(self, model: LlamaForCausalLM, cross: GlideCrossAttention,
                 target: _CachedModel, max_seq_len: int):
        super().__init__(model, max_seq_len)
        self.cross = cross.to(self.device, self.dtype)
        self.target = target

    @torch.inference_mode()
    def forward_chunk(self, token_ids: List[int]) -> torch.Tensor:
        from ..ops import fused_add_rms_norm, rms_norm

        m = self.model.model
        ids = torch.tensor([token_ids], device=self.device)
        S = ids.shape[1]
        positions = (torch.arange(S, device=self.device) + self.len).int()
        table = m.rope_table(self.device)
        residual = m.embed_tokens(ids)
        hidden = rms_norm(residual, m.layers[0].input_layernorm_weight, m.eps)
        n = len(m.layers)
        for i, layer in enumerate(m.layers):
            kc, vc = self.caches[i]
            attn_out = _chunk_attend(layer.self_attn, hidden, table, kc, vc, self.len, positions)
            hidden, residual = fused_add_rms_norm(attn_out, residual,
                                                  layer.post_attention_layernorm_weight, m.eps)
            mlp_out = layer.mlp(hidden)
            next_w = m.layers[i + 1].input_layernorm_weight if i + 1 < n else m.norm_weight
            hidden, residual = fused_add_rms_norm(mlp_out, residual, next_w, m.eps)
        # GLIDE glance: target's LAST layer cache over the verified prefix
        t_kc, t_vc = self.target.caches[-1]
        hidden = self.cross(hidden, t_kc, t_vc, self.target.len)
        self.len += S
        return self.model.lm_head(hidden[0]).float()


class GlideSpeculativeEngine(SpeculativeEngine):
    """Speculative decoding with a GLIDE drafter: identical accept/verify
    loop (still lossless under greedy), but the draft model reuses the
    target's KV cache through a cross-attention glance."""

    def __init__(self, target: LlamaForCausalLM, draft: LlamaForCausalLM,
                 cross: Optional[GlideCrossAttention] = None,
                 config: Optional[InferenceConfig] = None, gamma: int = 4):
        super().__init__(target, draft, config, gamma)
        if cross is None:
            tcfg, dcfg = target.config, draft.config
            cross = GlideCrossAttention(dcfg.hidden_size, tcfg.num_key_value_heads, tcfg.head_dim)
        self.draft = _GlideCachedModel(draft, cross, self.target, self.config.max_seq_len)
