"""Paged KV cache manager (reference: colossalai/inference/kv_cache/
kvcache_manager.py — vLLM-style block pool, MI355X sizing).

One K and one V pool per layer, shaped [num_blocks, block_size, Hkv, D]
(block_size a power of two so the decode kernel folds addressing into
shift/mask). 288 GB HBM3E holds ~1.5M tokens of llama-7B KV at bf16, so
the default pool is sized generously and admission control, not eviction,
is the knob. Blocks are handed out from a free list; sequences own a
block-id list that becomes a row of the [B, max_blocks] block-table tensor
fed to ``decode_attention_paged``.
"""

from typing import Dict, List, Optional

import torch

__all__ = ["KVCacheManager"]


class KVCacheManager:
    def __init__(
        self,
        num_layers: int,
        num_kv_heads: int,
        head_dim: int,
        num_blocks: int = 2048,
        block_size: int = 16,
        device=None,
        dtype=torch.bfloat16,
    ):
        assert block_size & (block_size - 1) == 0, "block_size must be a power of two"
        self.num_layers = num_layers
        self.block_size = block_size
        self.num_blocks = num_blocks
        self.device = device or ("cuda" if torch.cuda.is_available() else "cpu")
        self.k_pools = [
            torch.zeros(num_blocks, block_size, num_kv_heads, head_dim, device=self.device, dtype=dtype)
            for _ in range(num_layers)
        ]
        self.v_pools = [
            torch.zeros(num_blocks, block_size, num_kv_heads, head_dim, device=self.device, dtype=dtype)
            for _ in range(num_layers)
        ]
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))
        self._tables: Dict[int, List[int]] = {}  # seq id -> block ids

    # ------------------------------------------------------------- accounting
    @property
    def free_blocks(self) -> int:
        return len(self._free)

    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size

    def can_allocate(self, num_tokens: int) -> bool:
        return self.blocks_needed(num_tokens) <= self.free_blocks

    # ------------------------------------------------------------- allocation
    def allocate(self, seq_id: int, num_tokens: int) -> None:
        need = self.blocks_needed(num_tokens)
        assert need <= self.free_blocks, "KV pool exhausted"
        self._tables[seq_id] = [self._free.pop() for _ in range(need)]

    def extend(self, seq_id: int, new_len: int) -> None:
        """Grow seq's table to cover new_len tokens (decode append)."""
        table = self._tables[seq_id]
        while len(table) * self.block_size < new_len:
            assert self._free, "KV pool exhausted"
            table.append(self._free.pop())

    def free(self, seq_id: int) -> None:
        self._free.extend(reversed(self._tables.pop(seq_id, [])))

    def table(self, seq_id: int) -> List[int]:
        return self._tables[seq_id]

    def block_tables_tensor(self, seq_ids: List[int]) -> torch.Tensor:
        mb = max(len(self._tables[s]) for s in seq_ids)
        bt = torch.zeros(len(seq_ids), mb, dtype=torch.int32, device=self.device)
        for i, s in enumerate(seq_ids):
            ids = self._tables[s]
            bt[i, : len(ids)] = torch.tensor(ids, dtype=torch.int32, device=self.device)
        return bt

    # ---------------------------------------------------------------- writes
    def write_prefill(self, layer: int, seq_id: int, k: torch.Tensor, v: torch.Tensor) -> None:
        """k/v [S, Hkv, D] for one sequence's prompt."""
        S = k.shape[0]
        bs = self.block_size
        table = self._tables[seq_id]
        for bi in range((S + bs - 1) // bs):
            lo, hi = bi * bs, min((bi + 1) * bs, S)
            self.k_pools[layer][table[bi], : hi - lo] = k[lo:hi]
            self.v_pools[layer][table[bi], : hi - lo] = v[lo:hi]

    def write_token(self, layer: int, seq_id: int, pos: int, k: torch.Tensor, v: torch.Tensor) -> None:
        """k/v [Hkv, D] for the token at position pos."""
        bs = self.block_size
        blk = self._tables[seq_id][pos // bs]
        self.k_pools[layer][blk, pos % bs] = k
        self.v_pools[layer][blk, pos % bs] = v

    # ----------------------------------------------------------------- reads
    def gather_contiguous(self, layer: int, seq_id: int, length: int):
        """CPU/reference path: materialize [length, Hkv, D] K and V."""
        table = self._tables[seq_id]
        nb = self.blocks_needed(length)
        idx = torch.tensor(table[:nb], dtype=torch.long, device=self.device)
        k = self.k_pools[layer][idx].reshape(-1, *self.k_pools[layer].shape[2:])[:length]
        v = self.v_pools[layer][idx].reshape(-1, *self.v_pools[layer].shape[2:])[:length]
        return k, v
