"""Vocab-parallel embedding (reference: colossalai/shardformer/layer/embedding.py:241).

Each rank holds a contiguous vocab slice; out-of-slice tokens embed to zero
and the partial embeddings are all-reduced.
"""

import torch
import torch.distributed as dist
import torch.nn as nn
from torch.nn.parameter import Parameter

from ._operation import reduce_forward
from .parallel_module import ParallelModule

__all__ = ["VocabParallelEmbedding1D"]


class VocabParallelEmbedding1D(ParallelModule):
    def __init__(self, num_embeddings: int, embedding_dim: int, device=None, dtype=None, process_group=None, **kwargs):
        super().__init__()
        self.process_group = process_group
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.rank = dist.get_rank(process_group) if dist.is_initialized() else 0
        assert num_embeddings % self.world == 0, "pad the vocab to a multiple of tp size first"
        self.num_embeddings = num_embeddings
        self.part = num_embeddings // self.world
        self.vocab_start = self.rank * self.part
        self.vocab_end = self.vocab_start + self.part
        self.weight = Parameter(torch.empty(self.part, embedding_dim, device=device, dtype=dtype))

    @classmethod
    def from_native_module(cls, module: nn.Embedding, process_group=None, **kwargs) -> "VocabParallelEmbedding1D":
        layer = cls.__new__(cls)
        ParallelModule.__init__(layer)
        layer.process_group = process_group
        layer.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        layer.rank = dist.get_rank(process_group) if dist.is_initialized() else 0
        assert module.num_embeddings % layer.world == 0, "vocab must divide tp size (use vocab padding)"
        layer.num_embeddings = module.num_embeddings
        layer.part = module.num_embeddings // layer.world
        layer.vocab_start = layer.rank * layer.part
        layer.vocab_end = layer.vocab_start + layer.part
        layer.weight = Parameter(module.weight.data[layer.vocab_start : layer.vocab_end].contiguous())
        layer.weight.tp_sharded = True
        layer.weight.tp_dim = 0
        return layer

    def gather_weight(self) -> torch.Tensor:
        if self.world == 1:
            return self.weight.data
        parts = [torch.empty_like(self.weight.data) for _ in range(self.world)]
        dist.all_gather(parts, self.weight.data.contiguous(), group=self.process_group)
        return torch.cat(parts, dim=0)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        if self.world == 1:
            return torch.nn.functional.embedding(input_ids, self.weight)
        mask = (input_ids < self.vocab_start) | (input_ids >= self.vocab_end)
        local = input_ids.clamp(self.vocab_start, self.vocab_end - 1) - self.vocab_start
        out = torch.nn.functional.embedding(local, self.weight)
        out = out.masked_fill(mask.unsqueeze(-1), 0.0)
        return reduce_forward(out, self.process_group)
