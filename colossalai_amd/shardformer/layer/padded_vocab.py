"""Vocab padding (reference: colossalai/tensor/padded_tensor/api.py).

Pads embedding / lm-head weights along the vocab dim to a multiple of
``make_vocab_size_divisible_by * tp_size`` so vocab-parallel sharding always
divides. Padded rows are zero and logits for them are masked to -inf.
"""

from typing import Tuple

import torch
import torch.nn as nn

__all__ = ["pad_vocab", "padded_vocab_size", "unpad_vocab_weight"]


def padded_vocab_size(vocab_size: int, divisor: int) -> int:
    return (vocab_size + divisor - 1) // divisor * divisor


def pad_vocab(model: nn.Module, tp_size: int, make_divisible_by: int = 64) -> Tuple[int, int]:
    """Pad `embed_tokens`/`lm_head`-style modules in place. Returns
    (original_vocab, padded_vocab)."""
    divisor = make_divisible_by * max(tp_size, 1)
    orig = None
    padded = None
    for module in model.modules():
        if isinstance(module, nn.Embedding):
            orig = module.num_embeddings
            padded = padded_vocab_size(orig, divisor)
            if padded != orig:
                w = module.weight.data
                new = torch.zeros(padded, w.shape[1], dtype=w.dtype, device=w.device)
                new[:orig] = w
                module.weight = nn.Parameter(new)
                module.num_embeddings = padded
        elif isinstance(module, nn.Linear) and getattr(module, "out_features", 0) == (orig or -1):
            # lm_head (untied): pad output rows to match
            w = module.weight.data
            new = torch.zeros(padded, w.shape[1], dtype=w.dtype, device=w.device)
            new[: w.shape[0]] = w
            module.weight = nn.Parameter(new)
            module.out_features = padded
    return orig, padded


def unpad_vocab_weight(weight: torch.Tensor, original_vocab: int) -> torch.Tensor:
    """Drop padded rows on checkpoint save."""
    return weight[:original_vocab]
