"""Gemini auto placement: chunk-size search + runtime residency pinning
(reference: zero/gemini/placement_policy.py:128 Auto, chunk/search_utils.py)."""

import torch

from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
from colossalai_amd.zero.gemini.gemini_ddp import GeminiDDP, search_chunk_size


def _tiny():
    return LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=4,
                       num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)


def test_search_chunk_size():
    model = LlamaForCausalLM(_tiny())
    m = search_chunk_size(model, candidates_m=(1, 2, 4))
    assert m in (1, 2, 4)
    # a model this small should prefer the smallest candidate (no undersized
    # tails either way, fewest-chunks tiebreak only kicks in above target)
    big = search_chunk_size(model, candidates_m=(64,))
    assert big == 64


def test_auto_residency_pins_chunks():
    model = GeminiDDP(LlamaForCausalLM(_tiny()).float(), chunk_size_m=1, precision="fp32")
    released = [c for c in model.chunks if not c.persistent]
    assert released, "expected releasable unit chunks"
    # plenty of (simulated) capacity: everything should pin
    pinned = model.auto_adjust_residency(memory_ratio=0.9, capacity_bytes=1 << 40)
    assert pinned == len(released)
    assert all(c.persistent and c.gathered for c in model.chunks)
    # a forward still works with everything resident
    x = torch.randint(0, 128, (2, 16))
    out = model(input_ids=x, labels=x)
    assert torch.isfinite(out["loss"])
    # zero capacity: nothing left to pin
    assert model.auto_adjust_residency(capacity_bytes=0) == 0
