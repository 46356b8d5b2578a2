"""RCCL self-consistency on one GPU (world 1, nccl backend): the ZeRO
engine's reduce/all-gather path and the fp8 collectives execute on real
RCCL rather than only on gloo oracles."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_zero2_rccl_world1():
    if not torch.cuda.is_available():
        pytest.skip("GPU only")
    import torch.distributed as dist

    import colossalai_amd
    from colossalai_amd import Booster
    from colossalai_amd.booster.plugin import LowLevelZeroPlugin
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
    from colossalai_amd.nn import FusedAdam

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29519")
    if not dist.is_initialized():
        colossalai_amd.launch(0, 1, "127.0.0.1", 29519, backend="nccl", verbose=False)

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=256, intermediate_size=512, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    model = LlamaForCausalLM(cfg)
    plugin = LowLevelZeroPlugin(stage=2, precision="bf16", overlap_communication=True)
    booster = Booster(plugin=plugin)
    opt = FusedAdam(model.parameters(), lr=1e-2)
    model_b, opt_b, *_ = booster.boost(model, opt)

    x = torch.randint(0, 128, (2, 32), device="cuda")
    losses = []
    for _ in range(4):
        out = model_b(input_ids=x, labels=x)
        booster.backward(out["loss"], opt_b)
        opt_b.step()
        opt_b.zero_grad()
        losses.append(float(out["loss"]))
    assert losses[-1] < losses[0], f"RCCL world-1 ZeRO training did not learn: {losses}"

    # fp8 collectives execute on RCCL
    from colossalai_amd.quantization.fp8 import all_gather_fp8, all_reduce_fp8

    t = torch.randn(1024, device="cuda", dtype=torch.bfloat16)
    ref = t.clone()
    all_reduce_fp8(t)
    g = all_gather_fp8(ref)
    assert g.shape[0] == 1024 and torch.isfinite(t).all()
