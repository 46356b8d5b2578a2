"""FP8 collectives vs fp32 reference (CPU/gloo; fp8 payloads as uint8 views)."""

import pytest
import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd.quantization import (
    all_gather_fp8,
    all_reduce_fp8,
    all_to_all_single_fp8,
    cast_from_fp8,
    cast_to_fp8,
    reduce_scatter_fp8,
)
from colossalai_amd.testing import rerun_if_address_is_in_use, spawn


def test_cast_roundtrip():
    torch.manual_seed(0)
    x = torch.randn(1000) * 5
    for fmt in ("e4m3", "e5m2"):
        f, s = cast_to_fp8(x, fmt)
        y = cast_from_fp8(f, s, torch.float32)
        rel = ((x - y).abs() / x.abs().clamp(min=1e-3)).median()
        assert rel < 0.08, f"{fmt}: median rel err {rel}"


def _run(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    g = dist.group.WORLD
    torch.manual_seed(rank)
    x = torch.randn(64)
    ref = x.clone()
    dist.all_reduce(ref, group=g)
    y = x.clone()
    all_reduce_fp8(y, group=g)
    assert ((y - ref).abs() / ref.abs().clamp(min=1e-2)).median() < 0.1

    shard = torch.randn(32)
    out = all_gather_fp8(shard, group=g)
    refs = [torch.empty_like(shard) for _ in range(world_size)]
    dist.all_gather(refs, shard, group=g)
    ref_full = torch.cat(refs)
    assert ((out - ref_full).abs() / ref_full.abs().clamp(min=1e-2)).median() < 0.1

    a2a_in = torch.randn(world_size * 4, 8)
    out2 = all_to_all_single_fp8(a2a_in, group=g)
    assert out2.shape == a2a_in.shape
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_fp8_collectives():
    spawn(_run, 2)


def _run_plugins(rank, world_size, port):
    import copy

    import colossalai_amd
    from colossalai_amd import Booster
    from colossalai_amd.booster.plugin import LowLevelZeroPlugin, TorchDDPPlugin
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
    from colossalai_amd.nn import FusedAdam

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    base = LlamaForCausalLM(cfg)
    x = torch.randint(0, 128, (4, 16))

    # ---- DDP with fp8 grad comm hook vs plain DDP: grads close, not exact
    m_fp8 = copy.deepcopy(base)
    m_ref = copy.deepcopy(base)
    b1 = Booster(plugin=TorchDDPPlugin(fp8_communication=True))
    b2 = Booster(plugin=TorchDDPPlugin())
    m_fp8, o1, *_ = b1.boost(m_fp8, FusedAdam(m_fp8.parameters(), lr=1e-3))
    m_ref, o2, *_ = b2.boost(m_ref, FusedAdam(m_ref.parameters(), lr=1e-3))
    for model, opt in ((m_fp8, o1), (m_ref, o2)):
        loss = model(input_ids=x, labels=x)["loss"]
        opt.backward(loss)
    for p8, pr in zip(m_fp8.unwrap().parameters(), m_ref.unwrap().parameters()):
        assert p8.grad is not None and torch.isfinite(p8.grad).all()
        denom = pr.grad.abs().max().clamp_min(1e-6)
        assert (p8.grad - pr.grad).abs().max() / denom < 0.1, "fp8 grad too far from fp32 grad"
    o1.step(); o2.step()

    # ---- ZeRO-1 with fp8 weight all-gather: steps run, ranks stay in sync
    m_z = copy.deepcopy(base)
    bz = Booster(plugin=LowLevelZeroPlugin(stage=1, precision="fp32", overlap_communication=False,
                                           fp8_communication=True))
    m_z, oz, *_ = bz.boost(m_z, FusedAdam(m_z.parameters(), lr=1e-3))
    for _ in range(2):
        loss = m_z(input_ids=x, labels=x)["loss"]
        assert torch.isfinite(loss)
        oz.backward(loss)
        oz.step()
        oz.zero_grad()
    # replicated params must be BITWISE identical across ranks after the
    # fp8 gather (both ranks decode the same wire bytes)
    for p in m_z.unwrap().parameters():
        clone = p.detach().clone()
        dist.broadcast(clone, src=0)
        assert torch.equal(clone, p.detach()), "ranks diverged after fp8 all-gather"
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_fp8_plugin_flags():
    spawn(_run_plugins, 2)


def test_fp8_linear_cpu_wiring():
    """CPU path: plain linear fallback with working autograd."""
    from colossalai_amd.quantization import Fp8Linear, fp8_linear

    torch.manual_seed(0)
    x = torch.randn(4, 8, requires_grad=True)
    lin = torch.nn.Linear(8, 6)
    out = fp8_linear(x, lin.weight, lin.bias)
    ref = torch.nn.functional.linear(x, lin.weight, lin.bias)
    torch.testing.assert_close(out, ref)
    out.sum().backward()
    assert x.grad is not None and lin.weight.grad is not None

    f = Fp8Linear.from_linear(lin)
    torch.testing.assert_close(f(x.detach()), ref.detach())
