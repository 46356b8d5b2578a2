"""Native chunk-sharded Gemini (ZeRO-3) oracle tests: GeminiDDP +
GeminiOptimizer on gloo world 2 must track an unsharded fp32 AdamW model
bit-for-bit step by step (same seed, same per-rank batch averaged by hand
on the oracle). Mirrors the reference's tests/test_zero/test_gemini suite."""

import copy

import pytest
import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
from colossalai_amd.nn import FusedAdam
from colossalai_amd.testing import rerun_if_address_is_in_use, spawn
from colossalai_amd.zero import GeminiDDP, GeminiOptimizer


def _cfg():
    return LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                       num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)


def run_gemini_oracle(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    model = LlamaForCausalLM(_cfg())
    ref = copy.deepcopy(model).float()
    gm = GeminiDDP(copy.deepcopy(model), precision="fp32", chunk_size_m=1)
    opt = GeminiOptimizer(FusedAdam(gm.parameters(), lr=1e-2), gm)
    ref_opt = FusedAdam(ref.parameters(), lr=1e-2)

    # every chunk must actually be sharded (flat released between steps)
    assert any(not c.persistent for c in gm.chunks)

    for step in range(3):
        torch.manual_seed(100 + step)
        xs = [torch.randint(0, 128, (2, 16)) for _ in range(world_size)]
        out = gm(input_ids=xs[rank], labels=xs[rank])
        opt.backward(out["loss"])
        opt.step()
        opt.zero_grad()

        # oracle: average of per-rank grads on the full model
        ref_opt.zero_grad()
        losses = []
        for x in xs:
            r = ref(input_ids=x, labels=x)["loss"] / world_size
            r.backward()
            losses.append(r)
        ref_opt.step()
        ref_loss = sum(l.item() for l in losses) * world_size / world_size
        assert torch.isfinite(out["loss"])

    # non-persistent chunks are released outside fwd/bwd
    for c in gm.chunks:
        if not c.persistent:
            assert not c.gathered

    # final weights must match the oracle exactly (fp32, same math)
    sd = gm.state_dict()
    ref_sd = ref.state_dict()
    for k, v in sd.items():
        assert torch.allclose(v.float(), ref_sd[k].float(), atol=1e-5, rtol=1e-5), f"{k} diverged"
    dist.destroy_process_group()


def run_gemini_plugin(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    from colossalai_amd import Booster
    from colossalai_amd.booster.plugin import GeminiPlugin

    torch.manual_seed(0)
    model = LlamaForCausalLM(_cfg())
    booster = Booster(plugin=GeminiPlugin(shard_param_frac=1.0, precision="fp32", min_chunk_size_m=1))
    optimizer = FusedAdam(model.parameters(), lr=1e-3)
    model_b, optimizer_b, *_ = booster.boost(model, optimizer)
    x = torch.randint(0, 128, (4, 16))
    for _ in range(2):
        out = model_b(input_ids=x, labels=x)
        assert torch.isfinite(out["loss"])
        optimizer_b.backward(out["loss"])
        optimizer_b.step()
        optimizer_b.zero_grad()
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_gemini_ddp_oracle():
    spawn(run_gemini_oracle, 2)


@rerun_if_address_is_in_use()
def test_gemini_plugin_sharded():
    spawn(run_gemini_plugin, 2)


def run_gemini_no_sync(rank, world_size, port):
    """Grad accumulation: no_sync backward + sync backward + step must
    match the oracle that sums both micro-grads."""
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    model = LlamaForCausalLM(_cfg())
    ref = copy.deepcopy(model).float()
    gm = GeminiDDP(copy.deepcopy(model), precision="fp32", chunk_size_m=1)
    opt = GeminiOptimizer(FusedAdam(gm.parameters(), lr=1e-2), gm)
    ref_opt = FusedAdam(ref.parameters(), lr=1e-2)

    torch.manual_seed(50)
    micros = [[torch.randint(0, 128, (2, 16)) for _ in range(world_size)] for _ in range(2)]

    with gm.no_sync():
        opt.backward(gm(input_ids=micros[0][rank], labels=micros[0][rank])["loss"])
    opt.backward(gm(input_ids=micros[1][rank], labels=micros[1][rank])["loss"])
    opt.step()
    opt.zero_grad()

    ref_opt.zero_grad()
    for mb in micros:
        for x in mb:
            (ref(input_ids=x, labels=x)["loss"] / world_size).backward()
    ref_opt.step()

    sd = gm.state_dict()
    ref_sd = ref.state_dict()
    for k, v in sd.items():
        assert torch.allclose(v.float(), ref_sd[k].float(), atol=1e-5, rtol=1e-5), f"{k} diverged"
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_gemini_no_sync():
    spawn(run_gemini_no_sync, 2)


def run_gemini_fp16(rank, world_size, port):
    """fp16 Gemini: scaled backward, overflow skip (scale shrinks, params
    unchanged), then a normal step makes progress."""
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    model = LlamaForCausalLM(_cfg())
    gm = GeminiDDP(copy.deepcopy(model), precision="fp16", chunk_size_m=1)
    opt = GeminiOptimizer(FusedAdam(gm.parameters(), lr=1e-3), gm, initial_scale=2.0**8, hysteresis=1)
    assert opt.mixin is not None
    x = torch.randint(0, 128, (2, 16))

    before = gm.state_dict()
    # force an overflow: backward, then poison one grad shard
    out = gm(input_ids=x, labels=x)
    opt.backward(out["loss"])
    for c in gm.chunks:
        if c.grad_shard is not None:
            c.grad_shard[0] = float("inf")
            break
    scale0 = float(opt.mixin.loss_scale)
    opt.step()
    opt.zero_grad()
    assert float(opt.mixin.loss_scale) < scale0, "scale did not back off"
    after = gm.state_dict()
    for k in before:
        assert torch.equal(before[k], after[k]), f"{k} changed on a skipped step"

    # normal steps still learn
    losses = []
    for _ in range(3):
        out = gm(input_ids=x, labels=x)
        opt.backward(out["loss"])
        opt.step()
        opt.zero_grad()
        losses.append(float(out["loss"]))
    assert losses[-1] < losses[0], losses
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_gemini_fp16_scaler():
    spawn(run_gemini_fp16, 2)
