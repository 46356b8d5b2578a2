"""HybridParallelPlugin pp2(+tp2) vs unsharded oracle on CPU/gloo."""

import copy

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import HybridParallelPlugin
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def _tiny():
    return LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=4,
                       num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)


def _run(rank, world_size, port, tp_size, pp_size):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = _tiny()
    ref = LlamaForCausalLM(cfg)
    model = copy.deepcopy(ref)

    plugin = HybridParallelPlugin(tp_size=tp_size, pp_size=pp_size, precision="fp32",
                                  num_microbatches=2, zero_stage=0)
    booster = Booster(plugin=plugin)
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3)
    criterion = lambda out, micro: out["loss"]
    model_b, optimizer_b, criterion, _, _ = booster.boost(model, optimizer, criterion)

    torch.manual_seed(7)
    x = torch.randint(0, 128, (4, 16))
    batch = {"input_ids": x, "labels": x}

    result = booster.execute_pipeline(iter([batch]), model_b, criterion, optimizer_b, return_loss=True)

    # reference
    out_ref = ref(x, labels=x)
    out_ref["loss"].backward()

    sm = plugin.stage_manager
    if sm.is_last_stage():
        assert result["loss"] is not None
        assert_close_loose(result["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)

    # replicated norm-weight grads of locally-held layers match the reference
    start, end = model_b.module.stage_range
    for i in range(start, end):
        g = model_b.module.model.layers[i].input_layernorm_weight.grad
        rg = ref.model.layers[i].input_layernorm_weight.grad
        assert g is not None, f"layer {i} norm grad missing"
        assert_close_loose(g, rg, rtol=1e-3, atol=1e-5)

    # optimizer step must run without error on every rank
    optimizer_b.step()
    optimizer_b.zero_grad()
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_pp2():
    spawn(_run, 2, tp_size=1, pp_size=2)


@rerun_if_address_is_in_use()
def test_pp2_tp2():
    spawn(_run, 4, tp_size=2, pp_size=2)


def _run_z(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = _tiny()
    ref = LlamaForCausalLM(cfg)
    model = copy.deepcopy(ref)
    from colossalai_amd.nn import FusedAdam

    plugin = HybridParallelPlugin(tp_size=2, pp_size=1, precision="fp32", zero_stage=1,
                                  overlap_communication=False)
    booster = Booster(plugin=plugin)
    optimizer = FusedAdam(model.parameters(), lr=1e-3)
    model_b, optimizer_b, *_ = booster.boost(model, optimizer)

    torch.manual_seed(7)
    x = torch.randint(0, 128, (4, 16))
    out = model_b(input_ids=x, labels=x)
    out_ref = ref(x, labels=x)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    booster.backward(out["loss"], optimizer_b)
    optimizer_b.step()
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp2_zero1():
    spawn(_run_z, 2)


if __name__ == "__main__":
    test_pp2()


def _run_bf16_pp(rank, world_size, port):
    """pp2 in bf16: HybridParallelNaiveOptimizer (bf16 master path) through
    the pipeline schedule — finite losses, steps run, params update."""
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny())
    plugin = HybridParallelPlugin(tp_size=1, pp_size=2, precision="bf16",
                                  num_microbatches=2, zero_stage=0)
    booster = Booster(plugin=plugin)
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-2)
    criterion = lambda out, micro: out["loss"]
    model_b, optimizer_b, criterion, _, _ = booster.boost(model, optimizer, criterion)

    x = torch.randint(0, 128, (4, 16))
    batch = {"input_ids": x, "labels": x}
    start, end = model_b.module.stage_range
    before = model_b.module.model.layers[start].input_layernorm_weight.detach().clone()
    losses = []
    for _ in range(2):
        result = booster.execute_pipeline(iter([batch]), model_b, criterion, optimizer_b,
                                          return_loss=True)
        if plugin.stage_manager.is_last_stage():
            assert result["loss"] is not None and torch.isfinite(result["loss"])
            losses.append(float(result["loss"]))
        optimizer_b.step()
        optimizer_b.zero_grad()
    after = model_b.module.model.layers[start].input_layernorm_weight.detach()
    assert not torch.equal(before, after), "params did not update"
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_pp2_bf16():
    spawn(_run_bf16_pp, 2)
