"""Expert-parallel Mixtral vs unsharded oracle (CPU/gloo, ep=2)."""

import copy

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import MoeHybridParallelPlugin
from colossalai_amd.models.mixtral import MIXTRAL_CONFIGS, MixtralForCausalLM
from colossalai_amd.nn import FusedAdam
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def run_ep(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = MIXTRAL_CONFIGS["mixtral-tiny"]
    ref = MixtralForCausalLM(cfg)
    model = copy.deepcopy(ref)
    ref_opt = torch.optim.AdamW(ref.parameters(), lr=1e-2, weight_decay=0.1, eps=1e-8)

    plugin = MoeHybridParallelPlugin(ep_size=2, tp_size=1, pp_size=1, precision="fp32", zero_stage=1,
                                     overlap_communication=False)
    booster = Booster(plugin=plugin)
    optimizer = FusedAdam(model.parameters(), lr=1e-2, weight_decay=0.1)
    model_b, optimizer_b, *_ = booster.boost(model, optimizer)

    # expert slicing happened
    blk = model_b.module.model.layers[0].mlp
    assert blk.num_local_experts == cfg.num_local_experts // 2
    assert blk.ep_size == 2

    torch.manual_seed(7)
    x = torch.randint(0, 128, (2, 16))  # same batch on both ranks
    for it in range(2):
        out = model_b(input_ids=x, labels=x)
        out_ref = ref(x, labels=x)
        assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
        optimizer_b.backward(out["loss"])
        optimizer_b.step()
        out_ref["loss"].backward()
        ref_opt.step()
        ref_opt.zero_grad()

    # post-step params match the oracle: dense ...
    assert_close_loose(model_b.module.model.layers[0].input_layernorm_weight.detach(),
                       ref.model.layers[0].input_layernorm_weight.detach(), rtol=1e-4, atol=1e-5)
    # ... and this rank's expert slice
    lo = blk.expert_start
    assert_close_loose(blk.w_gate_up.detach(),
                       ref.model.layers[0].mlp.w_gate_up.detach()[lo : lo + blk.num_local_experts],
                       rtol=1e-4, atol=1e-5)
    dist.destroy_process_group()


def test_mixtral_single():
    torch.manual_seed(0)
    cfg = MIXTRAL_CONFIGS["mixtral-tiny"]
    m = MixtralForCausalLM(cfg)
    x = torch.randint(0, 128, (2, 16))
    out = m(x, labels=x)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n


@rerun_if_address_is_in_use()
def test_mixtral_ep2():
    spawn(run_ep, 2)


def run_deepseek_ep(rank, world_size, port):
    """DeepSeek-MoE: EP-sharded must equal the dense single-rank oracle."""
    import copy

    from colossalai_amd import Booster
    from colossalai_amd.booster.plugin import MoeHybridParallelPlugin
    from colossalai_amd.models.deepseek import DEEPSEEK_CONFIGS, DeepseekForCausalLM
    from colossalai_amd.nn import FusedAdam

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    model = DeepseekForCausalLM(DEEPSEEK_CONFIGS["deepseek-tiny"])
    ref = copy.deepcopy(model).float()

    plugin = MoeHybridParallelPlugin(ep_size=2, zero_stage=1, precision="fp32",
                                     overlap_communication=False)
    booster = Booster(plugin=plugin)
    opt = FusedAdam(model.parameters(), lr=1e-3)
    model_b, opt_b, *_ = booster.boost(model, opt)

    # each MoE block holds only its expert slice
    moe = model_b.module.model.layers[-1].mlp
    assert moe.num_local_experts == 2 and moe.ep_size == 2

    torch.manual_seed(9)
    xs = [torch.randint(0, 128, (2, 16)) for _ in range(world_size)]
    out = model_b(input_ids=xs[rank], labels=xs[rank])
    losses_ref = [ref(input_ids=x, labels=x)["loss"] for x in xs]
    assert_close_loose(out["loss"], losses_ref[rank], rtol=1e-4, atol=1e-5)

    opt_b.backward(out["loss"])
    opt_b.step()
    opt_b.zero_grad()
    out2 = model_b(input_ids=xs[rank], labels=xs[rank])
    assert torch.isfinite(out2["loss"])
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_deepseek_ep2():
    spawn(run_deepseek_ep, 2)


def run_deepseek_v3_ep(rank, world_size, port):
    """DeepSeek-V3: MLA + noaux-tc routing, EP-sharded vs dense oracle."""
    import copy

    from colossalai_amd import Booster
    from colossalai_amd.booster.plugin import MoeHybridParallelPlugin
    from colossalai_amd.models.deepseek_v3 import DEEPSEEK_V3_CONFIGS, DeepseekV3ForCausalLM
    from colossalai_amd.nn import FusedAdam

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    model = DeepseekV3ForCausalLM(DEEPSEEK_V3_CONFIGS["deepseek-v3-tiny"])
    ref = copy.deepcopy(model).float()

    plugin = MoeHybridParallelPlugin(ep_size=2, zero_stage=1, precision="fp32",
                                     overlap_communication=False)
    booster = Booster(plugin=plugin)
    opt = FusedAdam(model.parameters(), lr=1e-3)
    model_b, opt_b, *_ = booster.boost(model, opt)

    moe = model_b.module.model.layers[-1].mlp
    assert moe.num_local_experts == 4 and moe.ep_size == 2

    torch.manual_seed(9)
    xs = [torch.randint(0, 128, (2, 16)) for _ in range(world_size)]
    out = model_b(input_ids=xs[rank], labels=xs[rank])
    losses_ref = [ref(input_ids=x, labels=x)["loss"] for x in xs]
    assert_close_loose(out["loss"], losses_ref[rank], rtol=1e-4, atol=1e-5)

    opt_b.backward(out["loss"])
    opt_b.step()
    opt_b.zero_grad()
    out2 = model_b(input_ids=xs[rank], labels=xs[rank])
    assert torch.isfinite(out2["loss"])
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_deepseek_v3_ep2():
    spawn(run_deepseek_v3_ep, 2)
