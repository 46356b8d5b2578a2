"""Ulysses (all_to_all) sequence parallelism vs unsharded oracle (CPU/gloo, sp=2)."""

import copy

import torch
import torch.distributed as dist
import torch.nn.functional as F

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import HybridParallelPlugin
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def run_sp(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    ref = LlamaForCausalLM(cfg)
    model = copy.deepcopy(ref)

    plugin = HybridParallelPlugin(
        tp_size=1, pp_size=1, sp_size=2, precision="fp32", zero_stage=0,
        enable_sequence_parallelism=True, sequence_parallelism_mode="all_to_all",
    )
    booster = Booster(plugin=plugin)
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3)
    model_b, optimizer_b, *_ = booster.boost(model, optimizer)

    torch.manual_seed(7)
    S = 16
    x = torch.randint(0, 128, (2, S))
    out = model_b(input_ids=x, labels=x)

    # reference: same per-shard loss composition (mean of local means)
    logits_ref = ref(x)["logits"]
    shifted = torch.full_like(x, -100)
    shifted[:, :-1] = x[:, 1:]
    shard = S // world_size
    local_losses = []
    for r in range(world_size):
        lg = logits_ref[:, r * shard : (r + 1) * shard]
        lb = shifted[:, r * shard : (r + 1) * shard]
        local_losses.append(F.cross_entropy(lg.reshape(-1, 128).float(), lb.reshape(-1), ignore_index=-100))
    ref_loss = sum(local_losses) / world_size

    # this rank's loss must equal its shard's reference loss
    assert_close_loose(out["loss"], local_losses[rank], rtol=1e-4, atol=1e-5)

    # backward + sp grad sync = gradient of the averaged loss
    booster.backward(out["loss"], optimizer_b)
    ref_loss.backward()
    g = model_b.module.model.layers[0].input_layernorm_weight.grad
    assert_close_loose(g, ref.model.layers[0].input_layernorm_weight.grad, rtol=1e-3, atol=1e-5)
    g2 = model_b.module.model.layers[1].self_attn.qkv_proj.weight.grad
    assert_close_loose(g2, ref.model.layers[1].self_attn.qkv_proj.weight.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_sp_ulysses():
    spawn(run_sp, 2)


def run_tp_sp(rank, world_size, port):
    """tp2 x sp2 (all_to_all) world 4 vs unsharded oracle."""
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=4, max_position_embeddings=64)
    ref = LlamaForCausalLM(cfg)
    model = copy.deepcopy(ref)

    plugin = HybridParallelPlugin(
        tp_size=2, pp_size=1, sp_size=2, precision="fp32", zero_stage=0,
        enable_sequence_parallelism=True, sequence_parallelism_mode="all_to_all",
    )
    booster = Booster(plugin=plugin)
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3)
    model_b, optimizer_b, *_ = booster.boost(model, optimizer)

    torch.manual_seed(7)
    S = 16
    x = torch.randint(0, 128, (2, S))
    out = model_b(input_ids=x, labels=x)
    assert out["loss"] is not None and torch.isfinite(out["loss"])

    # oracle: per-sp-shard loss of the full model
    import torch.nn.functional as F

    sp_rank = torch.distributed.get_rank(plugin.sp_group)
    logits_ref = ref(x)["logits"]
    shifted = torch.full_like(x, -100)
    shifted[:, :-1] = x[:, 1:]
    shard = S // 2
    lg = logits_ref[:, sp_rank * shard : (sp_rank + 1) * shard]
    lb = shifted[:, sp_rank * shard : (sp_rank + 1) * shard]
    local_ref = F.cross_entropy(lg.reshape(-1, 128).float(), lb.reshape(-1), ignore_index=-100)
    assert_close_loose(out["loss"], local_ref, rtol=1e-4, atol=1e-5)

    booster.backward(out["loss"], optimizer_b)
    optimizer_b.step()
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp2_sp2_ulysses():
    spawn(run_tp_sp, 4)
