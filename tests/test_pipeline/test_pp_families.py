"""pp2 pipeline forwards beyond the Llama family (GPT-2, GPT-J, BLOOM) vs
unsharded oracles (CPU/gloo). Tied embed/lm_head weights sync over the
embed group in every family."""

import copy

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import HybridParallelPlugin
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def _pp2_roundtrip(model, ref, x, loss_tol=1e-5):
    plugin = HybridParallelPlugin(tp_size=1, pp_size=2, precision="fp32",
                                  num_microbatches=2, zero_stage=0)
    booster = Booster(plugin=plugin)
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3)
    criterion = lambda out, micro: out["loss"]
    model_b, optimizer_b, criterion, _, _ = booster.boost(model, optimizer, criterion)

    batch = {"input_ids": x, "labels": x}
    result = booster.execute_pipeline(iter([batch]), model_b, criterion, optimizer_b, return_loss=True)

    out_ref = ref(x, labels=x)
    out_ref["loss"].backward()
    if plugin.stage_manager.is_last_stage():
        assert result["loss"] is not None
        assert_close_loose(result["loss"], out_ref["loss"], rtol=1e-4, atol=loss_tol)
    optimizer_b.step()
    optimizer_b.zero_grad()
    return plugin, model_b


def run_pp_gpt2(rank, world_size, port):
    from colossalai_amd.models.gpt2 import GPT2Config, GPT2LMHeadModel

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = GPT2Config(vocab_size=256, n_embd=64, n_layer=4, n_head=4, n_positions=64)
    ref = GPT2LMHeadModel(cfg)
    _pp2_roundtrip(copy.deepcopy(ref), ref, torch.randint(0, 256, (4, 16)))
    dist.destroy_process_group()


def run_pp_gptj(rank, world_size, port):
    from colossalai_amd.models.gptj import GPTJConfig, GPTJForCausalLM

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = GPTJConfig(vocab_size=256, n_embd=64, n_layer=4, n_head=4, rotary_dim=8, n_positions=64)
    ref = GPTJForCausalLM(cfg)
    _pp2_roundtrip(copy.deepcopy(ref), ref, torch.randint(0, 256, (4, 16)))
    dist.destroy_process_group()


def run_pp_bloom(rank, world_size, port):
    from colossalai_amd.models.bloom import BloomConfig, BloomForCausalLM

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = BloomConfig(vocab_size=256, hidden_size=64, num_hidden_layers=4, num_attention_heads=4)
    ref = BloomForCausalLM(cfg)
    _pp2_roundtrip(copy.deepcopy(ref), ref, torch.randint(0, 256, (4, 16)))
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_pp2_gpt2():
    spawn(run_pp_gpt2, 2)


@rerun_if_address_is_in_use()
def test_pp2_gptj():
    spawn(run_pp_gptj, 2)


@rerun_if_address_is_in_use()
def test_pp2_bloom():
    spawn(run_pp_bloom, 2)
