"""Hybrid (tp2[+pp2]) checkpoint save → single-process load round trip (CPU/gloo)."""

import copy
import os

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import HybridParallelPlugin
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
from colossalai_amd.testing import check_state_dict_equal, rerun_if_address_is_in_use, spawn


def _tiny():
    return LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=4,
                       num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)


def _run(rank, world_size, port, tmp_path, tp_size, pp_size):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    ref = LlamaForCausalLM(_tiny())
    model = copy.deepcopy(ref)
    plugin = HybridParallelPlugin(tp_size=tp_size, pp_size=pp_size, precision="fp32", zero_stage=0,
                                  num_microbatches=2 if pp_size > 1 else None)
    booster = Booster(plugin=plugin)
    model_b, *_ = booster.boost(model)

    path = os.path.join(tmp_path, "model.bin")
    booster.save_model(model_b, path)
    dist.barrier()

    # full (unsharded) reload must equal the original single-process weights
    if rank == 0:
        loaded = torch.load(path, map_location="cpu", weights_only=False)
        ref_sd = {k: v for k, v in ref.state_dict().items()}
        check_state_dict_equal(loaded, ref_sd)

    # sharded reload: perturb local weights, reload, re-save, compare again
    with torch.no_grad():
        for p in model_b.module.parameters():
            p.add_(1.0)
    booster.load_model(model_b, path)
    path2 = os.path.join(tmp_path, "model2.bin")
    booster.save_model(model_b, path2)
    dist.barrier()
    if rank == 0:
        loaded2 = torch.load(path2, map_location="cpu", weights_only=False)
        check_state_dict_equal(loaded2, {k: v for k, v in ref.state_dict().items()})
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp2_ckpt(tmp_path):
    spawn(_run, 2, tmp_path=str(tmp_path), tp_size=2, pp_size=1)


@rerun_if_address_is_in_use()
def test_tp2_pp2_ckpt(tmp_path):
    spawn(_run, 4, tmp_path=str(tmp_path), tp_size=2, pp_size=2)
