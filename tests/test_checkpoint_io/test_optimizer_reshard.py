"""Topology-independent optimizer checkpoints: save under one (dp, tp, pp)
topology, resume under another, and match a single-process reference exactly
(reference: colossalai/checkpoint_io/hybrid_parallel_checkpoint_io.py:469,
1017 gather_from_sharded_optimizer_state, 1082 shard_from_complete_...)."""

import copy
import os

import pytest
import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import HybridParallelPlugin
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
from colossalai_amd.nn import FusedAdam
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def _tiny():
    return LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                       num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)


def _data(seed):
    g = torch.Generator().manual_seed(seed)
    return torch.randint(0, 128, (4, 16), generator=g)


def _fresh_model():
    torch.manual_seed(0)
    return LlamaForCausalLM(_tiny())


def _boost(tp, pp, zero, world_batch=None):
    plugin = HybridParallelPlugin(tp_size=tp, pp_size=pp, precision="fp32", zero_stage=zero,
                                  overlap_communication=False,
                                  num_microbatches=2 if pp > 1 else None)
    booster = Booster(plugin=plugin)
    model = _fresh_model()
    optimizer = FusedAdam(model.parameters(), lr=1e-2)
    criterion = lambda out, micro: out["loss"]
    model_b, optimizer_b, criterion, _, _ = booster.boost(model, optimizer, criterion)
    return plugin, booster, model_b, optimizer_b, criterion


def _train_step(plugin, booster, model_b, optimizer_b, criterion, x):
    if plugin.pp_size > 1:
        booster.execute_pipeline(iter([{"input_ids": x, "labels": x}]), model_b, criterion,
                                 optimizer_b, return_loss=True)
    else:
        out = model_b(input_ids=x, labels=x)
        booster.backward(out["loss"], optimizer_b)
    optimizer_b.step()
    optimizer_b.zero_grad()


def _run_save(rank, world_size, port, tmpdir=None, tp=1, pp=1, zero=0, shard=False):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    plugin, booster, model_b, optimizer_b, criterion = _boost(tp, pp, zero)
    _train_step(plugin, booster, model_b, optimizer_b, criterion, _data(1))
    booster.save_model(model_b, os.path.join(tmpdir, "model.bin"))
    booster.save_optimizer(optimizer_b, os.path.join(tmpdir, "optim"), shard=shard)
    dist.destroy_process_group()


def _run_resume(rank, world_size, port, tmpdir=None, tp=1, pp=1, zero=0, out_file="final.bin"):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    plugin, booster, model_b, optimizer_b, criterion = _boost(tp, pp, zero)
    booster.load_model(model_b, os.path.join(tmpdir, "model.bin"))
    if hasattr(optimizer_b, "update_master_params"):
        optimizer_b.update_master_params(model_b.unwrap())
    booster.load_optimizer(optimizer_b, os.path.join(tmpdir, "optim"))
    _train_step(plugin, booster, model_b, optimizer_b, criterion, _data(2))
    # write the full post-step model state from rank 0 (TP-gathered)
    booster.save_model(model_b, os.path.join(tmpdir, out_file))
    dist.destroy_process_group()


def _reference_final():
    ref = _fresh_model()
    opt = FusedAdam(ref.parameters(), lr=1e-2)
    for seed in (1, 2):
        x = _data(seed)
        ref(x, labels=x)["loss"].backward()
        opt.step()
        opt.zero_grad()
    return ref


@pytest.mark.parametrize("save_topo,load_topo,shard", [
    ((1, 1, 1), (1, 1, 0), False),   # dp2+zero1  -> single process
    ((1, 1, 1), (1, 1, 1), False),   # dp2+zero1  -> dp2+zero1 (same)
    ((2, 1, 0), (1, 1, 0), False),   # tp2        -> single process
    ((1, 1, 0), (2, 1, 0), False),   # single     -> tp2 (reshard on load)
    ((2, 1, 1), (1, 1, 1), False),   # tp2 x dp2 zero1 -> dp2 zero1
    ((1, 1, 1), (1, 1, 0), True),    # sharded dir format -> single process
])
def test_optimizer_reshard(tmp_path, save_topo, load_topo, shard):
    tmpdir = str(tmp_path)
    stp, spp, szero = save_topo
    ltp, lpp, lzero = load_topo
    save_world = stp * spp * (2 if szero else 1)
    load_world = ltp * lpp * (2 if lzero else 1)

    _spawn_seq(_run_save, save_world, tmpdir=tmpdir, tp=stp, pp=spp, zero=szero, shard=shard)
    _spawn_seq(_run_resume, load_world, tmpdir=tmpdir, tp=ltp, pp=lpp, zero=lzero, out_file="final.bin")

    ref = _reference_final()
    final = torch.load(os.path.join(tmpdir, "final.bin"), weights_only=False)
    ref_sd = {k: v for k, v in ref.state_dict().items()}
    for k, v in ref_sd.items():
        assert k in final, f"missing {k} in resumed state dict"
        assert_close_loose(final[k].float(), v.float(), rtol=2e-4, atol=2e-5)


def _run_pp_save(rank, world_size, port, tmpdir=None):
    _run_save(rank, world_size, port, tmpdir=tmpdir, tp=1, pp=2, zero=0, shard=False)


def _run_pp_resume(rank, world_size, port, tmpdir=None):
    _run_resume(rank, world_size, port, tmpdir=tmpdir, tp=1, pp=1, zero=0, out_file="final.bin")


def test_optimizer_reshard_pp2_to_single(tmp_path):
    """pp2 save -> single-process resume: stage-disjoint states merge."""
    tmpdir = str(tmp_path)
    _spawn_seq(_run_pp_save, 2, tmpdir=tmpdir)
    _spawn_seq(_run_pp_resume, 1, tmpdir=tmpdir)
    ref = _reference_final()
    final = torch.load(os.path.join(tmpdir, "final.bin"), weights_only=False)
    for k, v in ref.state_dict().items():
        assert_close_loose(final[k].float(), v.float(), rtol=2e-4, atol=2e-5)


@rerun_if_address_is_in_use()
def _spawn_seq(fn, nprocs, **kwargs):
    spawn(fn, nprocs, **kwargs)


if __name__ == "__main__":
    import tempfile
    with tempfile.TemporaryDirectory() as d:
        from pathlib import Path
        test_optimizer_reshard(Path(d), (2, 1, 1), (1, 1, 0), False)
