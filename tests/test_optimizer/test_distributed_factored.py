"""DistributedAdafactor / DistributedCAME at tp2 vs their single-rank
originals on the full (unsharded) weight (reference:
tests/test_optimizer/test_dist_adafactor.py idiom)."""

import copy

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd.nn.optimizer import (CAME, Adafactor, DistributedAdafactor, DistributedCAME,
                                         cast_to_distributed)
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def _run_one(rank, world, kind, tp_dim):
    torch.manual_seed(0)
    full = torch.randn(8, 12)
    grad_full = torch.randn(8, 12)

    # single-rank oracle
    ref = full.clone().requires_grad_(True)
    opt_cls = {"adafactor": Adafactor, "came": CAME}[kind]
    ref_opt = opt_cls([ref], lr=1e-2)
    for _ in range(3):
        ref.grad = grad_full.clone()
        ref_opt.step()

    # tp2 shard
    shard = full.chunk(world, dim=tp_dim)[rank].contiguous().requires_grad_(True)
    shard.tp_sharded = True
    shard.tp_dim = tp_dim
    d_opt = cast_to_distributed(opt_cls([shard], lr=1e-2))
    assert type(d_opt) in (DistributedAdafactor, DistributedCAME)
    d_opt.setup_distributed(tp_group=dist.group.WORLD)
    for _ in range(3):
        shard.grad = grad_full.chunk(world, dim=tp_dim)[rank].contiguous()
        d_opt.step()

    gathered = [torch.empty_like(shard.data) for _ in range(world)]
    dist.all_gather(gathered, shard.data, group=dist.group.WORLD)
    assert_close_loose(torch.cat(gathered, dim=tp_dim), ref.data, rtol=1e-5, atol=1e-6)


def _run(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    for kind in ("adafactor", "came"):
        for tp_dim in (0, 1):
            _run_one(rank, world_size, kind, tp_dim)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_distributed_factored_tp2():
    spawn(_run, 2)
