"""Layout conversion between sharding specs (reference:
colossalai/tensor/d_tensor/layout_converter.py) — world-2 CPU/gloo."""

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd.tensor.d_tensor import (DTensorSpec, comm_bytes, distribute_tensor,
                                            gather_distributed, redistribute, shard_rowwise)
from colossalai_amd.testing import rerun_if_address_is_in_use, spawn


def _run(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    W = dist.group.WORLD
    torch.manual_seed(0)
    full = torch.randn(8, 12)

    # row shard -> col shard: same-group shard-dim move (single a2a path)
    row = shard_rowwise(full.clone(), W)
    col_spec = DTensorSpec(dims={1: W}, global_shape=full.shape)
    col = redistribute(row, col_spec)
    expect = full.chunk(world_size, dim=1)[rank]
    torch.testing.assert_close(col, expect)

    # back again
    row2 = redistribute(col, DTensorSpec(dims={0: W}, global_shape=full.shape))
    torch.testing.assert_close(row2, full.chunk(world_size, dim=0)[rank])

    # shard -> replicated (gather)
    rep = redistribute(row2, DTensorSpec(dims={}, global_shape=full.shape))
    torch.testing.assert_close(rep, full)

    # replicated-stamped -> 2D shard via distribute + gather roundtrip
    both = distribute_tensor(full.clone(), DTensorSpec(dims={0: W}, global_shape=full.shape))
    torch.testing.assert_close(gather_distributed(both), full)

    # cost model: a2a move is cheaper than gather+split
    a2a_cost = comm_bytes(DTensorSpec(dims={0: W}, global_shape=full.shape),
                          DTensorSpec(dims={1: W}, global_shape=full.shape))
    gather_cost = comm_bytes(DTensorSpec(dims={0: W}, global_shape=full.shape),
                             DTensorSpec(dims={}, global_shape=full.shape))
    assert a2a_cost < gather_cost
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_redistribute_world2():
    spawn(_run, 2)
