"""d_tensor API + DeviceMesh/alpha-beta profiler (gloo world 2)."""

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd.device import AlphaBetaProfiler, DeviceMesh
from colossalai_amd.tensor import (
    DTensorSpec,
    distribute_tensor,
    gather_distributed,
    get_sharding_spec,
    is_distributed_tensor,
    shard_colwise,
    shard_rowwise,
)
from colossalai_amd.testing import rerun_if_address_is_in_use, spawn


def _run(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    g = dist.group.WORLD
    torch.manual_seed(0)
    full = torch.randn(8, 6)

    r = shard_rowwise(full.clone(), g)
    assert r.shape == (4, 6) and is_distributed_tensor(r)
    assert torch.equal(r, full.chunk(2, 0)[rank])
    assert torch.equal(gather_distributed(r), full)

    c = shard_colwise(full.clone(), g)
    assert c.shape == (8, 3)
    assert torch.equal(gather_distributed(c), full)

    d = distribute_tensor(full.clone(), DTensorSpec(dims={1: g}))
    assert torch.equal(d, c) and get_sharding_spec(d).global_shape == full.shape

    # mesh: 2x1, axis-0 group == world
    mesh = DeviceMesh((2, 1))
    pg = mesh.get_process_group(0)
    assert dist.get_world_size(pg) == 2
    # cost model sanity: monotonic in bytes, zero on singleton axis
    assert mesh.all_reduce_cost(1 << 20, 0) < mesh.all_reduce_cost(1 << 24, 0)
    assert mesh.all_reduce_cost(1 << 20, 1) == 0.0
    assert mesh.all_to_all_cost(1 << 20, 0) < mesh.all_reduce_cost(1 << 20, 0)

    alpha, beta = AlphaBetaProfiler(pg).profile(small=1024, large=1 << 20)
    assert alpha > 0 and beta > 0
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_dtensor_and_device_mesh():
    spawn(_run, 2)
