import torch.distributed as dist

import colossalai_amd
from colossalai_amd.cluster import DistCoordinator, ProcessGroupMesh
from colossalai_amd.testing import rerun_if_address_is_in_use, spawn


def check_mesh(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    # 2x2 mesh on 4 ranks: axes (dp, tp)
    mesh = ProcessGroupMesh(2, 2)
    assert mesh.shape == (2, 2)
    assert mesh.rank == rank
    coord = mesh.coordinate()
    assert ProcessGroupMesh.ravel(coord, mesh.shape) == rank

    tp_group = mesh.get_group_along_axis(1)
    dp_group = mesh.get_group_along_axis(0)
    assert dist.get_world_size(tp_group) == 2
    assert dist.get_world_size(dp_group) == 2
    # tp group ranks share the dp coordinate
    tp_ranks = mesh.get_ranks_in_group(tp_group)
    expected_tp = [rank // 2 * 2, rank // 2 * 2 + 1]
    assert tp_ranks == expected_tp, (tp_ranks, expected_tp)
    dp_ranks = mesh.get_ranks_in_group(dp_group)
    assert dp_ranks == [rank % 2, rank % 2 + 2]

    # flattened group over both axes = whole world
    full_group = mesh.get_group_along_axis([0, 1])
    assert dist.get_world_size(full_group) == 4

    # an all-reduce over the tp group only sums within the group
    import torch

    t = torch.tensor([float(rank)])
    dist.all_reduce(t, group=tp_group)
    assert t.item() == sum(expected_tp)

    coordinator = DistCoordinator()
    assert coordinator.world_size == world_size
    assert coordinator.is_master() == (rank == 0)
    dist.destroy_process_group()


def check_coordinator_priority(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    coordinator = DistCoordinator()
    with coordinator.priority_execution():
        pass
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_process_group_mesh():
    spawn(check_mesh, 4)


@rerun_if_address_is_in_use()
def test_coordinator():
    spawn(check_coordinator_priority, 2)
