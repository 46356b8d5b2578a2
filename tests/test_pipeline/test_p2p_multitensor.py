"""Multi-tensor stage IO round-trip through PipelineP2PCommunication
(reference: colossalai/pipeline/p2p.py:364 batched arbitrary-object P2P)."""

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd.cluster import ProcessGroupMesh
from colossalai_amd.pipeline import PipelineStageManager
from colossalai_amd.pipeline.p2p import PipelineP2PCommunication
from colossalai_amd.testing import rerun_if_address_is_in_use, spawn


def _run(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    mesh = ProcessGroupMesh(1, 2, 1, 1)
    sm = PipelineStageManager(mesh, 1)
    comm = PipelineP2PCommunication(sm)

    if sm.is_first_stage():
        # forward direction carries a multi-tensor payload (structure is
        # constant per direction — the metadata caches after microbatch 1)
        for _ in range(2):
            comm.send_forward([torch.arange(6).reshape(2, 3).float(),
                               torch.tensor([7, 8], dtype=torch.int64),
                               torch.ones(1, 2, 2, dtype=torch.bfloat16)])
        comm.flush_sends()
        single = comm.recv_backward()
        assert isinstance(single, torch.Tensor) and torch.equal(single, torch.full((2, 8), 3.0))
    else:
        for _ in range(2):
            multi = comm.recv_forward()
            assert isinstance(multi, list) and len(multi) == 3
            assert torch.equal(multi[0], torch.arange(6).reshape(2, 3).float())
            assert multi[1].dtype == torch.int64 and multi[1].tolist() == [7, 8]
            assert multi[2].dtype == torch.bfloat16 and multi[2].shape == (1, 2, 2)
        # backward direction keeps the single-tensor contract
        comm.send_backward(torch.full((2, 8), 3.0))
        comm.flush_sends()
    dist.barrier()
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_p2p_multi_tensor():
    spawn(_run, 2)
