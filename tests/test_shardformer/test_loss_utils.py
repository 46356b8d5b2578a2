"""DistLogProb vs dense log_softmax oracle + Randomizer regimes (gloo tp2)."""

import copy

import torch
import torch.distributed as dist
import torch.nn.functional as F

import colossalai_amd
from colossalai_amd.shardformer.layer import Randomizer, dist_log_prob
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def run_logprob(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    N, V = 6, 32
    full = torch.randn(N, V, requires_grad=True)
    labels = torch.randint(0, V, (N,))
    part = V // world_size
    shard = full.detach()[:, rank * part : (rank + 1) * part].clone().requires_grad_(True)

    lp = dist_log_prob(shard, labels, dist.group.WORLD)
    ref = F.log_softmax(full.float(), dim=-1).gather(-1, labels.unsqueeze(-1)).squeeze(-1)
    assert_close_loose(lp, ref, rtol=1e-5, atol=1e-6)

    dout = torch.randn(N)
    lp.backward(dout)
    ref.backward(dout)
    assert_close_loose(shard.grad, full.grad[:, rank * part : (rank + 1) * part], rtol=1e-5, atol=1e-6)

    # Randomizer: same-regime identical across ranks, diff-regime not
    rnd = Randomizer(1234, dist.group.WORLD)
    with rnd.fork_rng():
        same = torch.rand(8)
    with rnd.fork_rng_diff():
        diff = torch.rand(8)
    gathered_same = [torch.empty_like(same) for _ in range(world_size)]
    gathered_diff = [torch.empty_like(diff) for _ in range(world_size)]
    dist.all_gather(gathered_same, same)
    dist.all_gather(gathered_diff, diff)
    assert torch.equal(gathered_same[0], gathered_same[1])
    assert not torch.equal(gathered_diff[0], gathered_diff[1])
    # forked draws must not disturb the global stream
    torch.manual_seed(7)
    a = torch.rand(4)
    torch.manual_seed(7)
    with rnd.fork_rng():
        torch.rand(4)
    b = torch.rand(4)
    assert torch.equal(a, b)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_dist_log_prob_and_randomizer():
    spawn(run_logprob, 2)
