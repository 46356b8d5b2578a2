"""fp16 ZeRO: dynamic loss scaling skips overflowed steps and recovers."""

import torch
import torch.distributed as dist
import torch.nn as nn

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import LowLevelZeroPlugin
from colossalai_amd.nn import FusedAdam
from colossalai_amd.testing import rerun_if_address_is_in_use, spawn


def run_fp16(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 4))
    optimizer = FusedAdam(model.parameters(), lr=1e-3)
    plugin = LowLevelZeroPlugin(stage=2, precision="fp16", initial_scale=2**10,
                               overlap_communication=False)
    booster = Booster(plugin=plugin)
    model, optimizer, *_ = booster.boost(model, optimizer)

    x = torch.randn(8, 16, dtype=torch.float16)
    # normal step: params change
    before = model.unwrap()[0].weight.detach().clone()
    loss = model(x).float().pow(2).mean()
    optimizer.backward(loss)
    scale0 = optimizer.loss_scale
    assert scale0 == 2**10
    optimizer.step()
    after = model.unwrap()[0].weight.detach().clone()
    assert not torch.equal(before, after)

    # poison the grads -> overflow detected, step skipped, scale halves
    loss = model(x).float().pow(2).mean() * float("inf")
    optimizer.backward(loss)
    before = model.unwrap()[0].weight.detach().clone()
    optimizer.step()
    after = model.unwrap()[0].weight.detach().clone()
    assert torch.equal(before, after), "overflowed step must be skipped"
    # hysteresis=2: first overflow decrements; second halves the scale
    loss = model(x).float().pow(2).mean() * float("inf")
    optimizer.backward(loss)
    optimizer.step()
    assert optimizer.loss_scale < scale0
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_zero_fp16_scaler():
    spawn(run_fp16, 2)
