import torch
import torch.distributed as dist
import torch.nn as nn
from torch.optim import SGD

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import TorchDDPPlugin
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


class TinyModel(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(8, 16)
        self.fc2 = nn.Linear(16, 4)

    def forward(self, x):
        return self.fc2(torch.relu(self.fc1(x)))


def run_ddp(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(42)  # same init on all ranks

    model = TinyModel()
    optimizer = SGD(model.parameters(), lr=0.1)
    criterion = nn.MSELoss()
    plugin = TorchDDPPlugin()
    booster = Booster(plugin=plugin)
    model, optimizer, criterion, _, _ = booster.boost(model, optimizer, criterion)

    torch.manual_seed(rank)  # different data per rank
    x = torch.randn(4, 8)
    y = torch.randn(4, 4)
    loss = criterion(model(x), y)
    booster.backward(loss, optimizer)

    # grads must be identical across ranks after DDP all-reduce
    for p in model.unwrap().parameters():
        g = p.grad.clone()
        g_list = [torch.empty_like(g) for _ in range(world_size)]
        dist.all_gather(g_list, g)
        for other in g_list:
            assert_close_loose(g, other, rtol=1e-5, atol=1e-5)

    optimizer.step()
    # params identical across ranks after step
    for p in model.unwrap().parameters():
        p_list = [torch.empty_like(p) for _ in range(world_size)]
        dist.all_gather(p_list, p.detach())
        for other in p_list:
            assert_close_loose(p.detach(), other, rtol=1e-6, atol=1e-6)

    # no_sync context works
    with booster.no_sync(model):
        loss = criterion(model(x), y)
        booster.backward(loss, optimizer)

    dist.destroy_process_group()


def run_ckpt(rank, world_size, port, tmp_path: str):
    import os

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(42)
    model = TinyModel()
    optimizer = SGD(model.parameters(), lr=0.1)
    plugin = TorchDDPPlugin()
    booster = Booster(plugin=plugin)
    model, optimizer, *_ = booster.boost(model, optimizer)

    x = torch.randn(4, 8)
    loss = model(x).sum()
    booster.backward(loss, optimizer)
    optimizer.step()

    model_path = os.path.join(tmp_path, "model.safetensors")
    optim_path = os.path.join(tmp_path, "optim.bin")
    booster.save_model(model, model_path, use_safetensors=True)
    booster.save_optimizer(optimizer, optim_path)
    dist.barrier()

    model2 = TinyModel()
    optimizer2 = SGD(model2.parameters(), lr=0.1)
    booster2 = Booster(plugin=TorchDDPPlugin())
    model2, optimizer2, *_ = booster2.boost(model2, optimizer2)
    booster2.load_model(model2, model_path)
    booster2.load_optimizer(optimizer2, optim_path)

    from colossalai_amd.testing import check_state_dict_equal

    check_state_dict_equal(model.unwrap().state_dict(), model2.unwrap().state_dict())
    check_state_dict_equal(optimizer.state_dict(), optimizer2.state_dict())

    # sharded save/load round-trip
    shard_dir = os.path.join(tmp_path, "sharded")
    booster.save_model(model, shard_dir, shard=True, size_per_shard=1, use_safetensors=True)
    dist.barrier()
    model3 = TinyModel()
    booster2.load_model(colossalai_amd.Booster(plugin=TorchDDPPlugin()).boost(model3)[0], shard_dir)
    check_state_dict_equal(model.unwrap().state_dict(), model3.state_dict())

    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_torch_ddp_plugin():
    spawn(run_ddp, 2)


@rerun_if_address_is_in_use()
def test_torch_ddp_checkpoint(tmp_path):
    spawn(run_ckpt, 2, tmp_path=str(tmp_path))
