"""EP-sharded Mixtral checkpoint round trip (gloo world 2, ep=2): the saved
file holds FULL expert tensors; reload restores each rank's slice and the
model's outputs."""

import os

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import MoeHybridParallelPlugin
from colossalai_amd.models.mixtral import MIXTRAL_CONFIGS, MixtralForCausalLM
from colossalai_amd.nn import FusedAdam
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def _boost():
    torch.manual_seed(0)
    model = MixtralForCausalLM(MIXTRAL_CONFIGS["mixtral-tiny"])
    plugin = MoeHybridParallelPlugin(ep_size=2, zero_stage=1, precision="fp32",
                                     overlap_communication=False)
    booster = Booster(plugin=plugin)
    opt = FusedAdam(model.parameters(), lr=1e-3)
    model_b, opt_b, *_ = booster.boost(model, opt)
    return booster, model_b, opt_b


def _run(rank, world_size, port, tmp_path):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    booster, model_b, opt_b = _boost()
    x = torch.randint(0, 128, (4, 16))
    for _ in range(2):
        out = model_b(input_ids=x, labels=x)
        opt_b.backward(out["loss"])
        opt_b.step()
        opt_b.zero_grad()
    want_loss = model_b(input_ids=x, labels=x)["loss"]
    mpath = os.path.join(tmp_path, "moe.pt")
    booster.save_model(model_b, mpath)
    dist.barrier()
    if rank == 0:
        sd = torch.load(mpath, weights_only=True)
        # the file must hold the FULL expert dim (4 experts, 2 per rank)
        k = next(kk for kk in sd if kk.endswith("w_gate_up"))
        assert sd[k].shape[0] == 4, sd[k].shape

    booster2, model2, _ = _boost()
    booster2.load_model(model2, mpath)
    got_loss = model2(input_ids=x, labels=x)["loss"]
    assert_close_loose(got_loss, want_loss, rtol=1e-5, atol=1e-6)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_moe_ckpt_roundtrip(tmp_path):
    spawn(_run, 2, tmp_path=str(tmp_path))
