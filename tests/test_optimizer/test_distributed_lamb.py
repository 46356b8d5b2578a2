"""DistributedLamb on a tp2-sharded linear must match single-process Lamb
on the unsharded weight exactly (the trust ratio sees the global norms)."""

import copy

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd.nn import DistributedLamb, Lamb
from colossalai_amd.shardformer import ShardConfig, ShardFormer
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def _run(rank, world_size, port):
    from colossalai_amd.models.gpt2 import GPT2Config, GPT2LMHeadModel

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = GPT2Config(vocab_size=256, n_positions=64, n_embd=64, n_layer=2, n_head=4)
    ref = GPT2LMHeadModel(cfg)
    model = copy.deepcopy(ref)
    model, _ = ShardFormer(ShardConfig(tensor_parallel_process_group=dist.group.WORLD)).optimize(model)

    opt = DistributedLamb(model.parameters(), lr=1e-2)
    opt.setup_distributed(tp_group=dist.group.WORLD)
    ref_opt = Lamb(ref.parameters(), lr=1e-2)

    x = torch.randint(0, 256, (2, 16))
    for _ in range(3):
        out = model(x, labels=x)
        out_ref = ref(x, labels=x)
        opt.zero_grad(); ref_opt.zero_grad()
        out["loss"].backward()
        out_ref["loss"].backward()
        opt.step(); ref_opt.step()

    # sharded weight must equal the matching slice of the reference weight
    lay = model.transformer.layers[0].attn.c_attn
    full_ref = ref.transformer.layers[0].attn.c_attn.weight
    got = lay.gather_weight()
    assert_close_loose(got, full_ref, rtol=1e-4, atol=1e-5)
    # replicated norm weights also match
    assert_close_loose(model.transformer.layers[0].ln_1_weight,
                       ref.transformer.layers[0].ln_1_weight, rtol=1e-4, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_distributed_lamb_tp2():
    spawn(_run, 2)
