"""TP-sharded native GPT-2 vs unsharded oracle (CPU/gloo, tp=2)."""

import copy

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd.models.gpt2 import GPT2Config, GPT2LMHeadModel
from colossalai_amd.shardformer import ShardConfig, ShardFormer
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def run_tp(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = GPT2Config(vocab_size=256, n_positions=64, n_embd=64, n_layer=2, n_head=4)
    ref = GPT2LMHeadModel(cfg)
    model = copy.deepcopy(ref)
    model, _ = ShardFormer(ShardConfig(tensor_parallel_process_group=dist.group.WORLD)).optimize(model)
    assert model.transformer.layers[0].attn.num_heads == 2

    x = torch.randint(0, 256, (2, 16))
    out = model(x, labels=x)
    out_ref = ref(x, labels=x)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.transformer.layers[0].ln_1_weight.grad,
                       ref.transformer.layers[0].ln_1_weight.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp_gpt2():
    spawn(run_tp, 2)
