"""SP "ring" mode (ring-pipelined gather/reduce-scatter matmuls) vs oracle (CPU/gloo)."""

import copy

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import HybridParallelPlugin
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def run_ring_mode(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    ref = LlamaForCausalLM(cfg)
    model = copy.deepcopy(ref)

    plugin = HybridParallelPlugin(
        tp_size=2, pp_size=1, precision="fp32", zero_stage=0,
        enable_sequence_parallelism=True, sequence_parallelism_mode="ring",
        parallel_output=True,
    )
    booster = Booster(plugin=plugin)
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3)
    model_b, optimizer_b, *_ = booster.boost(model, optimizer)

    torch.manual_seed(7)
    x = torch.randint(0, 128, (2, 16))
    out = model_b(input_ids=x, labels=x)
    out_ref = ref(x, labels=x)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)

    booster.backward(out["loss"], optimizer_b)
    out_ref["loss"].backward()

    # norm weights (partial grads summed over the sp/tp group)
    assert_close_loose(model_b.module.model.layers[0].input_layernorm_weight.grad,
                       ref.model.layers[0].input_layernorm_weight.grad, rtol=1e-3, atol=1e-5)
    # row-parallel o_proj grad: gather along input dim
    g = model_b.module.model.layers[0].self_attn.o_proj.weight.grad
    parts = [torch.empty_like(g) for _ in range(world_size)]
    dist.all_gather(parts, g.contiguous(), group=plugin.tp_group)
    assert_close_loose(torch.cat(parts, dim=1),
                       ref.model.layers[0].self_attn.o_proj.weight.grad, rtol=1e-3, atol=1e-4)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_sp_ring_mode():
    spawn(run_ring_mode, 2)
