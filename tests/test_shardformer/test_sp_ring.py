"""Ring-attention SP vs unsharded oracle (CPU/gloo, sp=2)."""

import copy

import torch
import torch.distributed as dist
import torch.nn.functional as F

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import HybridParallelPlugin
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def run_ring(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    ref = LlamaForCausalLM(cfg)
    model = copy.deepcopy(ref)

    plugin = HybridParallelPlugin(
        tp_size=1, pp_size=1, sp_size=2, precision="fp32", zero_stage=0,
        enable_sequence_parallelism=True, sequence_parallelism_mode="ring_attn",
    )
    booster = Booster(plugin=plugin)
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3)
    model_b, optimizer_b, *_ = booster.boost(model, optimizer)

    torch.manual_seed(7)
    S = 16
    x = torch.randint(0, 128, (2, S))
    out = model_b(input_ids=x, labels=x)

    logits_ref = ref(x)["logits"]
    shifted = torch.full_like(x, -100)
    shifted[:, :-1] = x[:, 1:]
    shard = S // world_size
    local_losses = []
    for r in range(world_size):
        lg = logits_ref[:, r * shard : (r + 1) * shard]
        lb = shifted[:, r * shard : (r + 1) * shard]
        local_losses.append(F.cross_entropy(lg.reshape(-1, 128).float(), lb.reshape(-1), ignore_index=-100))
    ref_loss = sum(local_losses) / world_size

    assert_close_loose(out["loss"], local_losses[rank], rtol=1e-4, atol=1e-5)

    booster.backward(out["loss"], optimizer_b)
    ref_loss.backward()
    assert_close_loose(model_b.module.model.layers[0].input_layernorm_weight.grad,
                       ref.model.layers[0].input_layernorm_weight.grad, rtol=1e-3, atol=1e-5)
    assert_close_loose(model_b.module.model.layers[1].self_attn.qkv_proj.weight.grad,
                       ref.model.layers[1].self_attn.qkv_proj.weight.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_sp_ring():
    spawn(run_ring, 2)


def run_ring_zigzag_op(rank, world_size, port):
    """Op-level zigzag ring vs dense fp32 attention, fwd + bwd."""
    from colossalai_amd.ops.attention import attention_ref
    from colossalai_amd.shardformer.layer.ring_attn import (
        ring_flash_attention,
        zigzag_gather,
        zigzag_split,
    )

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    B, S, Hq, Hkv, D = 2, 32, 4, 2, 16
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    dout = torch.randn(B, S, Hq, D)

    ql = zigzag_split(q, world_size, rank).requires_grad_(True)
    kl = zigzag_split(k, world_size, rank).requires_grad_(True)
    vl = zigzag_split(v, world_size, rank).requires_grad_(True)
    out = ring_flash_attention(ql, kl, vl, dist.group.WORLD, causal=True, zigzag=True)
    out.backward(zigzag_split(dout, world_size, rank))

    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ref = attention_ref(qr, kr, vr, causal=True)
    ref.backward(dout)

    assert_close_loose(out, zigzag_split(ref, world_size, rank), rtol=1e-4, atol=1e-5)
    assert_close_loose(ql.grad, zigzag_split(qr.grad, world_size, rank), rtol=1e-4, atol=1e-5)
    assert_close_loose(kl.grad, zigzag_split(kr.grad, world_size, rank), rtol=1e-4, atol=1e-5)
    assert_close_loose(vl.grad, zigzag_split(vr.grad, world_size, rank), rtol=1e-4, atol=1e-5)
    dist.destroy_process_group()


def run_ring_zigzag_model(rank, world_size, port):
    """Full llama with zigzag ring SP vs unsharded oracle."""
    from colossalai_amd.shardformer.layer.ring_attn import zigzag_split

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    ref = LlamaForCausalLM(cfg)
    model = copy.deepcopy(ref)

    plugin = HybridParallelPlugin(
        tp_size=1, pp_size=1, sp_size=2, precision="fp32", zero_stage=0,
        enable_sequence_parallelism=True, sequence_parallelism_mode="ring_attn", sp_zigzag=True,
    )
    booster = Booster(plugin=plugin)
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3)
    model_b, optimizer_b, *_ = booster.boost(model, optimizer)

    torch.manual_seed(7)
    S = 16
    x = torch.randint(0, 128, (2, S))
    out = model_b(input_ids=x, labels=x)

    logits_ref = ref(x)["logits"]
    shifted = torch.full_like(x, -100)
    shifted[:, :-1] = x[:, 1:]
    local_losses = []
    for r in range(world_size):
        lg = zigzag_split(logits_ref, world_size, r)
        lb = zigzag_split(shifted, world_size, r)
        local_losses.append(F.cross_entropy(lg.reshape(-1, 128).float(), lb.reshape(-1), ignore_index=-100))
    ref_loss = sum(local_losses) / world_size

    assert_close_loose(out["loss"], local_losses[rank], rtol=1e-4, atol=1e-5)

    booster.backward(out["loss"], optimizer_b)
    ref_loss.backward()
    assert_close_loose(model_b.module.model.layers[0].input_layernorm_weight.grad,
                       ref.model.layers[0].input_layernorm_weight.grad, rtol=1e-3, atol=1e-5)
    assert_close_loose(model_b.module.model.layers[1].self_attn.qkv_proj.weight.grad,
                       ref.model.layers[1].self_attn.qkv_proj.weight.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_sp_ring_zigzag_op():
    spawn(run_ring_zigzag_op, 2)


@rerun_if_address_is_in_use()
def test_sp_ring_zigzag_model():
    spawn(run_ring_zigzag_model, 2)


def run_ring_zigzag_padded_op(rank, world_size, port):
    """Padded (ragged) batches under zigzag ring: masked dense oracle,
    fwd + bwd, world 4 — the reference's prepare_varlen_batch role."""
    from colossalai_amd.ops.attention import attention_ref
    from colossalai_amd.shardformer.layer.ring_attn import (
        ring_flash_attention,
        zigzag_split,
    )

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    B, S, Hq, Hkv, D = 3, 64, 4, 2, 16
    seqlens = torch.tensor([64, 37, 9], dtype=torch.int32)  # ragged, right-padded
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    dout = torch.randn(B, S, Hq, D)
    # zero pad-region dout so grad comparison ignores don't-care rows
    for b in range(B):
        dout[b, int(seqlens[b]):] = 0

    ql = zigzag_split(q, world_size, rank).requires_grad_(True)
    kl = zigzag_split(k, world_size, rank).requires_grad_(True)
    vl = zigzag_split(v, world_size, rank).requires_grad_(True)
    out = ring_flash_attention(ql, kl, vl, dist.group.WORLD, causal=True, zigzag=True,
                               seqlens=seqlens)
    out.backward(zigzag_split(dout, world_size, rank))

    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ref = attention_ref(qr, kr, vr, causal=True, seqlens=seqlens)
    ref.backward(dout)

    assert_close_loose(out, zigzag_split(ref, world_size, rank), rtol=1e-4, atol=1e-5)
    assert_close_loose(ql.grad, zigzag_split(qr.grad, world_size, rank), rtol=1e-4, atol=1e-5)
    assert_close_loose(kl.grad, zigzag_split(kr.grad, world_size, rank), rtol=1e-4, atol=1e-5)
    assert_close_loose(vl.grad, zigzag_split(vr.grad, world_size, rank), rtol=1e-4, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_sp_ring_zigzag_padded():
    spawn(run_ring_zigzag_padded_op, 4)


def run_ring_zigzag_padded_model(rank, world_size, port):
    """Full llama under zigzag ring SP with a right-padded attention_mask
    vs the unsharded padded oracle (loss + grads)."""
    from colossalai_amd.shardformer.layer.ring_attn import zigzag_split

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    ref = LlamaForCausalLM(cfg)
    model = copy.deepcopy(ref)

    plugin = HybridParallelPlugin(
        tp_size=1, pp_size=1, sp_size=2, precision="fp32", zero_stage=0,
        enable_sequence_parallelism=True, sequence_parallelism_mode="ring_attn", sp_zigzag=True,
    )
    booster = Booster(plugin=plugin)
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3)
    model_b, optimizer_b, *_ = booster.boost(model, optimizer)

    torch.manual_seed(7)
    S = 16
    seqlens = [16, 9]
    x = torch.randint(0, 128, (2, S))
    mask = torch.zeros(2, S, dtype=torch.long)
    labels = torch.full_like(x, -100)
    for b, L in enumerate(seqlens):
        mask[b, :L] = 1
        labels[b, :L] = x[b, :L]

    out = model_b(input_ids=x, attention_mask=mask, labels=labels)

    logits_ref = ref(x, attention_mask=mask)["logits"]
    shifted = torch.full_like(x, -100)
    shifted[:, :-1] = labels[:, 1:]
    local_losses = []
    for r in range(world_size):
        lg = zigzag_split(logits_ref, world_size, r)
        lb = zigzag_split(shifted, world_size, r)
        local_losses.append(F.cross_entropy(lg.reshape(-1, 128).float(), lb.reshape(-1), ignore_index=-100))
    ref_loss = sum(local_losses) / world_size

    assert_close_loose(out["loss"], local_losses[rank], rtol=1e-4, atol=1e-5)

    booster.backward(out["loss"], optimizer_b)
    ref_loss.backward()
    assert_close_loose(model_b.module.model.layers[0].input_layernorm_weight.grad,
                       ref.model.layers[0].input_layernorm_weight.grad, rtol=1e-3, atol=1e-5)
    assert_close_loose(model_b.module.model.layers[1].self_attn.qkv_proj.weight.grad,
                       ref.model.layers[1].self_attn.qkv_proj.weight.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_sp_ring_zigzag_padded_model():
    spawn(run_ring_zigzag_padded_model, 2)
