"""TP-sharded native OPT + BERT vs unsharded oracles (CPU/gloo, tp=2)."""

import copy

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd.shardformer import ShardConfig, ShardFormer
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def run_tp_opt(rank, world_size, port):
    from colossalai_amd.models.opt import OPTConfig, OPTForCausalLM

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = OPTConfig(vocab_size=256, hidden_size=64, ffn_dim=128, num_hidden_layers=2,
                    num_attention_heads=4, max_position_embeddings=64)
    ref = OPTForCausalLM(cfg)
    model = copy.deepcopy(ref)
    model, _ = ShardFormer(ShardConfig(tensor_parallel_process_group=dist.group.WORLD)).optimize(model)
    assert model.model.layers[0].self_attn.num_heads == 2

    x = torch.randint(0, 256, (2, 16))
    out = model(x, labels=x)
    out_ref = ref(x, labels=x)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.model.layers[0].self_attn_layer_norm_weight.grad,
                       ref.model.layers[0].self_attn_layer_norm_weight.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


def run_tp_bert(rank, world_size, port):
    from colossalai_amd.models.bert import BertConfig, BertForMaskedLM

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = BertConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2, num_attention_heads=4,
                     intermediate_size=128, max_position_embeddings=64)
    ref = BertForMaskedLM(cfg)
    model = copy.deepcopy(ref)
    model, _ = ShardFormer(ShardConfig(tensor_parallel_process_group=dist.group.WORLD)).optimize(model)
    assert model.bert.layers[0].attention.num_heads == 2

    x = torch.randint(0, 256, (2, 16))
    out = model(x, labels=x)
    out_ref = ref(x, labels=x)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.bert.layers[0].attn_ln_weight.grad,
                       ref.bert.layers[0].attn_ln_weight.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp_opt():
    spawn(run_tp_opt, 2)


@rerun_if_address_is_in_use()
def test_tp_bert():
    spawn(run_tp_bert, 2)


def run_tp_t5(rank, world_size, port):
    from colossalai_amd.models.t5 import T5Config, T5ForConditionalGeneration

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = T5Config(vocab_size=256, d_model=64, d_kv=16, d_ff=128, num_layers=2,
                   num_decoder_layers=2, num_heads=4, relative_attention_num_buckets=8,
                   relative_attention_max_distance=32)
    ref = T5ForConditionalGeneration(cfg)
    model = copy.deepcopy(ref)
    model, _ = ShardFormer(ShardConfig(tensor_parallel_process_group=dist.group.WORLD)).optimize(model)
    assert model.encoder.block[0].self_attn.num_heads == 2
    assert model.encoder.block[0].self_attn.relative_attention_bias.weight.shape[1] == 2

    x = torch.randint(0, 256, (2, 12))
    y = torch.randint(0, 256, (2, 8))
    out = model(input_ids=x, labels=y)
    out_ref = ref(input_ids=x, labels=y)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.encoder.block[0].self_ln_weight.grad,
                       ref.encoder.block[0].self_ln_weight.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp_t5():
    spawn(run_tp_t5, 2)


def run_tp_vit(rank, world_size, port):
    from colossalai_amd.models.vit import ViTConfig, ViTForImageClassification

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = ViTConfig(image_size=32, patch_size=8, hidden_size=64, num_hidden_layers=2,
                    num_attention_heads=4, intermediate_size=128, num_labels=5)
    ref = ViTForImageClassification(cfg)
    model = copy.deepcopy(ref)
    model, _ = ShardFormer(ShardConfig(tensor_parallel_process_group=dist.group.WORLD)).optimize(model)
    assert model.vit.layers[0].attention.num_heads == 2

    x = torch.randn(2, 3, 32, 32)
    y = torch.randint(0, 5, (2,))
    out = model(x, labels=y)
    out_ref = ref(x, labels=y)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.vit.layers[0].ln1_weight.grad,
                       ref.vit.layers[0].ln1_weight.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp_vit():
    spawn(run_tp_vit, 2)
