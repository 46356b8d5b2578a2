"""TP policies for the remaining native families (BLOOM, GPT-J, Whisper,
Cohere) vs their unsharded oracles (CPU/gloo, tp=2)
(reference registry: colossalai/shardformer/policies/auto_policy.py:30-291)."""

import copy

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd.shardformer import ShardConfig, ShardFormer
from colossalai_amd.testing import assert_close_loose, rerun_if_address_is_in_use, spawn


def _shard(model):
    sharded, _ = ShardFormer(ShardConfig(tensor_parallel_process_group=dist.group.WORLD)).optimize(model)
    return sharded


def run_tp_bloom(rank, world_size, port):
    from colossalai_amd.models.bloom import BloomConfig, BloomForCausalLM

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = BloomConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2, num_attention_heads=4)
    ref = BloomForCausalLM(cfg)
    model = _shard(copy.deepcopy(ref))
    attn = model.transformer.h[0].self_attention
    assert attn.num_heads == 2 and attn.slopes.numel() == 2

    x = torch.randint(0, 256, (2, 16))
    out = model(x, labels=x)
    out_ref = ref(x, labels=x)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.transformer.h[0].ln1_w.grad, ref.transformer.h[0].ln1_w.grad,
                       rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


def run_tp_gptj(rank, world_size, port):
    from colossalai_amd.models.gptj import GPTJConfig, GPTJForCausalLM

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = GPTJConfig(vocab_size=256, n_embd=64, n_layer=2, n_head=4, rotary_dim=8, n_positions=64)
    ref = GPTJForCausalLM(cfg)
    model = _shard(copy.deepcopy(ref))
    assert model.transformer.h[0].attn.num_heads == 2

    x = torch.randint(0, 256, (2, 16))
    out = model(x, labels=x)
    out_ref = ref(x, labels=x)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.transformer.h[0].ln_1_weight.grad,
                       ref.transformer.h[0].ln_1_weight.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


def run_tp_whisper(rank, world_size, port):
    from colossalai_amd.models.whisper import WhisperConfig, WhisperForConditionalGeneration

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = WhisperConfig(vocab_size=256, num_mel_bins=16, d_model=64, encoder_layers=2,
                        decoder_layers=2, num_heads=4, d_ff=128, max_source_positions=32,
                        max_target_positions=32, decoder_start_token_id=1, pad_token_id=0)
    ref = WhisperForConditionalGeneration(cfg)
    model = _shard(copy.deepcopy(ref))
    assert model.encoder.layers[0].self_attn.num_heads == 2
    assert model.decoder.layers[0].cross_attn.num_heads == 2

    mel = torch.randn(2, 16, 64)  # [B, mel, frames]
    labels = torch.randint(2, 256, (2, 8))
    out = model(mel, labels=labels)
    out_ref = ref(mel, labels=labels)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.encoder.layers[0].self_ln_w.grad,
                       ref.encoder.layers[0].self_ln_w.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


def run_tp_cohere(rank, world_size, port):
    from colossalai_amd.models.cohere import CohereConfig, CohereForCausalLM

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = CohereConfig(vocab_size=256, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                       num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    ref = CohereForCausalLM(cfg)
    model = _shard(copy.deepcopy(ref))
    attn = model.model.layers[0].self_attn
    assert attn.num_heads == 2 and attn.num_kv_heads == 1

    x = torch.randint(0, 256, (2, 16))
    out = model(x, labels=x)
    out_ref = ref(x, labels=x)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.model.layers[0].ln_weight.grad,
                       ref.model.layers[0].ln_weight.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp_bloom():
    spawn(run_tp_bloom, 2)


@rerun_if_address_is_in_use()
def test_tp_gptj():
    spawn(run_tp_gptj, 2)


@rerun_if_address_is_in_use()
def test_tp_whisper():
    spawn(run_tp_whisper, 2)


@rerun_if_address_is_in_use()
def test_tp_cohere():
    spawn(run_tp_cohere, 2)


def run_tp_blip2(rank, world_size, port):
    from colossalai_amd.models.blip2 import Blip2Config, Blip2ForConditionalGeneration
    from colossalai_amd.models.opt import OPTConfig
    from colossalai_amd.models.vit import ViTConfig

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = Blip2Config(
        vision=ViTConfig(image_size=16, patch_size=8, num_channels=3, hidden_size=64,
                         num_hidden_layers=2, num_attention_heads=4, intermediate_size=128),
        text=OPTConfig(vocab_size=256, hidden_size=64, ffn_dim=128, num_hidden_layers=2,
                       num_attention_heads=4, max_position_embeddings=64),
        qformer_hidden=64, qformer_layers=2, qformer_heads=4, num_query_tokens=4,
    )
    ref = Blip2ForConditionalGeneration(cfg)
    model = _shard(copy.deepcopy(ref))
    assert model.vision_model.layers[0].attention.num_heads == 2
    assert model.qformer_layers[0].self_attn.num_heads == 2
    assert model.language_model.model.layers[0].self_attn.num_heads == 2

    px = torch.randn(2, 3, 16, 16)
    ids = torch.randint(0, 256, (2, 12))
    out = model(px, ids, labels=ids)
    out_ref = ref(px, ids, labels=ids)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.qformer_ln_w.grad, ref.qformer_ln_w.grad, rtol=1e-3, atol=1e-5)
    assert_close_loose(model.query_tokens.grad, ref.query_tokens.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp_blip2():
    spawn(run_tp_blip2, 2)


def run_tp_chatglm(rank, world_size, port):
    from colossalai_amd.models.chatglm2 import ChatGLMConfig, ChatGLMForConditionalGeneration

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = ChatGLMConfig(vocab_size=256, hidden_size=64, ffn_hidden_size=96,
                        num_hidden_layers=2, num_attention_heads=4, multi_query_group_num=2)
    ref = ChatGLMForConditionalGeneration(cfg)
    model = _shard(copy.deepcopy(ref))
    attn = model.transformer.layers[0].self_attention
    assert attn.num_heads == 2 and attn.num_kv_heads == 1

    x = torch.randint(0, 256, (2, 16))
    out = model(x, labels=x)
    out_ref = ref(x, labels=x)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.transformer.layers[0].input_ln_w.grad,
                       ref.transformer.layers[0].input_ln_w.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp_chatglm():
    spawn(run_tp_chatglm, 2)


def run_tp_deepseek_v3(rank, world_size, port):
    """MLA attention under TP: per-head shards of q_b/kv_b, replicated latents."""
    from colossalai_amd.models.deepseek_v3 import DEEPSEEK_V3_CONFIGS, DeepseekV3ForCausalLM

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    ref = DeepseekV3ForCausalLM(DEEPSEEK_V3_CONFIGS["deepseek-v3-tiny"])
    model = _shard(copy.deepcopy(ref))
    attn = model.model.layers[0].self_attn
    assert attn.num_heads == 2 and attn.q_b_proj.weight.shape[0] == 2 * (16 + 8)

    x = torch.randint(0, 128, (2, 16))
    out = model(x, labels=x)
    out_ref = ref(x, labels=x)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.model.layers[0].self_attn.kv_a_ln_w.grad,
                       ref.model.layers[0].self_attn.kv_a_ln_w.grad, rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp_deepseek_v3():
    spawn(run_tp_deepseek_v3, 2)


def run_tp_sam(rank, world_size, port):
    from colossalai_amd.models.sam import SamConfig, SamModel, SamVisionConfig

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = SamConfig(
        vision=SamVisionConfig(image_size=64, patch_size=8, hidden_size=64, num_hidden_layers=3,
                               num_attention_heads=4, window_size=4, global_attn_indexes=(1,),
                               output_channels=32),
        prompt_embed_dim=32, decoder_hidden=32, decoder_heads=4, decoder_layers=2,
        decoder_mlp_dim=64)
    ref = SamModel(cfg)
    model = _shard(copy.deepcopy(ref))
    assert model.vision_encoder.layers[0].attn.num_heads == 2
    assert model.mask_decoder.layers[0].self_attn.num_heads == 2

    px = torch.randn(2, 3, 64, 64)
    pts = torch.rand(2, 3, 2)
    lbl = torch.randint(0, 2, (2, 3))
    tgt = torch.rand(2, 32, 32) > 0.5
    out = model(px, pts, lbl, mask_labels=tgt)
    out_ref = ref(px, pts, lbl, mask_labels=tgt)
    assert_close_loose(out["loss"], out_ref["loss"], rtol=1e-4, atol=1e-5)
    out["loss"].backward()
    out_ref["loss"].backward()
    assert_close_loose(model.vision_encoder.layers[0].ln1_w.grad,
                       ref.vision_encoder.layers[0].ln1_w.grad, rtol=1e-3, atol=1e-5)
    # rel-pos grads are partial per rank (local heads only) and marked
    # _sp_partial_grad; the plugin all-reduces them — emulate that here
    rel = model.vision_encoder.layers[0].attn.rel_pos_h
    assert getattr(rel, "_sp_partial_grad", False)
    g = rel.grad.clone()
    dist.all_reduce(g)
    assert_close_loose(g, ref.vision_encoder.layers[0].attn.rel_pos_h.grad,
                       rtol=1e-3, atol=1e-8)
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_tp_sam():
    spawn(run_tp_sam, 2)
