"""GPU smoke of the round-2 model families: bf16 forward+backward on the
HIP compute path (flash attention where head dims allow)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _check(loss, model):
    assert torch.isfinite(loss).all()
    loss.backward()
    n_grad = sum(p.grad is not None for p in model.parameters() if p.requires_grad)
    assert n_grad > 0


def test_chatglm_gpu():
    from colossalai_amd.models.chatglm2 import ChatGLMConfig, ChatGLMForConditionalGeneration

    if not torch.cuda.is_available():
        pytest.skip("GPU only")
    torch.manual_seed(0)
    cfg = ChatGLMConfig(vocab_size=512, hidden_size=512, ffn_hidden_size=1024,
                        num_hidden_layers=2, num_attention_heads=4, multi_query_group_num=2)
    m = ChatGLMForConditionalGeneration(cfg).cuda().bfloat16()
    x = torch.randint(0, 512, (2, 64), device="cuda")
    _check(m(x, labels=x)["loss"], m)


def test_deepseek_v3_gpu():
    from colossalai_amd.models.deepseek_v3 import DEEPSEEK_V3_CONFIGS, DeepseekV3ForCausalLM

    if not torch.cuda.is_available():
        pytest.skip("GPU only")
    torch.manual_seed(0)
    m = DeepseekV3ForCausalLM(DEEPSEEK_V3_CONFIGS["deepseek-v3-tiny"]).cuda().bfloat16()
    x = torch.randint(0, 128, (2, 32), device="cuda")
    _check(m(x, labels=x)["loss"], m)


def test_sam_gpu():
    from colossalai_amd.models.sam import SamConfig, SamModel, SamVisionConfig

    if not torch.cuda.is_available():
        pytest.skip("GPU only")
    torch.manual_seed(0)
    cfg = SamConfig(
        vision=SamVisionConfig(image_size=64, patch_size=8, hidden_size=64, num_hidden_layers=2,
                               num_attention_heads=4, window_size=4, global_attn_indexes=(1,),
                               output_channels=32),
        prompt_embed_dim=32, decoder_hidden=32, decoder_heads=4, decoder_layers=2,
        decoder_mlp_dim=64)
    m = SamModel(cfg).cuda().bfloat16()
    px = torch.randn(2, 3, 64, 64, device="cuda", dtype=torch.bfloat16)
    tgt = torch.rand(2, 32, 32, device="cuda") > 0.5
    _check(m(px, mask_labels=tgt)["loss"], m)


def test_blip2_gpu():
    from colossalai_amd.models.blip2 import Blip2Config, Blip2ForConditionalGeneration
    from colossalai_amd.models.opt import OPTConfig
    from colossalai_amd.models.vit import ViTConfig

    if not torch.cuda.is_available():
        pytest.skip("GPU only")
    torch.manual_seed(0)
    cfg = Blip2Config(
        vision=ViTConfig(image_size=32, patch_size=8, num_channels=3, hidden_size=256,
                         num_hidden_layers=2, num_attention_heads=4, intermediate_size=512),
        text=OPTConfig(vocab_size=512, hidden_size=256, ffn_dim=512, num_hidden_layers=2,
                       num_attention_heads=4, max_position_embeddings=128),
        qformer_hidden=256, qformer_layers=2, qformer_heads=4, num_query_tokens=4)
    m = Blip2ForConditionalGeneration(cfg).cuda().bfloat16()
    px = torch.randn(2, 3, 32, 32, device="cuda", dtype=torch.bfloat16)
    ids = torch.randint(0, 512, (2, 16), device="cuda")
    _check(m(px, ids, labels=ids)["loss"], m)


def test_dit_gpu():
    from colossalai_amd.models.dit import DiT, DiTConfig

    if not torch.cuda.is_available():
        pytest.skip("GPU only")
    torch.manual_seed(0)
    cfg = DiTConfig(input_size=16, patch_size=2, in_channels=4, hidden_size=256,
                    num_hidden_layers=2, num_attention_heads=4, num_classes=10)
    m = DiT(cfg).cuda().bfloat16()
    x = torch.randn(2, 4, 16, 16, device="cuda", dtype=torch.bfloat16)
    t = torch.randint(0, 1000, (2,), device="cuda")
    noise = torch.randn_like(x)
    _check(m(x, t, torch.tensor([1, 5], device="cuda"), noise_target=noise)["loss"], m)
