"""State-dict round trips for the round-2 families: save -> load into a
fresh instance -> identical forward outputs (catches unregistered
buffers/params and non-deterministic construction)."""

import torch

from colossalai_amd.models.blip2 import Blip2Config, Blip2ForConditionalGeneration
from colossalai_amd.models.chatglm2 import ChatGLMConfig, ChatGLMForConditionalGeneration
from colossalai_amd.models.deepseek_v3 import DEEPSEEK_V3_CONFIGS, DeepseekV3ForCausalLM
from colossalai_amd.models.dit import DiT, DiTConfig
from colossalai_amd.models.opt import OPTConfig
from colossalai_amd.models.sam import SamConfig, SamModel, SamVisionConfig
from colossalai_amd.models.vit import ViTConfig


def _roundtrip(make, run):
    torch.manual_seed(0)
    a = make().eval()
    torch.manual_seed(123)  # different init for the destination
    b = make().eval()
    b.load_state_dict(a.state_dict())
    with torch.no_grad():
        oa, ob = run(a), run(b)
    torch.testing.assert_close(oa, ob, rtol=0.0, atol=0.0)


def test_chatglm_roundtrip():
    cfg = ChatGLMConfig(vocab_size=128, hidden_size=64, ffn_hidden_size=96,
                        num_hidden_layers=2, num_attention_heads=4, multi_query_group_num=2)
    x = torch.randint(0, 128, (2, 12))
    _roundtrip(lambda: ChatGLMForConditionalGeneration(cfg), lambda m: m(x)["logits"])


def test_deepseek_v3_roundtrip():
    cfg = DEEPSEEK_V3_CONFIGS["deepseek-v3-tiny"]
    x = torch.randint(0, 128, (2, 12))
    _roundtrip(lambda: DeepseekV3ForCausalLM(cfg), lambda m: m(x)["logits"])


def test_sam_roundtrip():
    cfg = SamConfig(
        vision=SamVisionConfig(image_size=32, patch_size=8, hidden_size=32, num_hidden_layers=2,
                               num_attention_heads=4, window_size=2, global_attn_indexes=(1,),
                               output_channels=16),
        prompt_embed_dim=16, decoder_hidden=16, decoder_heads=4, decoder_layers=1,
        decoder_mlp_dim=32)
    px = torch.randn(1, 3, 32, 32)
    _roundtrip(lambda: SamModel(cfg), lambda m: m(px)["pred_masks"])


def test_blip2_roundtrip():
    cfg = Blip2Config(
        vision=ViTConfig(image_size=16, patch_size=8, num_channels=3, hidden_size=32,
                         num_hidden_layers=1, num_attention_heads=4, intermediate_size=64),
        text=OPTConfig(vocab_size=128, hidden_size=32, ffn_dim=64, num_hidden_layers=1,
                       num_attention_heads=4, max_position_embeddings=32),
        qformer_hidden=32, qformer_layers=1, qformer_heads=4, num_query_tokens=2)
    px = torch.randn(1, 3, 16, 16)
    ids = torch.randint(0, 128, (1, 8))
    _roundtrip(lambda: Blip2ForConditionalGeneration(cfg), lambda m: m(px, ids)["logits"])


def test_dit_roundtrip():
    cfg = DiTConfig(input_size=8, patch_size=2, in_channels=4, hidden_size=32,
                    num_hidden_layers=2, num_attention_heads=4, num_classes=5)
    x = torch.randn(2, 4, 8, 8)
    t = torch.tensor([3, 500])
    lab = torch.tensor([1, 4])
    _roundtrip(lambda: DiT(cfg), lambda m: m(x, t, lab)["sample"])
