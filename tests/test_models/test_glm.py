"""Native GLM-4 (Llama-family flag) vs HF transformers parity."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_native_glm_matches_hf():
    from transformers import GlmConfig as HFConfig
    from transformers import GlmForCausalLM as HFGlm

    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
    from colossalai_amd.models.hf_compat import hf_to_native_llama

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=256, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, head_dim=16,
                      partial_rotary_factor=0.5, attention_bias=True, pad_token_id=0,
                      max_position_embeddings=64, rope_theta=10000.0, tie_word_embeddings=False,
                      attn_implementation="eager", attention_dropout=0.0)
    hf = HFGlm(hf_cfg).eval()
    native = LlamaForCausalLM(LlamaConfig(vocab_size=256, hidden_size=64, intermediate_size=128,
                                          num_hidden_layers=2, num_attention_heads=4,
                                          num_key_value_heads=2, head_dim_override=16,
                                          max_position_embeddings=64, attention_bias=True,
                                          rms_norm_eps=hf_cfg.rms_norm_eps,
                                          partial_interleaved_rotary_factor=0.5)).eval()
    sd = hf_to_native_llama(hf.state_dict())
    missing, unexpected = native.load_state_dict(sd, strict=False)
    missing = [m for m in missing if not any(s in m for s in
               ("q_proj", "k_proj", "v_proj", "gate_proj", "up_proj"))]
    assert not missing, missing

    x = torch.randint(0, 256, (2, 24))
    with torch.no_grad():
        ref = hf(x).logits
        out = native(x)["logits"]
    torch.testing.assert_close(out, ref, rtol=2e-3, atol=2e-3)
