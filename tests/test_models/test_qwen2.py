"""Qwen2-family (qkv-bias Llama arch) vs HF parity."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_native_matches_hf_qwen2():
    from transformers import Qwen2Config as HFConfig
    from transformers import Qwen2ForCausalLM as HFQwen2

    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
    from colossalai_amd.models.hf_compat import hf_to_native_llama

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64,
                      rms_norm_eps=1e-5, rope_theta=10000.0, tie_word_embeddings=False,
                      attn_implementation="eager", attention_bias=True)
    hf = HFQwen2(hf_cfg).eval()
    # HF qwen2 initializes biases to 0 — randomize so parity is meaningful
    with torch.no_grad():
        for n, p in hf.named_parameters():
            if "bias" in n:
                p.normal_(0, 0.1)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64,
                      rms_norm_eps=1e-5, attention_bias=True)
    native = LlamaForCausalLM(cfg).eval()
    native.load_state_dict(hf_to_native_llama(hf.state_dict()), strict=True)

    x = torch.randint(0, 128, (2, 16))
    with torch.no_grad():
        ref = hf(x).logits
        out = native(x)["logits"]
    torch.testing.assert_close(out, ref, rtol=2e-3, atol=2e-3)
