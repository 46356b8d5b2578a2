"""Native model ≡ HF transformers Llama: same weights → same logits."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_native_matches_hf_llama():
    from transformers import LlamaConfig as HFConfig
    from transformers import LlamaForCausalLM as HFLlama

    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
    from colossalai_amd.models.hf_compat import hf_to_native_llama, native_to_hf_llama

    torch.manual_seed(0)
    hf_cfg = HFConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64,
        rms_norm_eps=1e-5, rope_theta=10000.0, tie_word_embeddings=False,
        attn_implementation="eager",
    )
    hf_model = HFLlama(hf_cfg).eval()

    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64,
                      rms_norm_eps=1e-5)
    native = LlamaForCausalLM(cfg).eval()
    missing = native.load_state_dict(hf_to_native_llama(hf_model.state_dict()), strict=True)

    x = torch.randint(0, 128, (2, 16))
    with torch.no_grad():
        hf_logits = hf_model(x).logits
        native_logits = native(x)["logits"]
    torch.testing.assert_close(native_logits, hf_logits, rtol=2e-3, atol=2e-3)

    # round-trip back to HF layout
    back = native_to_hf_llama(native.state_dict(), cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim)
    hf2 = HFLlama(hf_cfg)
    incompat = hf2.load_state_dict(back, strict=False)
    assert not incompat.unexpected_keys, incompat.unexpected_keys
    leftover = [k for k in incompat.missing_keys if "rotary" not in k]
    assert not leftover, leftover
