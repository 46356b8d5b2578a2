"""Native Qwen3 (qk-norm Llama variant) vs HF transformers parity."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_native_qwen3_matches_hf():
    from transformers import Qwen3Config as HFConfig
    from transformers import Qwen3ForCausalLM as HFQwen3

    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
    from colossalai_amd.models.hf_compat import hf_to_native_llama

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=256, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, head_dim=32,
                      max_position_embeddings=64, rope_theta=10000.0, tie_word_embeddings=False,
                      attention_bias=False, attn_implementation="eager")
    hf = HFQwen3(hf_cfg).eval()
    native = LlamaForCausalLM(LlamaConfig(vocab_size=256, hidden_size=64, intermediate_size=128,
                                          num_hidden_layers=2, num_attention_heads=4,
                                          num_key_value_heads=2, head_dim_override=32,
                                          max_position_embeddings=64, rms_norm_eps=hf_cfg.rms_norm_eps,
                                          qk_norm=True)).eval()
    sd = hf_to_native_llama(hf.state_dict())
    missing, unexpected = native.load_state_dict(sd, strict=False)
    missing = [m for m in missing if "q_proj" not in m and "k_proj" not in m and "v_proj" not in m
               and "gate_proj" not in m and "up_proj" not in m]
    assert not missing, missing

    x = torch.randint(0, 256, (2, 24))
    with torch.no_grad():
        ref = hf(x).logits
        out = native(x)["logits"]
    torch.testing.assert_close(out, ref, rtol=2e-3, atol=2e-3)


def test_qwen3_train_step():
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(0)
    m = LlamaForCausalLM(LlamaConfig(vocab_size=256, hidden_size=64, intermediate_size=128,
                                     num_hidden_layers=2, num_attention_heads=4,
                                     num_key_value_heads=2, head_dim_override=32,
                                     max_position_embeddings=64, qk_norm=True))
    x = torch.randint(0, 256, (2, 24))
    out = m(x, labels=x)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
