"""Native Cohere/Command-R vs HF transformers parity (CPU) + train step."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_native_cohere_matches_hf():
    from transformers import CohereConfig as HFConfig
    from transformers import CohereForCausalLM as HFCohere

    from colossalai_amd.models.cohere import CohereConfig, CohereForCausalLM, hf_cohere_to_native

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=256, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64,
                      rope_theta=10000.0, logit_scale=0.125, layer_norm_eps=1e-5,
                      use_qk_norm=False, attn_implementation="eager", attention_dropout=0.0)
    hf = HFCohere(hf_cfg).eval()
    native = CohereForCausalLM(CohereConfig(vocab_size=256, hidden_size=64, intermediate_size=128,
                                            num_hidden_layers=2, num_attention_heads=4,
                                            num_key_value_heads=2, max_position_embeddings=64,
                                            rope_theta=10000.0, logit_scale=0.125)).eval()
    missing, unexpected = native.load_state_dict(hf_cohere_to_native(hf.state_dict()), strict=False)
    assert not missing, missing
    assert not unexpected, unexpected

    x = torch.randint(0, 256, (2, 24))
    with torch.no_grad():
        ref = hf(x).logits
        out = native(x)["logits"]
    torch.testing.assert_close(out, ref, rtol=2e-3, atol=2e-3)


def test_cohere_train_step():
    from colossalai_amd.models.cohere import CohereConfig, CohereForCausalLM

    torch.manual_seed(0)
    m = CohereForCausalLM(CohereConfig(vocab_size=256, hidden_size=64, intermediate_size=128,
                                       num_hidden_layers=2, num_attention_heads=4,
                                       num_key_value_heads=2, max_position_embeddings=64))
    x = torch.randint(0, 256, (2, 24))
    out = m(x, labels=x)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
