"""Native Falcon vs HF transformers parity (CPU) + train step."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_native_falcon_matches_hf():
    from transformers import FalconConfig as HFConfig
    from transformers import FalconForCausalLM as HFFalcon

    from colossalai_amd.models.falcon import FalconConfig, FalconForCausalLM, hf_falcon_to_native

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2, num_attention_heads=4,
                      multi_query=True, new_decoder_architecture=False, parallel_attn=True,
                      bias=False, alibi=False, hidden_dropout=0.0, attention_dropout=0.0,
                      max_position_embeddings=64, attn_implementation="eager")
    hf = HFFalcon(hf_cfg).eval()
    native = FalconForCausalLM(FalconConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2,
                                            num_attention_heads=4, max_position_embeddings=64)).eval()
    missing, unexpected = native.load_state_dict(hf_falcon_to_native(hf.state_dict()), strict=False)
    assert not missing, missing
    assert not unexpected, unexpected

    x = torch.randint(0, 256, (2, 32))
    with torch.no_grad():
        ref = hf(x).logits
        out = native(x)["logits"]
    torch.testing.assert_close(out, ref, rtol=2e-3, atol=2e-3)


def test_falcon_train_step():
    from colossalai_amd.models.falcon import FalconConfig, FalconForCausalLM

    torch.manual_seed(0)
    m = FalconForCausalLM(FalconConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2,
                                       num_attention_heads=4, max_position_embeddings=64))
    x = torch.randint(0, 256, (2, 32))
    out = m(x, labels=x)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
