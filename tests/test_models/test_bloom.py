"""Native BLOOM vs HF transformers parity (CPU) + train step."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_native_bloom_matches_hf():
    from transformers import BloomConfig as HFConfig
    from transformers import BloomForCausalLM as HFBloom

    from colossalai_amd.models.bloom import BloomConfig, BloomForCausalLM, hf_bloom_to_native

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=256, hidden_size=64, n_layer=2, n_head=4,
                      hidden_dropout=0.0, attention_dropout=0.0, attn_implementation="eager")
    hf = HFBloom(hf_cfg).eval()
    native = BloomForCausalLM(BloomConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2,
                                          num_attention_heads=4)).eval()
    missing, unexpected = native.load_state_dict(hf_bloom_to_native(hf.state_dict(), 4), strict=False)
    assert not missing, missing
    assert not unexpected, unexpected

    x = torch.randint(0, 256, (2, 24))
    with torch.no_grad():
        ref = hf(x).logits
        out = native(x)["logits"]
    torch.testing.assert_close(out, ref, rtol=2e-3, atol=2e-3)


def test_bloom_train_step():
    from colossalai_amd.models.bloom import BloomConfig, BloomForCausalLM

    torch.manual_seed(0)
    m = BloomForCausalLM(BloomConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2,
                                     num_attention_heads=4))
    x = torch.randint(0, 256, (2, 24))
    out = m(x, labels=x)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
