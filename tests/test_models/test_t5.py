"""Native T5 vs HF transformers parity (CPU) + train-step smoke."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def _pair():
    from transformers import T5Config as HFConfig
    from transformers import T5ForConditionalGeneration as HFT5

    from colossalai_amd.models.t5 import T5Config, T5ForConditionalGeneration, hf_t5_to_native

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=256, d_model=64, d_kv=16, d_ff=128, num_layers=2,
                      num_decoder_layers=2, num_heads=4, relative_attention_num_buckets=8,
                      relative_attention_max_distance=32, dropout_rate=0.0,
                      decoder_start_token_id=0, tie_word_embeddings=True)
    hf = HFT5(hf_cfg).eval()
    native = T5ForConditionalGeneration(T5Config(
        vocab_size=256, d_model=64, d_kv=16, d_ff=128, num_layers=2, num_decoder_layers=2,
        num_heads=4, relative_attention_num_buckets=8, relative_attention_max_distance=32)).eval()
    missing, unexpected = native.load_state_dict(hf_t5_to_native(hf.state_dict()), strict=False)
    assert not missing, missing
    assert not unexpected, unexpected
    return hf, native


def test_native_t5_matches_hf():
    hf, native = _pair()
    x = torch.randint(0, 256, (2, 18))
    y = torch.randint(0, 256, (2, 11))
    with torch.no_grad():
        ref = hf(input_ids=x, labels=y)
        out = native(input_ids=x, labels=y)
    torch.testing.assert_close(out["logits"], ref.logits, rtol=2e-3, atol=2e-3)
    torch.testing.assert_close(out["loss"], ref.loss, rtol=1e-3, atol=1e-4)


def test_t5_train_step():
    from colossalai_amd.models.t5 import T5Config, T5ForConditionalGeneration

    torch.manual_seed(0)
    m = T5ForConditionalGeneration(T5Config(vocab_size=256, d_model=64, d_kv=16, d_ff=128,
                                            num_layers=2, num_decoder_layers=2, num_heads=4))
    x = torch.randint(0, 256, (2, 16))
    y = torch.randint(0, 256, (2, 10))
    out = m(input_ids=x, labels=y)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
