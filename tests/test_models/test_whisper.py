"""Native Whisper vs HF transformers parity (CPU) + train step."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def _cfgs():
    from transformers import WhisperConfig as HFConfig

    from colossalai_amd.models.whisper import WhisperConfig

    hf = HFConfig(vocab_size=256, num_mel_bins=16, d_model=64, encoder_layers=2,
                  decoder_layers=2, encoder_attention_heads=4, decoder_attention_heads=4,
                  encoder_ffn_dim=128, decoder_ffn_dim=128, max_source_positions=32,
                  max_target_positions=32, dropout=0.0, attention_dropout=0.0,
                  activation_dropout=0.0, decoder_start_token_id=3, pad_token_id=2,
                  attn_implementation="eager")
    native = WhisperConfig(vocab_size=256, num_mel_bins=16, d_model=64, encoder_layers=2,
                           decoder_layers=2, num_heads=4, d_ff=128, max_source_positions=32,
                           max_target_positions=32, decoder_start_token_id=3, pad_token_id=2)
    return hf, native


def test_native_whisper_matches_hf():
    from transformers import WhisperForConditionalGeneration as HFWhisper

    from colossalai_amd.models.whisper import WhisperForConditionalGeneration, hf_whisper_to_native

    torch.manual_seed(0)
    hf_cfg, cfg = _cfgs()
    hf = HFWhisper(hf_cfg).eval()
    native = WhisperForConditionalGeneration(cfg).eval()
    missing, unexpected = native.load_state_dict(hf_whisper_to_native(hf.state_dict()), strict=False)
    assert not missing, missing
    assert not unexpected, unexpected

    feats = torch.randn(2, 16, 64)  # [B, mels, T] -> encoder S = 32
    y = torch.randint(0, 256, (2, 12))
    with torch.no_grad():
        ref = hf(input_features=feats, labels=y)
        out = native(input_features=feats, labels=y)
    torch.testing.assert_close(out["logits"], ref.logits, rtol=2e-3, atol=2e-3)
    torch.testing.assert_close(out["loss"], ref.loss, rtol=1e-3, atol=1e-4)


def test_whisper_train_step():
    from colossalai_amd.models.whisper import WhisperForConditionalGeneration

    torch.manual_seed(0)
    _, cfg = _cfgs()
    m = WhisperForConditionalGeneration(cfg)
    feats = torch.randn(2, 16, 64)
    y = torch.randint(0, 256, (2, 12))
    out = m(input_features=feats, labels=y)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
