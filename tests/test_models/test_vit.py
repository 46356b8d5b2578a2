"""Native ViT vs HF transformers parity (CPU) + train-step smoke."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_native_vit_matches_hf():
    from transformers import ViTConfig as HFConfig
    from transformers import ViTForImageClassification as HFViT

    from colossalai_amd.models.vit import ViTConfig, ViTForImageClassification, hf_vit_to_native

    torch.manual_seed(0)
    hf_cfg = HFConfig(image_size=32, patch_size=8, num_channels=3, hidden_size=64,
                      num_hidden_layers=2, num_attention_heads=4, intermediate_size=128,
                      hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
                      num_labels=5, attn_implementation="eager")
    hf = HFViT(hf_cfg).eval()
    native = ViTForImageClassification(ViTConfig(image_size=32, patch_size=8, hidden_size=64,
                                                 num_hidden_layers=2, num_attention_heads=4,
                                                 intermediate_size=128, num_labels=5)).eval()
    missing, unexpected = native.load_state_dict(hf_vit_to_native(hf.state_dict()), strict=False)
    assert not missing, missing
    assert not unexpected, unexpected

    x = torch.randn(2, 3, 32, 32)
    y = torch.randint(0, 5, (2,))
    with torch.no_grad():
        ref = hf(x, labels=y)
        out = native(x, labels=y)
    torch.testing.assert_close(out["logits"], ref.logits, rtol=2e-3, atol=2e-3)
    torch.testing.assert_close(out["loss"], ref.loss, rtol=1e-3, atol=1e-4)


def test_vit_train_step():
    from colossalai_amd.models.vit import ViTConfig, ViTForImageClassification

    torch.manual_seed(0)
    m = ViTForImageClassification(ViTConfig(image_size=32, patch_size=8, hidden_size=64,
                                            num_hidden_layers=2, num_attention_heads=4,
                                            intermediate_size=128, num_labels=5))
    x = torch.randn(2, 3, 32, 32)
    y = torch.randint(0, 5, (2,))
    out = m(x, labels=y)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
