"""Native GPT-2 vs HF transformers parity (CPU)."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_native_gpt2_matches_hf():
    from transformers import GPT2Config as HFConfig
    from transformers import GPT2LMHeadModel as HFGPT2

    from colossalai_amd.models.gpt2 import GPT2Config, GPT2LMHeadModel, hf_gpt2_to_native

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=256, n_positions=64, n_embd=64, n_layer=2, n_head=4,
                      attn_implementation="eager")
    hf = HFGPT2(hf_cfg).eval()
    native = GPT2LMHeadModel(GPT2Config(vocab_size=256, n_positions=64, n_embd=64, n_layer=2, n_head=4)).eval()
    native.load_state_dict(hf_gpt2_to_native(hf.state_dict()), strict=True)

    x = torch.randint(0, 256, (2, 32))
    with torch.no_grad():
        ref = hf(x).logits
        out = native(x)["logits"]
    torch.testing.assert_close(out, ref, rtol=2e-3, atol=2e-3)


def test_gpt2_train_step():
    from colossalai_amd.models.gpt2 import GPT2Config, GPT2LMHeadModel

    torch.manual_seed(0)
    m = GPT2LMHeadModel(GPT2Config(vocab_size=256, n_positions=64, n_embd=64, n_layer=2, n_head=4))
    x = torch.randint(0, 256, (2, 32))
    out = m(x, labels=x)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
