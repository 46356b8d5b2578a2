"""Native GPT-J vs HF transformers parity (CPU) + train step."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_native_gptj_matches_hf():
    from transformers import GPTJConfig as HFConfig
    from transformers import GPTJForCausalLM as HFGPTJ

    from colossalai_amd.models.gptj import GPTJConfig, GPTJForCausalLM, hf_gptj_to_native

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=256, n_embd=64, n_layer=2, n_head=4, rotary_dim=8,
                      n_positions=64, resid_pdrop=0.0, embd_pdrop=0.0, attn_pdrop=0.0,
                      attn_implementation="eager")
    hf = HFGPTJ(hf_cfg).eval()
    native = GPTJForCausalLM(GPTJConfig(vocab_size=256, n_embd=64, n_layer=2, n_head=4,
                                        rotary_dim=8, n_positions=64)).eval()
    missing, unexpected = native.load_state_dict(hf_gptj_to_native(hf.state_dict()), strict=False)
    assert not missing, missing
    assert not unexpected, unexpected

    x = torch.randint(0, 256, (2, 24))
    with torch.no_grad():
        ref = hf(x).logits
        out = native(x)["logits"]
    torch.testing.assert_close(out, ref, rtol=2e-3, atol=2e-3)


def test_gptj_train_step():
    from colossalai_amd.models.gptj import GPTJConfig, GPTJForCausalLM

    torch.manual_seed(0)
    m = GPTJForCausalLM(GPTJConfig(vocab_size=256, n_embd=64, n_layer=2, n_head=4, rotary_dim=8,
                                   n_positions=64))
    x = torch.randint(0, 256, (2, 24))
    out = m(x, labels=x)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
