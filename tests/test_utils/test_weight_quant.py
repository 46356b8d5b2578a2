"""Native weight-only quantization (reference: colossalai/quantization/bnb.py)."""

import torch

from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
from colossalai_amd.quantization import NF4Linear, W8Linear, quantize_model


def test_w8_roundtrip():
    torch.manual_seed(0)
    w = torch.randn(64, 32)
    q = W8Linear(w, torch.randn(64))
    deq = q.dequantize(torch.float32)
    # per-channel int8: error ~ scale/2 = absmax/254, plus fp16 scale rounding
    bound = w.abs().amax(dim=1, keepdim=True) / 200 + 1e-6
    assert ((deq - w).abs() <= bound).all()
    x = torch.randn(2, 32)
    assert q(x).shape == (2, 64)


def test_nf4_roundtrip():
    torch.manual_seed(1)
    w = torch.randn(48, 64)
    q = NF4Linear(w, None, block=64)
    deq = q.dequantize(torch.float32)
    assert q.qweight.numel() == 48 * 64 // 2  # 2 codes per byte
    # nf4 is coarse: check strong correlation, not tight error
    corr = torch.corrcoef(torch.stack([w.reshape(-1), deq.reshape(-1)]))[0, 1]
    assert corr > 0.98
    assert q(torch.randn(3, 64)).shape == (3, 48)


def test_quantize_model_logits_close():
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    model = LlamaForCausalLM(cfg).eval()
    x = torch.randint(0, 128, (2, 16))
    with torch.no_grad():
        ref = model(x)["logits"]
    q = quantize_model(model, bits=8)
    # lm_head skipped, attention/MLP linears swapped
    assert isinstance(q.model.layers[0].self_attn.qkv_proj, W8Linear)
    assert isinstance(q.lm_head, torch.nn.Linear)
    with torch.no_grad():
        out = q(x)["logits"]
    assert torch.allclose(out, ref, rtol=0.1, atol=0.2)
    # greedy argmax mostly preserved
    agree = (out.argmax(-1) == ref.argmax(-1)).float().mean()
    assert agree > 0.9


def test_quantize_model_nf4_runs():
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    model = LlamaForCausalLM(cfg).eval()
    q = quantize_model(model, bits=4)
    with torch.no_grad():
        out = q(torch.randint(0, 128, (2, 16)))["logits"]
    assert torch.isfinite(out).all()


def test_quantized_state_dict_roundtrip():
    """Quantized model state dicts restore exactly (buffers + bias)."""
    import copy

    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    qa = quantize_model(LlamaForCausalLM(cfg), bits=8).eval()
    torch.manual_seed(9)
    qb = quantize_model(LlamaForCausalLM(cfg), bits=8).eval()
    qb.load_state_dict(qa.state_dict())
    x = torch.randint(0, 128, (2, 12))
    with torch.no_grad():
        torch.testing.assert_close(qa(x)["logits"], qb(x)["logits"], rtol=0.0, atol=0.0)
