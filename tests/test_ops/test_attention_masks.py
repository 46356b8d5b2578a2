"""CPU semantics of the padded / varlen attention dispatcher (fp32 fallback
path). The HIP kernels are oracle-tested against the same semantics in
test_kernels_gpu.py. Reference mask surface:
colossalai/shardformer/layer/attn.py:139 (CAUSAL/PADDED/PADDED_CAUSAL +
varlen packing)."""

import math

import pytest
import torch

from colossalai_amd.ops import attention_ref, flash_attention, flash_attention_varlen
from colossalai_amd.ops.attention import seqlens_from_attention_mask


def test_seqlens_from_mask():
    am = torch.tensor([[1, 1, 1, 0], [1, 1, 1, 1], [1, 0, 0, 0]])
    assert seqlens_from_attention_mask(am).tolist() == [3, 4, 1]
    with pytest.raises(ValueError):
        seqlens_from_attention_mask(torch.tensor([[1, 0, 1, 0]]))


@pytest.mark.parametrize("causal", [True, False])
def test_padded_matches_per_sequence(causal):
    """Padded-batch attention == running each unpadded sequence alone."""
    torch.manual_seed(0)
    B, S, H, D = 3, 32, 2, 64
    seqlens = torch.tensor([32, 17, 9], dtype=torch.int32)
    q = torch.randn(B, S, H, D, requires_grad=True)
    k = torch.randn(B, S, H, D, requires_grad=True)
    v = torch.randn(B, S, H, D, requires_grad=True)

    out = flash_attention(q, k, v, causal=causal, seqlens=seqlens)
    loss_mask = (torch.arange(S).unsqueeze(0) < seqlens.unsqueeze(1)).float().view(B, S, 1, 1)
    (out * loss_mask).sum().backward()

    for b in range(B):
        L = int(seqlens[b])
        qs = q.detach()[b : b + 1, :L].requires_grad_(True)
        ks = k.detach()[b : b + 1, :L].requires_grad_(True)
        vs = v.detach()[b : b + 1, :L].requires_grad_(True)
        ref = attention_ref(qs, ks, vs, causal=causal)
        ref.sum().backward()
        torch.testing.assert_close(out[b, :L], ref[0], rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(q.grad[b, :L], qs.grad[0], rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(k.grad[b, :L], ks.grad[0], rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(v.grad[b, :L], vs.grad[0], rtol=1e-4, atol=1e-5)
        # pad rows produce zero output and zero grads
        if L < S:
            assert out[b, L:].abs().sum() == 0
            assert q.grad[b, L:].abs().sum() == 0


def test_varlen_matches_per_sequence():
    torch.manual_seed(1)
    H, D = 2, 64
    lens = [19, 32, 5]
    total = sum(lens)
    cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)), dtype=torch.int32)
    q = torch.randn(total, H, D, requires_grad=True)
    k = torch.randn(total, H, D, requires_grad=True)
    v = torch.randn(total, H, D, requires_grad=True)

    out = flash_attention_varlen(q, k, v, cu, causal=True)
    out.sum().backward()

    off = 0
    for L in lens:
        qs = q.detach()[off : off + L].unsqueeze(0).requires_grad_(True)
        ks = k.detach()[off : off + L].unsqueeze(0).requires_grad_(True)
        vs = v.detach()[off : off + L].unsqueeze(0).requires_grad_(True)
        ref = attention_ref(qs, ks, vs, causal=True)
        ref.sum().backward()
        torch.testing.assert_close(out[off : off + L], ref[0], rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(q.grad[off : off + L], qs.grad[0], rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(v.grad[off : off + L], vs.grad[0], rtol=1e-4, atol=1e-5)
        off += L


def test_model_padded_batch_loss():
    """LlamaForCausalLM with a right-padded attention_mask: loss over valid
    tokens matches running each sequence unpadded (SFT data shape)."""
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(3)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    model = LlamaForCausalLM(cfg).float()
    B, S = 2, 24
    lens = [24, 11]
    x = torch.randint(0, 128, (B, S))
    am = torch.zeros(B, S, dtype=torch.long)
    labels = torch.full((B, S), -100, dtype=torch.long)
    for b, L in enumerate(lens):
        am[b, :L] = 1
        labels[b, :L] = x[b, :L]

    out = model(input_ids=x, labels=labels, attention_mask=am)

    # per-sequence reference losses (token-summed, then averaged like CE mean)
    tot_loss, tot_tok = 0.0, 0
    for b, L in enumerate(lens):
        ref = model(input_ids=x[b : b + 1, :L], labels=x[b : b + 1, :L])
        n_tok = L - 1  # shifted CE
        tot_loss += float(ref["loss"]) * n_tok
        tot_tok += n_tok
    torch.testing.assert_close(out["loss"], torch.tensor(tot_loss / tot_tok), rtol=1e-4, atol=1e-5)


def test_model_packed_varlen_loss():
    """Packed cu_seqlens batch == per-sequence losses (token-weighted):
    the varlen path through LlamaForCausalLM + the SFT packer."""
    import sys

    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM

    sys.path.insert(0, "applications")
    from chat.packing import pack_sft_samples

    torch.manual_seed(4)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    model = LlamaForCausalLM(cfg).float()
    lens = [13, 24, 7]
    samples = [{"input_ids": torch.randint(0, 128, (L,)), "labels": None} for L in lens]
    for s in samples:
        s["labels"] = s["input_ids"].clone()
    batches = pack_sft_samples(samples, max_tokens=64)
    assert len(batches) == 1 and batches[0]["cu_seqlens"].tolist() == [0, 13, 37, 44]

    out = model(**batches[0])
    tot, n = 0.0, 0
    for s, L in zip(samples, lens):
        ref = model(input_ids=s["input_ids"].unsqueeze(0), labels=s["labels"].unsqueeze(0))
        tot += float(ref["loss"]) * (L - 1)
        n += L - 1
    torch.testing.assert_close(out["loss"], torch.tensor(tot / n), rtol=1e-4, atol=1e-5)
