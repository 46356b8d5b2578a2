"""Eval harness: perplexity matches hand-computed CE; multiple-choice picks
the continuation the model actually prefers (train a preference in)."""

import math

import torch
import torch.nn.functional as F

from applications.eval import evaluate_multiple_choice, evaluate_perplexity, sequence_loglikelihood
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM


def _tiny():
    return LlamaConfig(vocab_size=64, hidden_size=32, intermediate_size=64, num_hidden_layers=2,
                       num_attention_heads=2, num_key_value_heads=2, max_position_embeddings=64)


def test_perplexity_matches_manual():
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny()).eval()
    seq = list(range(20))
    res = evaluate_perplexity(model, [seq], max_len=64)
    ids = torch.tensor([seq])
    with torch.no_grad():
        logits = model(input_ids=ids)["logits"][0].float()
    nll = F.cross_entropy(logits[:-1], ids[0, 1:]).item()
    assert abs(res["nll"] - nll) < 1e-5
    assert abs(res["ppl"] - math.exp(nll)) < 1e-3
    assert res["tokens"] == 19


def test_perplexity_sliding_window_covers_all_tokens():
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny()).eval()
    seq = list(range(50))
    res = evaluate_perplexity(model, [seq], max_len=32, stride=16)
    assert res["tokens"] == 49  # every target token scored exactly once


def test_multiple_choice_prefers_trained_continuation():
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny())
    prompt = [1, 2, 3]
    good, bad = [10, 11, 12], [40, 41, 42]
    opt = torch.optim.AdamW(model.parameters(), lr=5e-3)
    x = torch.tensor([prompt + good])
    for _ in range(30):
        out = model(input_ids=x, labels=x)
        opt.zero_grad(); out["loss"].backward(); opt.step()
    model.eval()
    assert sequence_loglikelihood(model, prompt, good) > sequence_loglikelihood(model, prompt, bad)
    res = evaluate_multiple_choice(model, [{"prompt": prompt, "choices": [bad, good], "answer": 1}])
    assert res["accuracy"] == 1.0
