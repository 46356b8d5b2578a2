"""Model evaluation harness (reference: applications/ColossalEval — the
two primitives every task there reduces to, over the native model API).

- ``evaluate_perplexity``: token-level NLL / perplexity over a corpus of
  token sequences (sliding window for sequences beyond the context).
- ``evaluate_multiple_choice``: loglikelihood ranking — score each
  candidate continuation given the prompt, pick the argmax (the HellaSwag
  / ARC / MMLU protocol), optionally length-normalized.

Both run batched on any native causal LM (or a Booster-wrapped one).
"""

import math
from typing import Dict, List, Sequence

import torch
import torch.nn.functional as F

__all__ = ["evaluate_perplexity", "evaluate_multiple_choice", "sequence_loglikelihood"]


@torch.no_grad()
def sequence_loglikelihood(model, prompt: List[int], continuation: List[int]) -> float:
    """Sum log p(continuation | prompt) under the model."""
    ids = torch.tensor([prompt + continuation], device=_device(model))
    logits = model(input_ids=ids)["logits"][0].float()
    lp = F.log_softmax(logits[:-1], dim=-1)
    P = len(prompt)
    tgt = ids[0, 1:]
    picked = lp.gather(-1, tgt.unsqueeze(-1)).squeeze(-1)
    return float(picked[P - 1 :].sum())


def _device(model):
    return next(p for p in model.parameters() if p.numel() > 0).device


@torch.no_grad()
def evaluate_perplexity(model, sequences: Sequence[List[int]], max_len: int = 2048,
                        stride: int = None) -> Dict[str, float]:
    """-> {nll, ppl, tokens} over all sequences (teacher-forced)."""
    model.eval()
    stride = stride or max_len
    total_nll, total_tok = 0.0, 0
    dev = _device(model)
    for seq in sequences:
        covered = 1  # absolute index of the next unscored target token
        for lo in range(0, max(len(seq) - 1, 1), stride):
            window = seq[lo : lo + max_len]
            if len(window) < 2 or covered >= len(seq):
                break
            ids = torch.tensor([window], device=dev)
            logits = model(input_ids=ids)["logits"][0].float()
            lp = F.log_softmax(logits[:-1], dim=-1)
            tgt = ids[0, 1:]
            # window targets are absolute positions lo+1 .. lo+len(window)-1;
            # score only those not already covered by the previous window
            start = max(covered - (lo + 1), 0)
            picked = lp.gather(-1, tgt.unsqueeze(-1)).squeeze(-1)[start:]
            total_nll += float(-picked.sum())
            total_tok += picked.numel()
            covered = lo + len(window)
    nll = total_nll / max(total_tok, 1)
    return {"nll": nll, "ppl": math.exp(min(nll, 30.0)), "tokens": total_tok}


@torch.no_grad()
def evaluate_multiple_choice(model, examples: Sequence[dict], length_normalize: bool = True
                             ) -> Dict[str, float]:
    """examples: [{"prompt": [ids], "choices": [[ids], ...], "answer": int}]
    -> {accuracy, n}."""
    model.eval()
    correct = 0
    for ex in examples:
        scores = []
        for cont in ex["choices"]:
            s = sequence_loglikelihood(model, ex["prompt"], cont)
            scores.append(s / max(len(cont), 1) if length_normalize else s)
        if int(torch.tensor(scores).argmax()) == ex["answer"]:
            correct += 1
    return {"accuracy": correct / max(len(examples), 1), "n": len(examples)}
