"""Generation-based evaluation: greedy decode + text metric
(reference: applications/ColossalEval/colossal_eval/evaluate/
dataset_evaluator/dataset_evaluator.py generation metrics path).

``greedy_generate`` runs the plain model API (full-context re-forward —
tiny eval batches; the serving engines in colossalai_amd/inference own the
KV-cached path)."""

from typing import Callable, Dict, List, Optional, Sequence

import torch

__all__ = ["greedy_generate", "evaluate_generation"]


@torch.no_grad()
def greedy_generate(model, prompt_ids: List[int], max_new_tokens: int = 32,
                    eos_token_id: Optional[int] = None) -> List[int]:
    model.eval()
    device = next(p for p in model.parameters() if p.numel() > 0).device
    ids = list(prompt_ids)
    for _ in range(max_new_tokens):
        logits = model(input_ids=torch.tensor([ids], device=device))["logits"]
        nxt = int(logits[0, -1].argmax())
        ids.append(nxt)
        if eos_token_id is not None and nxt == eos_token_id:
            break
    return ids[len(prompt_ids):]


@torch.no_grad()
def evaluate_generation(model, examples: Sequence[Dict], detokenize: Callable[[List[int]], str],
                        metric: Callable[[str, str], float], max_new_tokens: int = 32,
                        eos_token_id: Optional[int] = None) -> Dict[str, float]:
    """examples: [{"prompt": [ids], "reference": str}] → {score, n}."""
    total = 0.0
    for ex in examples:
        out = greedy_generate(model, ex["prompt"], max_new_tokens, eos_token_id)
        total += metric(detokenize(out), ex["reference"])
    return {"score": total / max(len(examples), 1), "n": len(examples)}
