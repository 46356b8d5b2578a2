"""Text metrics for generation-based evaluation (reference:
applications/ColossalEval/colossal_eval/evaluate/dataset_evaluator/metrics.py
— the offline subset: exact match, token F1, single-choice extraction and
GSM-style numeric answer matching; the GPT-judge metrics need an API and
are out of scope offline)."""

import re
import string
from collections import Counter
from typing import List

__all__ = ["normalize_text", "exact_match", "f1_score", "first_choice",
           "extract_numeric_answer", "numeric_match"]

_GSM_FINAL = re.compile(r"####\s*(-?[0-9][0-9.,]*)")
_NUMBER = re.compile(r"-?\d[\d,]*\.?\d*")


def normalize_text(s: str) -> str:
    """Lowercase, strip punctuation/articles/extra whitespace (SQuAD norm)."""
    s = s.lower()
    s = "".join(ch for ch in s if ch not in string.punctuation)
    s = re.sub(r"\b(a|an|the)\b", " ", s)
    return " ".join(s.split())


def exact_match(prediction: str, reference: str) -> float:
    return float(normalize_text(prediction) == normalize_text(reference))


def f1_score(prediction: str, reference: str) -> float:
    """Token-overlap F1 between normalized texts."""
    p = normalize_text(prediction).split()
    r = normalize_text(reference).split()
    if not p or not r:
        return float(p == r)
    common = Counter(p) & Counter(r)
    overlap = sum(common.values())
    if overlap == 0:
        return 0.0
    precision = overlap / len(p)
    recall = overlap / len(r)
    return 2 * precision * recall / (precision + recall)


def first_choice(prediction: str, choices: str = "ABCD") -> str:
    """First standalone choice letter in a model response ('' if none)."""
    m = re.search(rf"\b([{choices}])\b", prediction.upper())
    return m.group(1) if m else ""


def extract_numeric_answer(text: str) -> str:
    """GSM8K protocol: the '#### N' answer if present, else the LAST number."""
    m = _GSM_FINAL.search(text)
    if m:
        return m.group(1).replace(",", "")
    nums: List[str] = _NUMBER.findall(text)
    return nums[-1].replace(",", "") if nums else ""


def numeric_match(prediction: str, reference: str) -> float:
    a, b = extract_numeric_answer(prediction), extract_numeric_answer(reference)
    if not a or not b:
        return 0.0
    try:
        return float(abs(float(a) - float(b)) < 1e-6)
    except ValueError:
        return float(a == b)
