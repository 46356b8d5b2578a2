"""Offline dataset adapters (reference: applications/ColossalEval/
colossal_eval/dataset/*.py — the loader layer, reduced to the two on-disk
formats everything there parses into: 4-choice CSV (MMLU/CMMLU/C-Eval
style) and JSONL QA (GSM8K/LongBench style). Loaders emit the example
dicts ``evaluate_multiple_choice`` / ``evaluate_generation`` consume;
``tokenize`` is any ids-producing callable so no network tokenizer is
required."""

import csv
import json
from pathlib import Path
from typing import Callable, Dict, List, Sequence

__all__ = ["load_choice_csv", "load_qa_jsonl", "build_choice_examples",
           "build_generation_examples"]

_LETTERS = "ABCDEFGH"


def load_choice_csv(path, has_header: bool = False) -> List[Dict]:
    """Rows ``question, choice_a..choice_n, answer_letter`` →
    [{"question", "choices": [str], "answer": int}]."""
    rows = []
    with open(path, newline="", encoding="utf-8") as f:
        reader = csv.reader(f)
        for i, row in enumerate(reader):
            if has_header and i == 0:
                continue
            if len(row) < 3:
                continue
            q, *choices, ans = [c.strip() for c in row]
            rows.append({"question": q, "choices": choices,
                         "answer": _LETTERS.index(ans.upper())})
    return rows


def load_qa_jsonl(path, question_key: str = "question", answer_key: str = "answer") -> List[Dict]:
    rows = []
    with open(path, encoding="utf-8") as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            obj = json.loads(line)
            rows.append({"question": str(obj[question_key]), "answer": str(obj[answer_key])})
    return rows


def build_choice_examples(rows: Sequence[Dict], tokenize: Callable[[str], List[int]],
                          template: str = "{question}\nAnswer:") -> List[Dict]:
    """→ the ``evaluate_multiple_choice`` protocol: prompt ids + per-choice
    continuation ids + gold index."""
    out = []
    for r in rows:
        prompt = tokenize(template.format(question=r["question"]))
        out.append({
            "prompt": prompt,
            "choices": [tokenize(" " + c) for c in r["choices"]],
            "answer": r["answer"],
        })
    return out


def build_generation_examples(rows: Sequence[Dict], tokenize: Callable[[str], List[int]],
                              template: str = "Question: {question}\nAnswer:") -> List[Dict]:
    out = []
    for r in rows:
        out.append({"prompt": tokenize(template.format(question=r["question"])),
                    "reference": r["answer"]})
    return out
