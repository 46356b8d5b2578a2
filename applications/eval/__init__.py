from .datasets import (
    build_choice_examples,
    build_generation_examples,
    load_choice_csv,
    load_qa_jsonl,
)
from .evaluator import evaluate_multiple_choice, evaluate_perplexity, sequence_loglikelihood
from .generation import evaluate_generation, greedy_generate
from .metrics import exact_match, extract_numeric_answer, f1_score, first_choice, numeric_match

__all__ = [
    "evaluate_perplexity", "evaluate_multiple_choice", "sequence_loglikelihood",
    "greedy_generate", "evaluate_generation",
    "exact_match", "f1_score", "first_choice", "extract_numeric_answer", "numeric_match",
    "load_choice_csv", "load_qa_jsonl", "build_choice_examples", "build_generation_examples",
]
