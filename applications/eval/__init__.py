from .evaluator import evaluate_multiple_choice, evaluate_perplexity, sequence_loglikelihood

__all__ = ["evaluate_perplexity", "evaluate_multiple_choice", "sequence_loglikelihood"]
