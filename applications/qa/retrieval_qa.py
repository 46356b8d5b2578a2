"""Retrieval-augmented QA (reference: applications/ColossalQA — re-scoped:
the reference stacks langchain + external vector stores; this stack keeps
the same shape — retriever + prompt assembly + LLM — with a local TF-IDF
retriever (scikit-learn) and the native inference engine, so it runs
offline on one box)."""

from dataclasses import dataclass
from typing import Callable, List, Optional, Sequence, Tuple

__all__ = ["TfidfRetriever", "RetrievalQA"]


class TfidfRetriever:
    """Cosine-similarity retrieval over TF-IDF vectors of a document set."""

    def __init__(self, documents: Sequence[str]):
        from sklearn.feature_extraction.text import TfidfVectorizer

        self.documents = list(documents)
        self._vec = TfidfVectorizer()
        self._mat = self._vec.fit_transform(self.documents)

    def retrieve(self, query: str, k: int = 3) -> List[Tuple[str, float]]:
        import numpy as np

        q = self._vec.transform([query])
        scores = (self._mat @ q.T).toarray().ravel()
        top = np.argsort(-scores)[:k]
        return [(self.documents[i], float(scores[i])) for i in top if scores[i] > 0]


@dataclass
class RetrievalQA:
    """retrieve -> assemble grounded prompt -> generate.

    ``generate_fn(prompt: str) -> str`` decouples the pipeline from the
    serving stack: pass ``LLMEngine.generate`` + a tokenizer round-trip, or
    any callable (tests use an echo model).
    """

    retriever: TfidfRetriever
    generate_fn: Callable[[str], str]
    k: int = 3
    template: str = (
        "Use the context to answer the question.\n"
        "{context}\n"
        "Question: {question}\nAnswer:"
    )

    def build_prompt(self, question: str) -> str:
        hits = self.retriever.retrieve(question, self.k)
        context = "\n".join(f"[{i + 1}] {doc}" for i, (doc, _) in enumerate(hits))
        return self.template.format(context=context, question=question)

    def answer(self, question: str) -> str:
        return self.generate_fn(self.build_prompt(question))
