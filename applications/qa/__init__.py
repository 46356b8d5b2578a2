from .retrieval_qa import RetrievalQA, TfidfRetriever

__all__ = ["RetrievalQA", "TfidfRetriever"]
