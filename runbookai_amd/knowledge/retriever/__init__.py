from .default import KnowledgeRetriever, create_retriever
from .hybrid import HybridRetriever, reciprocal_rank_fusion

__all__ = ["KnowledgeRetriever", "create_retriever", "HybridRetriever", "reciprocal_rank_fusion"]
