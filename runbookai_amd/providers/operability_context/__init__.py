from .types import AgentChangeClaim, VerifiedChangeFact
from .reconcile import reconcile_claims, trust_score
from .factory import create_adapter

__all__ = ["AgentChangeClaim", "VerifiedChangeFact", "reconcile_claims", "trust_score",
           "create_adapter"]
