from .types import (
    Criticality,
    TargetModel,
    InferenceModel,
    InferencePool,
    random_weighted_draw,
    is_critical,
)

__all__ = [
    "Criticality",
    "TargetModel",
    "InferenceModel",
    "InferencePool",
    "random_weighted_draw",
    "is_critical",
]
