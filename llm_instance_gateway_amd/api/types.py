"""Registry data model: the MI355X-native analogue of the reference CRD types.

Capability parity with the reference API plane (`api/v1alpha1/`):

  * ``InferenceModel`` — reference `inferencemodel_types.go:40-135`:
    model name exposed to clients, criticality tier, traffic-split across
    target model versions (base model or LoRA adapters), pool reference.
  * ``InferencePool`` — reference `inferencepool_types.go:26-46`:
    which serving engines belong to the pool.  On one MI355X node the
    reference's label-selector over pods becomes a GPU-set selector.
  * ``random_weighted_draw`` — reference `backend/datastore.go:78-98`
    (seedable weighted draw over TargetModels).
  * ``is_critical`` — reference `backend/datastore.go:100-105`.

Instead of K8s CRDs the registry is fed from YAML files (see
``llm_instance_gateway_amd.api.config``) or mutated through the admin API,
with the same field semantics.
"""

from __future__ import annotations

import enum
import random
from dataclasses import dataclass, field
from typing import List, Optional


class Criticality(str, enum.Enum):
    """Priority tier of an InferenceModel.

    Mirrors the reference's three-tier enum (`inferencemodel_types.go:78-95`):
    Critical traffic is never shed; Default/Sheddable traffic is dropped with
    429 when the pool is saturated (only Critical survives shedding in the
    scheduler — reference `scheduling/scheduler.go:74-90` treats everything
    non-critical as sheddable).
    """

    CRITICAL = "Critical"
    DEFAULT = "Default"
    SHEDDABLE = "Sheddable"


# Weight bounds for traffic-split entries (reference inferencemodel_types.go:
# TargetModel.Weight is 0..1e6).
MAX_TARGET_WEIGHT = 1_000_000


@dataclass
class TargetModel:
    """One arm of a traffic split: a concrete serveable model version.

    ``name`` is either the base model name or a registered LoRA adapter name;
    ``weight`` is the relative traffic share (reference
    `inferencemodel_types.go:119-135`).
    """

    name: str
    weight: int = 1

    def __post_init__(self) -> None:
        if not (0 <= self.weight <= MAX_TARGET_WEIGHT):
            raise ValueError(
                f"TargetModel weight {self.weight} outside [0, {MAX_TARGET_WEIGHT}]"
            )


@dataclass
class InferenceModel:
    """Client-visible model entry: routes ``model_name`` to weighted targets.

    Reference: `api/v1alpha1/inferencemodel_types.go:40-117`.
    """

    model_name: str
    criticality: Criticality = Criticality.DEFAULT
    target_models: List[TargetModel] = field(default_factory=list)
    pool_ref: str = ""

    @property
    def targets(self) -> List[TargetModel]:
        """Resolved traffic-split targets; defaults to the model itself."""
        if self.target_models:
            return self.target_models
        return [TargetModel(name=self.model_name, weight=1)]


@dataclass
class InferencePool:
    """A pool of serving engines.

    The reference selects backend pods by label + target port
    (`inferencepool_types.go:26-46`).  Here a pool selects local GPU engines
    by device index (``gpu_ids``) and/or remote engines by address
    (``engine_addresses``, kept for the scrape-based deployment mode).
    """

    name: str
    gpu_ids: List[int] = field(default_factory=list)
    engine_addresses: List[str] = field(default_factory=list)
    # Base model served by the pool's engines and its parallelism degree.
    base_model: str = ""
    tensor_parallel: int = 1
    # Explicit device names (overrides gpu_ids; lets CPU deployments/tests
    # drive runtime membership reconciliation the same way).
    devices: List[str] = field(default_factory=list)
    # Engine quantization mode: "" (bf16) or "fp8" (e4m3 projection GEMMs,
    # models/quant.py — vLLM --quantization fp8 analogue; dense models only).
    quant: str = ""


def random_weighted_draw(model: InferenceModel, seed: int = 0) -> str:
    """Draw one target model name according to the traffic-split weights.

    Semantics-compatible with the reference `RandomWeightedDraw`
    (`backend/datastore.go:78-98`): total the integer weights, draw uniformly
    in ``[0, total)``, walk the list subtracting.  ``seed > 0`` makes the draw
    deterministic (the reference seeds its PRNG the same way for tests).
    """
    targets = model.targets
    total = sum(t.weight for t in targets)
    if total <= 0:
        return ""
    rng = random.Random(seed) if seed > 0 else random
    val = rng.randrange(total)
    for t in targets:
        if val < t.weight:
            return t.name
        val -= t.weight
    return ""


def is_critical(model: Optional[InferenceModel]) -> bool:
    """Reference `backend/datastore.go:100-105`."""
    return model is not None and model.criticality == Criticality.CRITICAL
