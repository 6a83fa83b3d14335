"""Registry type tests (reference backend/datastore_test.go: seeded
RandomWeightedDraw determinism; api/v1alpha1 semantics)."""

import pytest

from llm_instance_gateway_amd.api.types import (
    Criticality,
    InferenceModel,
    TargetModel,
    is_critical,
    random_weighted_draw,
)


def test_weighted_draw_deterministic_with_seed():
    m = InferenceModel(
        model_name="base",
        target_models=[
            TargetModel("v1", 50),
            TargetModel("v2", 25),
            TargetModel("v3", 25),
        ],
    )
    first = random_weighted_draw(m, seed=420)
    for _ in range(10):
        assert random_weighted_draw(m, seed=420) == first


def test_weighted_draw_respects_weights():
    m = InferenceModel(
        model_name="base",
        target_models=[TargetModel("always", 10), TargetModel("never", 0)],
    )
    for s in range(1, 20):
        assert random_weighted_draw(m, seed=s) == "always"


def test_weighted_draw_distribution():
    m = InferenceModel(
        model_name="base",
        target_models=[TargetModel("a", 3), TargetModel("b", 1)],
    )
    counts = {"a": 0, "b": 0}
    for s in range(1, 2001):
        counts[random_weighted_draw(m, seed=s)] += 1
    assert 0.65 < counts["a"] / 2000 < 0.85


def test_targets_default_to_self():
    m = InferenceModel(model_name="solo")
    assert [t.name for t in m.targets] == ["solo"]


def test_is_critical():
    assert is_critical(InferenceModel(model_name="m", criticality=Criticality.CRITICAL))
    assert not is_critical(InferenceModel(model_name="m", criticality=Criticality.DEFAULT))
    assert not is_critical(None)


def test_weight_bounds():
    with pytest.raises(ValueError):
        TargetModel("x", weight=-1)
    with pytest.raises(ValueError):
        TargetModel("x", weight=1_000_001)
