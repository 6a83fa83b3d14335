from .types import Engine, EngineMetrics, EngineState, LLMRequest
from .datastore import Datastore
from .scheduler import Scheduler, ResourceExhausted
from .provider import Provider, MetricsSource

__all__ = [
    "Engine",
    "EngineMetrics",
    "EngineState",
    "LLMRequest",
    "Datastore",
    "Scheduler",
    "ResourceExhausted",
    "Provider",
    "MetricsSource",
]
