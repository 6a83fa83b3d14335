from .core import SimEngine, SimEnv, SimRequest, decode_delay, prefill_delay
from .router import ROUTING_POLICIES, SimRouter

__all__ = [
    "ROUTING_POLICIES",
    "SimEngine",
    "SimEnv",
    "SimRequest",
    "SimRouter",
    "decode_delay",
    "prefill_delay",
]
