from .frontdoor import build_app
from .pool import EnginePool, EngineSpec, FinishedRequest, PoolRequest, RequestTicket

__all__ = ["EnginePool", "EngineSpec", "FinishedRequest", "PoolRequest", "RequestTicket", "build_app"]
