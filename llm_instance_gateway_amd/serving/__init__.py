from .frontdoor import build_app
from .pool import EnginePool, FinishedRequest, PoolRequest, RequestTicket

__all__ = ["EnginePool", "FinishedRequest", "PoolRequest", "RequestTicket", "build_app"]
