"""In-process registry/state store.

Reference: `pkg/ext-proc/backend/datastore.go` (K8sDatastore).  The reference
caches the InferencePool CRD, a sync.Map of InferenceModels keyed by client
ModelName, and the discovered pod set.  Here the same three maps live behind
one lock; mutation comes from the config reconcilers
(`llm_instance_gateway_amd.api.config`) or the admin API instead of K8s watch
events.
"""

from __future__ import annotations

import threading
from typing import Dict, List, Optional

from ..api.types import InferenceModel, InferencePool
from .types import Engine


class Datastore:
    """Thread-safe store of pool / models / engines.

    Reference `datastore.go:26-76`: RWMutex'd pool + sync.Map models (keyed by
    ModelName) + sync.Map pods.
    """

    def __init__(
        self,
        pool: Optional[InferencePool] = None,
        models: Optional[Dict[str, InferenceModel]] = None,
        engines: Optional[List[Engine]] = None,
    ) -> None:
        self._lock = threading.RLock()
        self._pool = pool
        self._models: Dict[str, InferenceModel] = dict(models or {})
        self._engines: Dict[str, Engine] = {e.name: e for e in (engines or [])}

    # -- pool (reference datastore.go:46-59) --------------------------------
    def set_pool(self, pool: InferencePool) -> None:
        with self._lock:
            self._pool = pool

    def get_pool(self) -> Optional[InferencePool]:
        with self._lock:
            return self._pool

    def has_synced_pool(self) -> bool:
        """Reference gates endpoint updates on the pool existing
        (endpointslice_reconciler.go:82-89)."""
        with self._lock:
            return self._pool is not None

    # -- models (reference datastore.go:61-76) ------------------------------
    def store_model(self, model: InferenceModel) -> None:
        with self._lock:
            self._models[model.model_name] = model

    def delete_model(self, model_name: str) -> None:
        with self._lock:
            self._models.pop(model_name, None)

    def fetch_model(self, model_name: str) -> Optional[InferenceModel]:
        """Reference `FetchModelData` (datastore.go:70-76)."""
        with self._lock:
            return self._models.get(model_name)

    def all_models(self) -> List[InferenceModel]:
        with self._lock:
            return list(self._models.values())

    # -- engines (reference datastore.go pods sync.Map + GetPodIPs) ---------
    def add_engine(self, engine: Engine) -> None:
        with self._lock:
            self._engines[engine.name] = engine

    def remove_engine(self, name: str) -> None:
        with self._lock:
            self._engines.pop(name, None)

    def get_engines(self) -> List[Engine]:
        with self._lock:
            return list(self._engines.values())

    def get_engine(self, name: str) -> Optional[Engine]:
        with self._lock:
            return self._engines.get(name)
