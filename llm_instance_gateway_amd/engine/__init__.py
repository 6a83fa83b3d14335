from .config import EngineConfig
from .engine import LLMEngine
from .sequence import SamplingParams, Sequence, SequenceStatus
from .block_manager import BlockManager
from .scheduler import EngineScheduler, SchedulerOutput

__all__ = [
    "EngineConfig",
    "LLMEngine",
    "SamplingParams",
    "Sequence",
    "SequenceStatus",
    "BlockManager",
    "EngineScheduler",
    "SchedulerOutput",
]
