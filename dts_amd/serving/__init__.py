from dts_amd.serving.engine import GenerationResult, LocalBackend, ServingEngine
from dts_amd.serving.kv_cache import BlockManager, KVCachePool
from dts_amd.serving.scheduler import Scheduler
from dts_amd.serving.sequence import Sequence, SeqStatus

__all__ = [
    "ServingEngine",
    "LocalBackend",
    "GenerationResult",
    "BlockManager",
    "KVCachePool",
    "Scheduler",
    "Sequence",
    "SeqStatus",
]
