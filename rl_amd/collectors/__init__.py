from .collectors import BaseCollector, Collector, SyncDataCollector
from .multi import (
    AsyncCollector,
    MultiAsyncCollector,
    MultiSyncCollector,
    MultiSyncDataCollector,
    MultiaSyncDataCollector,
    _Interruptor,
)
from .utils import split_trajectories
from .distributed import DistributedCollector, DistributedSyncCollector
from .evaluator import Evaluator
from .llm import LLMCollector
from .weight_update import (
    DistributedWeightUpdater,
    RPCWeightUpdater,
    RayWeightUpdater,
    MultiProcessedWeightUpdater,
    RemoteModuleWeightUpdater,
    VanillaWeightUpdater,
    WeightUpdaterBase,
)
from .async_batched import AsyncBatchedCollector
from .profiling import ProfileConfig, ProfilerHook, enable_profile
from .graph import GraphedRollout
from .rpc import RPCCollector
from .ray import RayCollector, RayLLMCollector
from .utils import RandomPolicy, split_trajectories
from .multi import _MultiCollectorBase as MultiCollector
