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
