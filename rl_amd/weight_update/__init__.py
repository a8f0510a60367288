from .weight_sync_schemes import (
    DistributedWeightSyncScheme,
    MultiProcessWeightSyncScheme,
    NoWeightSyncScheme,
    SharedMemWeightSyncScheme,
    WeightStrategy,
    WeightSyncScheme,
)
