from .weight_sync_schemes import (
    DistributedWeightSyncScheme,
    MultiProcessWeightSyncScheme,
    NoWeightSyncScheme,
    SharedMemWeightSyncScheme,
    WeightStrategy,
    WeightSyncScheme,
)
from .llm import (
    LLMCollectiveWeightSyncScheme,
    LLMDoubleBufferWeightSyncScheme,
    get_model_metadata,
)
