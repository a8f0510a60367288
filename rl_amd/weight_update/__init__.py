from .weight_sync_schemes import (
    DistributedWeightSyncScheme,
    MultiProcessWeightSyncScheme,
    NoWeightSyncScheme,
    RPCWeightSyncScheme,
    SharedMemWeightSyncScheme,
    WeightReceiver,
    WeightSender,
    WeightStrategy,
    WeightSyncScheme,
    rpc_register_model,
)
from .llm import (
    LLMCollectiveWeightSyncScheme,
    LLMDoubleBufferWeightSyncScheme,
    get_model_metadata,
)
