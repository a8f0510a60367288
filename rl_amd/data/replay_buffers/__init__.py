from .buffers import (
    PrioritizedReplayBuffer,
    RemoteTensorDictReplayBuffer,
    ReplayBuffer,
    ReplayBufferEnsemble,
    TensorDictPrioritizedReplayBuffer,
    TensorDictReplayBuffer,
)
from .samplers import (
    PrioritizedSampler,
    PrioritizedSliceSampler,
    RandomSampler,
    Sampler,
    SamplerWithoutReplacement,
    SliceSampler,
    SliceSamplerWithoutReplacement,
)
from .segment_tree import MinSegmentTree, SumSegmentTree
from .storages import (
    LazyMemmapStorage,
    LazyStackStorage,
    LazyTensorStorage,
    ListStorage,
    Storage,
    StorageEnsemble,
    TensorStorage,
)
from .writers import (
    ImmutableDatasetWriter,
    RoundRobinWriter,
    TensorDictMaxValueWriter,
    TensorDictRoundRobinWriter,
    Writer,
)
from .checkpointers import (
    FlatStorageCheckpointer,
    H5StorageCheckpointer,
    ListStorageCheckpointer,
    NestedStorageCheckpointer,
    StorageCheckpointerBase,
    TensorStorageCheckpointer,
)
from .extras import (
    CompressedListStorage,
    ConsumingSampler,
    LinearScheduler,
    ParameterScheduler,
    PromptGroupSampler,
    StalenessAwareSampler,
    StepScheduler,
    LambdaScheduler,
    SchedulerList,
)
