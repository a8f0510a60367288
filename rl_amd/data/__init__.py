from .tensor_specs import (
    Binary,
    Bounded,
    Categorical,
    Choice,
    Composite,
    CompositeSpec,
    MultiCategorical,
    MultiOneHot,
    NonTensor,
    OneHot,
    TensorSpec,
    Unbounded,
    UnboundedContinuous,
    UnboundedDiscrete,
    stack_specs,
)
from .replay_buffers import (
    ImmutableDatasetWriter,
    LazyMemmapStorage,
    LazyStackStorage,
    LazyTensorStorage,
    ListStorage,
    MinSegmentTree,
    PrioritizedReplayBuffer,
    PrioritizedSampler,
    PrioritizedSliceSampler,
    RandomSampler,
    ReplayBuffer,
    ReplayBufferEnsemble,
    RoundRobinWriter,
    Sampler,
    SamplerWithoutReplacement,
    SliceSampler,
    SliceSamplerWithoutReplacement,
    Storage,
    StorageEnsemble,
    SumSegmentTree,
    TensorDictMaxValueWriter,
    TensorDictPrioritizedReplayBuffer,
    TensorDictReplayBuffer,
    TensorDictRoundRobinWriter,
    TensorStorage,
    Writer,
)
from .postprocs import DensifyReward, MultiStep
from .her import HERReplayBuffer, HindsightStrategy
from .map import (
    BinaryToDecimal,
    EXP3Score,
    HashToInt,
    MCTSForest,
    PUCTScore,
    QueryModule,
    RandomProjectionHash,
    SipHash,
    TensorDictMap,
    Tree,
    UCBScore,
)
from .llm import ContentBase, History
from .datasets import (
    AtariDQNExperienceReplay,
    BaseDatasetExperienceReplay,
    D4RLExperienceReplay,
    LocalHDF5ExperienceReplay,
    LocalMemmapExperienceReplay,
    MinariExperienceReplay,
    OpenXExperienceReplay,
    StreamingEpisodeStorage,
    convert_atari_shards,
    convert_d4rl_hdf5,
    convert_minari_hdf5,
    download_file,
)
from .replay_buffers import (
    CompressedListStorage,
    ConsumingSampler,
    LinearScheduler,
    ParameterScheduler,
    PromptGroupSampler,
    StalenessAwareSampler,
    StepScheduler,
)
from .offline_to_online import OfflineOnlineReplayBuffer
from .llm import AdaptiveKLController, ConstantKLController
from .offline_to_online import OfflineOnlineReplayBuffer as OfflineToOnlineReplayBuffer
from .llm.datasets import PairwisePreferenceDataset as PairwiseDataset
from .replay_buffers.checkpointers import (
    FlatStorageCheckpointer as TED2Flat,
    FlatStorageCheckpointer as Flat2TED,
    NestedStorageCheckpointer as TED2Nested,
    NestedStorageCheckpointer as Nested2TED,
)
from .llm import (
    PromptData,
    PromptTensorDictTokenizer,
    RewardData,
    RolloutFromModel,
    TensorDictTokenizer,
    TokenizedDatasetLoader,
    TopKRewardSelector,
)
from .video import VideoClipRef

# reference-parity re-exports (torchrl subpackage-level __all__)
from .replay_buffers import (  # noqa: F401
    FlatStorageCheckpointer,
    H5StorageCheckpointer,
    ListStorageCheckpointer,
    NestedStorageCheckpointer,
    RemoteTensorDictReplayBuffer,
    StorageCheckpointerBase,
    TensorStorageCheckpointer,
)
from .tensor_specs import (  # noqa: F401
    BoundedContinuous,
    DEVICE_TYPING,
)
from .vla import (  # noqa: F401
    ActionTokenizerBase,
    UniformActionTokenizer,
)
__all__ = sorted(set(list(globals().get('__all__', [])) + ['ActionTokenizerBase', 'BoundedContinuous', 'DEVICE_TYPING', 'FlatStorageCheckpointer', 'H5StorageCheckpointer', 'ListStorageCheckpointer', 'NestedStorageCheckpointer', 'RemoteTensorDictReplayBuffer', 'StorageCheckpointerBase', 'TensorStorageCheckpointer', 'UniformActionTokenizer']))

# reference-parity: sample units, trajectory queries, spec helpers,
# ensembles, VLA/video/llm data plumbing
from .extras import (  # noqa: F401
    ConditionalUpdateResult,
    DEFAULT_DONE_KEYS,
    H5Combine,
    H5Split,
    RayReplayBuffer,
    RobotDatasetMetadata,
    SampleUnit,
    SamplerEnsemble,
    Sequence,
    StorageEnsembleCheckpointer,
    StoreStorage,
    TensorMap,
    Trajectory,
    TrajectoryPredicate,
    Transition,
    VocabTailActionTokenizer,
    WriterEnsemble,
    check_no_exclusive_keys,
    clear_video_decoder_cache,
    consolidate_spec,
    contains_lazy_spec,
    create_infinite_iterator,
    filter_trajectories,
    find_start_stop_traj,
    get_dataloader,
    iter_trajectories,
    prefill_replay_buffer,
    set_video_decoder_cache_size,
    traj,
    validate_vla_tensordict,
)
from .tensor_specs import Stacked, StackedComposite  # noqa: F401
from .replay_buffers.checkpointers import CompressedListStorageCheckpointer  # noqa: F401
__all__ = sorted(set(__all__) | {
    "ConditionalUpdateResult", "DEFAULT_DONE_KEYS", "H5Combine", "H5Split",
    "RayReplayBuffer", "RobotDatasetMetadata", "SampleUnit", "SamplerEnsemble",
    "Sequence", "StorageEnsembleCheckpointer", "StoreStorage", "TensorMap",
    "Trajectory", "TrajectoryPredicate", "Transition", "VocabTailActionTokenizer",
    "WriterEnsemble", "check_no_exclusive_keys", "clear_video_decoder_cache",
    "consolidate_spec", "contains_lazy_spec", "create_infinite_iterator",
    "filter_trajectories", "find_start_stop_traj", "get_dataloader",
    "iter_trajectories", "prefill_replay_buffer", "set_video_decoder_cache_size",
    "traj", "validate_vla_tensordict", "Stacked", "StackedComposite",
    "CompressedListStorageCheckpointer",
})
