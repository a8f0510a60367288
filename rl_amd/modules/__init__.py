from .models import (
    MLP,
    Conv3dNet,
    ConvNet,
    DdpgCnnActor,
    DdpgCnnQNet,
    DdpgMlpActor,
    DdpgMlpQNet,
    DistributionalDQNnet,
    DuelingCnnDQNet,
    DuelingMlpDQNet,
    NormalParamExtractor,
)
from .distributions import (
    Delta,
    IndependentNormal,
    MaskedCategorical,
    MaskedOneHotCategorical,
    OneHotCategorical,
    Ordinal,
    TanhDelta,
    TanhNormal,
    TruncatedNormal,
)
from .tensordict_module import (
    Actor,
    ActorCriticOperator,
    ActorCriticWrapper,
    ActorValueOperator,
    AdditiveGaussianModule,
    DistributionalQValueActor,
    DistributionalQValueModule,
    EGreedyModule,
    MultiStepActorWrapper,
    OrnsteinUhlenbeckProcessModule,
    ProbabilisticActor,
    QValueActor,
    QValueModule,
    TanhModule,
    ValueOperator,
)
from .tensordict_module import (
    GRUCell,
    GRUModule,
    LSTMCell,
    LSTMModule,
    gru_scan,
    lstm_scan,
    recurrent_mode,
    set_recurrent_mode,
)
from .models import Mixer, MultiAgentConvNet, MultiAgentMLP, QMixer, VDNMixer
from .models import ConsistentDropout, NoisyLazyLinear, NoisyLinear, gSDEModule, reset_noise
from .planners import CEMPlanner, MPCPlannerBase, MPPIPlanner
from .inference_server import (
    InferenceServer,
    PolicyClient,
    ProcessInferenceServer,
    SlotPolicyClient,
)
from .llm import LLMWrapperBase, TransformersWrapper
from .models import (
    DTActor,
    DecisionTransformer,
    DreamerActor,
    ObsDecoder,
    ObsEncoder,
    RSSMPosterior,
    RSSMPrior,
    RSSMRollout,
    WorldModelWrapper,
)
from .tensordict_module import DecisionTransformerInferenceWrapper, LMHeadActorValueOperator
from .tensordict_module import SafeModule, SafeProbabilisticModule, SafeSequential
from .distributions import LLMMaskedCategorical
from .models import (
    BatchRenorm1d,
    ConsistentDropout,
    ConsistentDropoutModule,
    Squeeze2dLayer,
    SqueezeLayer,
    SymExpTwoHot,
)
from . import functional
from .models import (
    DreamerV3BlockGRU,
    DreamerV3MLP,
    DreamerV3RMSNorm,
    RSSMPosteriorV3,
    RSSMPriorV3,
    RSSMRolloutV3,
)
from .models import GPT2RewardModel, OnlineDTActor, RewardModel
from .models import ExactGPRegressor, GPWorldModel, RBFController
from .models import ACTModel
from .value_norm import PopArtValueNorm, RunningValueNorm, ValueNorm
from .vla import LeRobotPolicyWrapper, TinyVLA
from .value_transforms import (
    ComposeValueTransform,
    IdentityValueTransform,
    SignedHyperbolicValueTransform,
    SymLogValueTransform,
    ValueTransform,
)
from .legacy import (
    AdditiveGaussianWrapper,
    DistributionalQValueHook,
    EGreedyWrapper,
    GRU,
    GRUBase,
    LSTM,
    LSTMBase,
    NormalParamWrapper,
    OrnsteinUhlenbeckProcessWrapper,
    QValueHook,
    WorldModel,
)
from .models.exploration import LazygSDEModule

# reference-parity re-exports (torchrl subpackage-level __all__)
from .functional import (  # noqa: F401
    symexp,
    symlog,
)
from .value_transforms import (  # noqa: F401
    signed_hyperbolic,
    signed_parabolic,
)
from rl_amd.collectors import (  # noqa: F401
    RandomPolicy,
)
from rl_amd.data.map import (  # noqa: F401
    EXP3Score,
    PUCTScore,
    UCBScore,
)
from rl_amd.objectives.act import DiffusionActor  # noqa: F401
__all__ = sorted(set(list(globals().get('__all__', [])) + ['DiffusionActor', 'EXP3Score', 'PUCTScore', 'RandomPolicy', 'UCBScore', 'signed_hyperbolic', 'signed_parabolic', 'symexp', 'symlog']))

# reference-parity: MCTS scores, multi-agent/cross-group critics, RNN +
# primer utilities, discrete-distribution extras
from .extras import (  # noqa: F401
    AutocastPolicy,
    CrossCriticGroupSpec,
    CrossGroupCritic,
    MCTSScore,
    MCTSScores,
    MultiAgentNetBase,
    RecurrentMatmulPrecision,
    RecurrentMatmulPrecisionUserMode,
    UCB1TunedScore,
    VLAWrapperBase,
    VmapModule,
    canonicalize_rnn_subset,
    get_env_transforms_from_module,
    get_primers_from_module,
    get_recurrent_matmul_precision,
    set_exploration_modules_spec_from_env,
    set_recurrent_matmul_precision,
)
from .distributions.discrete import OneHotOrdinal, ReparamGradientStrategy  # noqa: F401
from ..tensordict.nn import (  # noqa: F401
    ProbabilisticTensorDictSequential as SafeProbabilisticTensorDictSequential,
)

def distributions_maps(distribution_class: str):
    """Map a distribution name to its class (reference
    distributions/__init__.py distributions_maps)."""
    from . import distributions as _d

    maps = {
        "delta": getattr(_d, "Delta", None),
        "tanhnormal": getattr(_d, "TanhNormal", None),
        "truncnormal": getattr(_d, "TruncatedNormal", None),
        "tanhdelta": getattr(_d, "TanhDelta", None),
        "onehotcategorical": getattr(_d, "OneHotCategorical", None),
        "categorical": getattr(_d, "MaskedCategorical", None),
        "ordinal": getattr(_d, "Ordinal", None),
        "onehotordinal": OneHotOrdinal,
    }
    key = distribution_class.lower().replace("_", "").replace("-", "")
    if key not in maps or maps[key] is None:
        raise NotImplementedError(f"unknown distribution {distribution_class!r}")
    return maps[key]

__all__ = sorted(set(__all__) | {
    "CrossCriticGroupSpec", "CrossGroupCritic", "MCTSScore", "MCTSScores",
    "MultiAgentNetBase", "RecurrentMatmulPrecision",
    "RecurrentMatmulPrecisionUserMode", "UCB1TunedScore", "VLAWrapperBase",
    "VmapModule", "canonicalize_rnn_subset", "get_env_transforms_from_module",
    "get_primers_from_module", "get_recurrent_matmul_precision",
    "set_exploration_modules_spec_from_env", "set_recurrent_matmul_precision",
    "OneHotOrdinal", "ReparamGradientStrategy",
    "SafeProbabilisticTensorDictSequential", "distributions_maps",
})
