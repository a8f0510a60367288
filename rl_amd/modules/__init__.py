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
from .inference_server import InferenceServer, PolicyClient
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
