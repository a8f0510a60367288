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
from .multiagent import Mixer, MultiAgentConvNet, MultiAgentMLP, QMixer, VDNMixer
from .exploration import ConsistentDropout, NoisyLazyLinear, NoisyLinear, gSDEModule, reset_noise
from .model_based import (
    DreamerActor,
    ObsDecoder,
    ObsEncoder,
    RSSMPosterior,
    RSSMPrior,
    RSSMRollout,
    WorldModelWrapper,
)
try:
    from .decision_transformer import DecisionTransformer, DTActor
except ImportError:  # transformers not installed
    DecisionTransformer = DTActor = None
from .extras import (
    BatchRenorm1d,
    ConsistentDropout,
    ConsistentDropoutModule,
    Squeeze2dLayer,
    SqueezeLayer,
    SymExpTwoHot,
)
from .dreamer_v3 import (
    DreamerV3BlockGRU,
    DreamerV3BlockLinear,
    DreamerV3MLP,
    DreamerV3RMSNorm,
    RSSMPosteriorV3,
    RSSMPriorV3,
    RSSMRolloutV3,
)
from .decision_transformer import OnlineDTActor
from .llm_models import GPT2RewardModel, RewardModel
from .gp import ExactGPRegressor, GPWorldModel, RBFController
from .act import ACTModel
