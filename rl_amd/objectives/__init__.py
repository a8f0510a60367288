from .a2c import A2CLoss, ReinforceLoss
from .common import LossModule
from .ddpg import DDPGLoss
from .dqn import DistributionalDQNLoss, DQNLoss
from .ppo import ClipPPOLoss, KLPENPPOLoss, PPOLoss
from .sac import DiscreteSACLoss, SACLoss
from .td3 import TD3BCLoss, TD3Loss
from .utils import (
    HardUpdate,
    SoftUpdate,
    TargetNetUpdater,
    ValueEstimators,
    distance_loss,
    hold_out_net,
)
from .value import GAE, MultiAgentGAE, TD0Estimator, TD1Estimator, TDLambdaEstimator, ValueEstimatorBase, VTrace
from .cql import CQLLoss, DiscreteCQLLoss
from .iql import DiscreteIQLLoss, IQLLoss
from .imitation import BCLoss, DTLoss, GAILLoss, OnlineDTLoss, RNDLoss
from .multiagent import IPPOLoss, MAPPOLoss, QMixerLoss
from .redq import CrossQLoss, REDQLoss
from .llm import CISPOLoss, DAPO, DistillationLoss, GRPOLoss, MCAdvantage, SFTLoss
from .dreamer import DreamerActorLoss, DreamerModelLoss, DreamerValueLoss, WorldModelLoss
from .pilco import ExponentialQuadraticCost
from .dreamer_v3 import (
    DreamerV3ActorLoss,
    DreamerV3ModelLoss,
    DreamerV3ValueLoss,
    categorical_kl_terms,
)
from .tqc import TQCLoss
from .act import ACTLoss, DiffusionActor, DiffusionBCLoss
from .common import AUTO_LOSS_MASK_KEYS

# reference-parity re-exports (torchrl subpackage-level __all__)
from .utils import (  # noqa: F401
    default_value_kwargs,
    hold_out_params,
    next_state_value,
)
from rl_amd.modules.functional import (  # noqa: F401
    symexp,
    symlog,
    two_hot_cross_entropy,
    two_hot_decode,
    two_hot_encode,
)
__all__ = sorted(set(list(globals().get('__all__', [])) + ['default_value_kwargs', 'hold_out_params', 'next_state_value', 'symexp', 'symlog', 'two_hot_cross_entropy', 'two_hot_decode', 'two_hot_encode']))

from .dreamer_v3 import categorical_kl_balanced  # noqa: F401
from .utils import add_random_module, group_optimizers  # noqa: F401
__all__ = sorted(set(__all__) | {"categorical_kl_balanced", "add_random_module", "group_optimizers"})
