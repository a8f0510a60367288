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
from .value import GAE, TD0Estimator, TD1Estimator, TDLambdaEstimator, ValueEstimatorBase, VTrace
