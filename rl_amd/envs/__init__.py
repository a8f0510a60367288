from .common import EnvBase, EnvMetaData
from .utils import (
    ExplorationType,
    RandomPolicy,
    check_env_specs,
    exploration_type,
    make_composite_from_td,
    set_exploration_type,
    step_mdp,
    terminated_or_truncated,
)
from .batched_envs import BatchedEnvBase, ParallelEnv, SerialEnv
from .custom import AntVec, HalfCheetahVec, HumanoidVec, PendulumEnv, SyntheticMuJoCoEnv
from .gym_like import GymLikeEnv, default_info_dict_reader
from .model_based import DreamerEnv, ModelBasedEnvBase
from . import transforms
from .llm import ChatEnv
from .async_envs import AsyncEnvPool, ProcessorAsyncEnvPool, ThreadingAsyncEnvPool
from .env_creator import EnvCreator, env_creator
from .custom import ChessEnv, TicTacToeEnv, ToyVLAEnv
from . import libs
from .transforms import HumanoidMacroAction
