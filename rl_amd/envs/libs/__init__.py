from .gym import GymEnv, GymWrapper, gym_backend, space_to_spec
from .dm_control import DMControlEnv, DMControlWrapper
from .misc_wrappers import BraxWrapper, MultiThreadedEnv, PettingZooWrapper, VmasWrapper
from .more_wrappers import (
    EnvPoolEnv,
    GenesisEnv,
    HabitatEnv,
    IsaacGymEnv,
    IsaacLabEnv,
    JumanjiEnv,
    LiberoEnv,
    MeltingpotEnv,
    MjLabEnv,
    MujocoPlaygroundEnv,
    OpenMLEnv,
    OpenSpielEnv,
    ProcgenEnv,
    RoboHiveEnv,
    SMACv2Env,
    SafetyGymnasiumEnv,
    UnityMLAgentsEnv,
)
