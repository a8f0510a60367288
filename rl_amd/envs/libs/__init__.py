from .gym import GymEnv, GymWrapper, gym_backend, space_to_spec
from .dm_control import DMControlEnv, DMControlWrapper
from .misc_wrappers import BraxWrapper, MultiThreadedEnv, PettingZooWrapper, VmasWrapper
