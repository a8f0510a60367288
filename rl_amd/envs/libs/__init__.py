from .gym import GymEnv, GymWrapper, gym_backend, space_to_spec
