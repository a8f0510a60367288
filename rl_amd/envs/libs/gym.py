"""Gym / Gymnasium wrapper with spec converters.

Reference: pytorch/rl torchrl/envs/libs/gym.py (GymEnv/GymWrapper, spec
converters :383-635).  Import-gated: gymnasium (preferred) or gym must be
installed; the converters map Box/Discrete/MultiDiscrete/MultiBinary/
Dict/Tuple spaces onto rl_amd specs.
"""
from __future__ import annotations

from typing import Any, Optional

import numpy as np
import torch

from ...data.tensor_specs import (
    Binary,
    Bounded,
    Categorical,
    Composite,
    MultiCategorical,
    OneHot,
    TensorSpec,
    Unbounded,
)
from ..gym_like import GymLikeEnv

__all__ = ["GymEnv", "GymWrapper", "gym_backend", "space_to_spec"]

_GYM = None


def gym_backend():
    """Import gymnasium (or legacy gym) on demand."""
    global _GYM
    if _GYM is None:
        try:
            import gymnasium as gym  # type: ignore

            _GYM = gym
        except ImportError:
            try:
                import gym  # type: ignore

                _GYM = gym
            except ImportError as e:
                raise ImportError(
                    "GymEnv requires gymnasium or gym to be installed"
                ) from e
    return _GYM


_NP_TO_TORCH = {
    np.dtype("float32"): torch.float32,
    np.dtype("float64"): torch.float64,
    np.dtype("int64"): torch.int64,
    np.dtype("int32"): torch.int32,
    np.dtype("uint8"): torch.uint8,
    np.dtype("bool"): torch.bool,
}


def space_to_spec(space, device=None, categorical_actions: bool = False) -> TensorSpec:
    """gym.Space → TensorSpec (reference gym.py:383-635)."""
    gym = gym_backend()
    sp = gym.spaces
    if isinstance(space, sp.Box):
        dtype = _NP_TO_TORCH.get(np.dtype(space.dtype), torch.float32)
        return Bounded(
            low=torch.as_tensor(space.low),
            high=torch.as_tensor(space.high),
            shape=space.shape,
            dtype=dtype,
            device=device,
        )
    if isinstance(space, sp.Discrete):
        if categorical_actions:
            return Categorical(int(space.n), shape=(), device=device)
        return OneHot(int(space.n), device=device)
    if isinstance(space, sp.MultiDiscrete):
        return MultiCategorical(space.nvec.tolist(), device=device)
    if isinstance(space, sp.MultiBinary):
        return Binary(int(space.n), device=device)
    if isinstance(space, sp.Dict):
        comp = Composite(device=device)
        for k, v in space.spaces.items():
            comp[k] = space_to_spec(v, device=device, categorical_actions=categorical_actions)
        return comp
    if isinstance(space, sp.Tuple):
        comp = Composite(device=device)
        for i, v in enumerate(space.spaces):
            comp[f"tuple_{i}"] = space_to_spec(v, device=device)
        return comp
    raise NotImplementedError(f"unsupported gym space {type(space)}")


class GymWrapper(GymLikeEnv):
    """Wrap an existing gym env object (reference GymWrapper)."""

    def __init__(self, env, *, device=None, categorical_action_encoding: bool = False, **kwargs):
        super().__init__(device=device, batch_size=None)
        self._env = env
        self.categorical_action_encoding = categorical_action_encoding
        self._make_specs()

    def _make_specs(self):
        obs_spec = space_to_spec(self._env.observation_space, device=self.device)
        if not isinstance(obs_spec, Composite):
            obs_spec = Composite(
                {"observation": obs_spec}, shape=(), device=self.device
            )
        self.observation_spec = obs_spec
        self.action_spec = space_to_spec(
            self._env.action_space,
            device=self.device,
            categorical_actions=self.categorical_action_encoding,
        )
        self.reward_spec = Unbounded(shape=(1,), device=self.device)

    def read_action(self, action: torch.Tensor):
        gym = gym_backend()
        if isinstance(self._env.action_space, gym.spaces.Discrete):
            if action.dtype != torch.int64 or action.dim() > 0 and action.shape[-1] > 1:
                return int(action.argmax(-1).item())
            return int(action.item())
        return action.detach().cpu().numpy()


class GymEnv(GymWrapper):
    """Build a gym env by name (reference GymEnv)."""

    def __init__(self, env_name: str, *, device=None, **kwargs):
        gym = gym_backend()
        env = gym.make(env_name, **{k: v for k, v in kwargs.items() if k not in ("categorical_action_encoding",)})
        self.env_name = env_name
        super().__init__(
            env,
            device=device,
            categorical_action_encoding=kwargs.get("categorical_action_encoding", False),
        )
