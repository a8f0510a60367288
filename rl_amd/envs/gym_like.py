"""GymLikeEnv — adapter from (obs, reward, done, info) tuples to TensorDict.

Reference: pytorch/rl torchrl/envs/gym_like.py:153 (GymLikeEnv),
:26-152 (info_dict_reader).
"""
from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional, Tuple

import numpy as np
import torch

from ..data.tensor_specs import Composite, TensorSpec, Unbounded
from ..tensordict import TensorDict, TensorDictBase
from .common import EnvBase

__all__ = ["GymLikeEnv", "default_info_dict_reader"]


class default_info_dict_reader:
    """Pull selected info-dict entries into the TensorDict
    (reference gym_like.py:26)."""

    def __init__(self, keys: Optional[List[str]] = None, spec: Optional[Composite] = None):
        self.keys = keys or []
        self.info_spec = spec

    def __call__(self, info: dict, td: TensorDictBase) -> TensorDictBase:
        for k in self.keys:
            if k in info:
                td.set(k, torch.as_tensor(info[k]))
        return td


class GymLikeEnv(EnvBase):
    """Base for wrappers over step()/reset() tuple APIs.

    Subclasses set ``self._env`` and implement ``_build_env`` spec
    inference; this class handles tuple → TensorDict conversion for both
    4-tuple (obs, r, done, info) and 5-tuple
    (obs, r, terminated, truncated, info) step conventions.
    """

    def __init__(self, *args, device=None, batch_size=None, **kwargs):
        super().__init__(device=device, batch_size=batch_size)
        self._info_dict_reader: List[Callable] = []

    def set_info_dict_reader(self, reader: Callable) -> "GymLikeEnv":
        self._info_dict_reader.append(reader)
        return self

    def read_obs(self, obs) -> dict:
        if isinstance(obs, dict):
            return {k: self._to_tensor(v) for k, v in obs.items()}
        return {"observation": self._to_tensor(obs)}

    def read_reward(self, reward):
        return self._to_tensor(reward).reshape(*self.batch_size, 1).to(torch.float32)

    def read_done(self, terminated, truncated=None):
        term = self._to_tensor(terminated).reshape(*self.batch_size, 1).bool()
        if truncated is None:
            trunc = torch.zeros_like(term)
        else:
            trunc = self._to_tensor(truncated).reshape(*self.batch_size, 1).bool()
        return term, trunc

    def read_action(self, action: torch.Tensor):
        spec = self.action_spec
        return spec.to_numpy(action) if hasattr(spec, "to_numpy") else action.cpu().numpy()

    def _to_tensor(self, x) -> torch.Tensor:
        if isinstance(x, torch.Tensor):
            return x.to(self.device)
        if isinstance(x, np.ndarray):
            return torch.as_tensor(x.copy(), device=self.device)
        return torch.as_tensor(x, device=self.device)

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        action = self.read_action(tensordict.get("action"))
        out = self._env.step(action)
        if len(out) == 5:
            obs, reward, terminated, truncated, info = out
        else:
            obs, reward, done, info = out
            terminated, truncated = done, None
        term, trunc = self.read_done(terminated, truncated)
        td = TensorDict(
            self.read_obs(obs), batch_size=self.batch_size, device=self.device
        )
        td.set("reward", self.read_reward(reward))
        td.set("terminated", term)
        td.set("truncated", trunc)
        td.set("done", term | trunc)
        for reader in self._info_dict_reader:
            td = reader(info or {}, td)
        return td

    def _reset(self, tensordict: Optional[TensorDictBase] = None, **kwargs) -> TensorDictBase:
        out = self._env.reset(**kwargs)
        if isinstance(out, tuple):
            obs, info = out
        else:
            obs, info = out, {}
        td = TensorDict(
            self.read_obs(obs), batch_size=self.batch_size, device=self.device
        )
        shape = (*self.batch_size, 1)
        td.set("done", torch.zeros(shape, dtype=torch.bool, device=self.device))
        td.set("terminated", torch.zeros(shape, dtype=torch.bool, device=self.device))
        for reader in self._info_dict_reader:
            td = reader(info or {}, td)
        return td

    def _set_seed(self, seed):
        if seed is not None and hasattr(self._env, "reset"):
            try:
                self._env.reset(seed=seed)
            except TypeError:
                if hasattr(self._env, "seed"):
                    self._env.seed(seed)
        return seed

    def close(self, raise_if_closed: bool = False):
        if hasattr(self._env, "close"):
            self._env.close()
        self.is_closed = True
