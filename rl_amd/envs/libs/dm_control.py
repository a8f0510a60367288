"""dm_control wrapper (import-gated).

Reference: pytorch/rl torchrl/envs/libs/dm_control.py (DMControlEnv /
DMControlWrapper): converts dm_env specs and TimeStep tuples.
"""
from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ...data.tensor_specs import Bounded, Composite, Unbounded
from ...tensordict import TensorDict, TensorDictBase
from ..common import EnvBase

__all__ = ["DMControlEnv", "DMControlWrapper"]


def _spec_from_dm(spec, device):
    import dm_env.specs as dm_specs  # type: ignore

    if isinstance(spec, dm_specs.BoundedArray):
        return Bounded(
            low=torch.as_tensor(np.broadcast_to(spec.minimum, spec.shape).copy()),
            high=torch.as_tensor(np.broadcast_to(spec.maximum, spec.shape).copy()),
            shape=spec.shape,
            dtype=torch.float32,
            device=device,
        )
    return Unbounded(shape=spec.shape, dtype=torch.float32, device=device)


class DMControlWrapper(EnvBase):
    """Wrap an existing dm_control environment object."""

    def __init__(self, env, *, device=None):
        super().__init__(device=device, batch_size=())
        self._env = env
        obs_spec = env.observation_spec()
        comp = Composite(shape=(), device=self.device)
        for k, v in obs_spec.items():
            comp[k] = _spec_from_dm(v, self.device)
        self.observation_spec = comp
        self.action_spec = _spec_from_dm(env.action_spec(), self.device)
        self.reward_spec = Unbounded(shape=(1,), device=self.device)

    def _read_obs(self, ts) -> dict:
        return {
            k: torch.as_tensor(np.asarray(v), device=self.device).float()
            for k, v in ts.observation.items()
        }

    def _reset(self, tensordict=None, **kwargs):
        ts = self._env.reset()
        td = TensorDict(self._read_obs(ts), batch_size=(), device=self.device)
        td.set("done", torch.zeros(1, dtype=torch.bool, device=self.device))
        td.set("terminated", torch.zeros(1, dtype=torch.bool, device=self.device))
        return td

    def _step(self, tensordict):
        action = tensordict.get("action").detach().cpu().numpy()
        ts = self._env.step(action)
        td = TensorDict(self._read_obs(ts), batch_size=(), device=self.device)
        td.set("reward", torch.tensor([ts.reward or 0.0], device=self.device))
        done = torch.tensor([ts.last()], device=self.device)
        td.set("done", done)
        td.set("terminated", done & torch.tensor([ts.discount == 0.0], device=self.device))
        td.set("truncated", done & torch.tensor([ts.discount != 0.0], device=self.device))
        return td

    def _set_seed(self, seed):
        return seed


class DMControlEnv(DMControlWrapper):
    """Build by (domain, task) name."""

    def __init__(self, env_name: str, task_name: str, *, device=None, **kwargs):
        try:
            from dm_control import suite  # type: ignore
        except ImportError as e:
            raise ImportError("DMControlEnv requires dm_control") from e
        env = suite.load(env_name, task_name, **kwargs)
        super().__init__(env, device=device)
