"""Additional third-party wrappers (import-gated): PettingZoo, VMAS,
Brax, EnvPool.

Reference: pytorch/rl torchrl/envs/libs/ (pettingzoo.py, vmas.py,
brax.py, envpool.py).  Each converts the lib's native spec/step
conventions to rl_amd specs/TensorDicts; all raise a clear ImportError
when the dependency is absent (none ship in this image).
"""
from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ...data.tensor_specs import Bounded, Categorical, Composite, OneHot, Unbounded
from ...tensordict import TensorDict, TensorDictBase
from ..common import EnvBase

__all__ = ["PettingZooWrapper", "VmasWrapper", "BraxWrapper", "MultiThreadedEnv"]


class PettingZooWrapper(EnvBase):
    """Parallel-API PettingZoo env → agent-stacked TensorDicts
    (reference pettingzoo.py)."""

    def __init__(self, env, *, device=None):
        try:
            import pettingzoo  # noqa: F401
        except ImportError as e:
            raise ImportError("PettingZooWrapper requires pettingzoo") from e
        super().__init__(device=device, batch_size=())
        self._env = env
        self.agents = list(env.possible_agents)
        n = len(self.agents)
        from .gym import space_to_spec

        obs_specs = [space_to_spec(env.observation_space(a), device=device) for a in self.agents]
        act_specs = [space_to_spec(env.action_space(a), device=device) for a in self.agents]
        comp = Composite(shape=(), device=self.device)
        comp[("agents", "observation")] = obs_specs[0].expand(n, *obs_specs[0].shape)
        self.observation_spec = comp
        self.full_action_spec = Composite(
            {("agents", "action"): act_specs[0].expand(n, *act_specs[0].shape)},
            shape=(),
            device=self.device,
        )
        self.reward_spec = Composite(
            {("agents", "reward"): Unbounded(shape=(n, 1), device=self.device)},
            shape=(),
            device=self.device,
        )

    def _stack_obs(self, obs_dict):
        return torch.stack(
            [torch.as_tensor(np.asarray(obs_dict[a]), device=self.device).float() for a in self.agents]
        )

    def _reset(self, tensordict=None, **kwargs):
        out = self._env.reset(**kwargs)
        obs = out[0] if isinstance(out, tuple) else out
        td = TensorDict({}, batch_size=(), device=self.device)
        td.set(("agents", "observation"), self._stack_obs(obs))
        td.set("done", torch.zeros(1, dtype=torch.bool, device=self.device))
        td.set("terminated", torch.zeros(1, dtype=torch.bool, device=self.device))
        return td

    def _step(self, tensordict):
        actions = tensordict.get(("agents", "action"))
        act_dict = {
            a: actions[i].detach().cpu().numpy() for i, a in enumerate(self.agents)
        }
        obs, rewards, terms, truncs, _infos = self._env.step(act_dict)
        td = TensorDict({}, batch_size=(), device=self.device)
        td.set(("agents", "observation"), self._stack_obs(obs))
        td.set(
            ("agents", "reward"),
            torch.tensor(
                [[rewards[a]] for a in self.agents], device=self.device, dtype=torch.float32
            ),
        )
        done = any(terms.values()) or any(truncs.values())
        td.set("reward", td.get(("agents", "reward")).sum(0))
        td.set("done", torch.tensor([done], device=self.device))
        td.set("terminated", torch.tensor([any(terms.values())], device=self.device))
        return td

    def _set_seed(self, seed):
        return seed


class VmasWrapper(EnvBase):
    """VMAS vectorized multi-agent env (reference vmas.py) — batched on
    the VMAS side, agent dim stacked under ("agents", ...)."""

    def __init__(self, env, *, device=None):
        try:
            import vmas  # noqa: F401
        except ImportError as e:
            raise ImportError("VmasWrapper requires vmas") from e
        super().__init__(device=device, batch_size=(env.num_envs,))
        self._env = env
        self.n_agents = env.n_agents
        obs_dim = env.observation_space[0].shape[-1]
        act_dim = env.action_space[0].shape[-1]
        bs = self.batch_size
        self.observation_spec = Composite(
            {("agents", "observation"): Unbounded(shape=(*bs, self.n_agents, obs_dim), device=self.device)},
            shape=bs,
            device=self.device,
        )
        self.full_action_spec = Composite(
            {("agents", "action"): Bounded(low=-1.0, high=1.0, shape=(*bs, self.n_agents, act_dim), device=self.device)},
            shape=bs,
            device=self.device,
        )
        self.reward_spec = Composite(
            {("agents", "reward"): Unbounded(shape=(*bs, self.n_agents, 1), device=self.device)},
            shape=bs,
            device=self.device,
        )

    def _reset(self, tensordict=None, **kwargs):
        obs = self._env.reset()
        td = TensorDict({}, batch_size=self.batch_size, device=self.device)
        td.set(("agents", "observation"), torch.stack(list(obs), 1))
        bs = self.batch_size
        td.set("done", torch.zeros((*bs, 1), dtype=torch.bool, device=self.device))
        td.set("terminated", torch.zeros((*bs, 1), dtype=torch.bool, device=self.device))
        return td

    def _step(self, tensordict):
        actions = tensordict.get(("agents", "action"))
        act_list = [actions[:, i] for i in range(self.n_agents)]
        obs, rews, dones, _info = self._env.step(act_list)
        td = TensorDict({}, batch_size=self.batch_size, device=self.device)
        td.set(("agents", "observation"), torch.stack(list(obs), 1))
        td.set(("agents", "reward"), torch.stack(list(rews), 1).unsqueeze(-1))
        td.set("reward", td.get(("agents", "reward")).sum(1))
        td.set("done", dones.unsqueeze(-1))
        td.set("terminated", dones.unsqueeze(-1).clone())
        return td

    def _set_seed(self, seed):
        if seed is not None and hasattr(self._env, "seed"):
            self._env.seed(seed)
        return seed


class BraxWrapper(EnvBase):
    """Brax (JAX) env via dlpack zero-copy (reference brax.py) —
    requires jax+brax (absent in this image)."""

    def __init__(self, env, *, device=None, batch_size=None):
        try:
            import brax  # noqa: F401
        except ImportError as e:
            raise ImportError("BraxWrapper requires brax (and jax)") from e
        raise NotImplementedError(
            "jax is not installed in this build; BraxWrapper is API surface only"
        )


class MultiThreadedEnv(EnvBase):
    """EnvPool batched env (reference envpool.py MultiThreadedEnv)."""

    def __init__(self, env_name: str, num_workers: int, *, device=None, **kwargs):
        try:
            import envpool  # type: ignore
        except ImportError as e:
            raise ImportError("MultiThreadedEnv requires envpool") from e
        super().__init__(device=device, batch_size=(num_workers,))
        self._env = envpool.make(env_name, env_type="gymnasium", num_envs=num_workers, **kwargs)
        from .gym import space_to_spec

        bs = self.batch_size
        obs = space_to_spec(self._env.observation_space, device=device)
        self.observation_spec = Composite(
            {"observation": obs.expand(num_workers, *obs.shape)}, shape=bs, device=self.device
        )
        act = space_to_spec(self._env.action_space, device=device, categorical_actions=True)
        self.action_spec = act.expand(num_workers, *act.shape)
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)

    def _reset(self, tensordict=None, **kwargs):
        obs, _ = self._env.reset()
        bs = self.batch_size
        return TensorDict(
            {
                "observation": torch.as_tensor(obs, device=self.device),
                "done": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
                "terminated": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
            },
            batch_size=bs,
            device=self.device,
        )

    def _step(self, tensordict):
        action = tensordict.get("action").detach().cpu().numpy()
        obs, rew, term, trunc, _ = self._env.step(action)
        bs = self.batch_size
        term_t = torch.as_tensor(term, device=self.device).reshape(*bs, 1)
        trunc_t = torch.as_tensor(trunc, device=self.device).reshape(*bs, 1)
        return TensorDict(
            {
                "observation": torch.as_tensor(obs, device=self.device),
                "reward": torch.as_tensor(rew, device=self.device).reshape(*bs, 1).float(),
                "terminated": term_t,
                "truncated": trunc_t,
                "done": term_t | trunc_t,
            },
            batch_size=bs,
            device=self.device,
        )

    def _set_seed(self, seed):
        return seed
