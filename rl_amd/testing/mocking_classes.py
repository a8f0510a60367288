"""Deterministic mock envs for unit testing (reference:
pytorch/rl torchrl/testing/mocking_classes.py — CountingEnv:1168,
NestedCountingEnv:1492, EnvWithDynamicSpec:2307, EnvThatErrorsAfter10Iters:2486).

Every collector/transform/objective test runs on these — no simulator deps.
"""
from __future__ import annotations

from typing import Optional

import torch

from ..data.tensor_specs import (
    Binary,
    Bounded,
    Categorical,
    Composite,
    OneHot,
    Unbounded,
)
from ..envs.common import EnvBase
from ..tensordict import TensorDict, TensorDictBase

__all__ = [
    "CountingEnv",
    "NestedCountingEnv",
    "ContinuousActionVecMockEnv",
    "DiscreteActionVecMockEnv",
    "MockSerialEnv",
    "EnvThatErrors",
]


class CountingEnv(EnvBase):
    """Obs counts steps since reset; reward always 1; terminates at
    ``max_steps``.  Bit-for-bit predictable — collectors and GAE outputs are
    assertable exactly (reference mocking_classes.py:1168)."""

    _supports_masked_reset = True

    def __init__(
        self,
        max_steps: int = 5,
        start_val: int = 0,
        batch_size=(),
        device=None,
    ):
        super().__init__(device=device, batch_size=batch_size)
        self.max_steps = max_steps
        self.start_val = start_val
        bs = self.batch_size
        self.observation_spec = Composite(
            {
                "observation": Unbounded(
                    shape=(*bs, 1), device=self.device, dtype=torch.float32
                )
            },
            shape=bs,
            device=self.device,
        )
        self.action_spec = Binary(shape=(*bs, 1), device=self.device, dtype=torch.bool)
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)
        self.count = torch.zeros(
            (*bs, 1), dtype=torch.float32, device=self.device
        )

    def _reset(self, tensordict: Optional[TensorDictBase] = None, **kwargs) -> TensorDictBase:
        if tensordict is not None and "_reset" in tensordict:
            mask = tensordict.get("_reset").reshape(self.count.shape)
            self.count = torch.where(
                mask, torch.full_like(self.count, float(self.start_val)), self.count
            )
        else:
            self.count.fill_(float(self.start_val))
        return TensorDict(
            {
                "observation": self.count.clone(),
                "done": torch.zeros_like(self.count, dtype=torch.bool),
                "terminated": torch.zeros_like(self.count, dtype=torch.bool),
            },
            batch_size=self.batch_size,
            device=self.device,
        )

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        action = tensordict.get("action")
        self.count = self.count + action.to(self.count.dtype).reshape(self.count.shape)
        done = self.count >= self.max_steps
        return TensorDict(
            {
                "observation": self.count.clone(),
                "reward": torch.ones_like(self.count),
                "done": done,
                "terminated": done,
            },
            batch_size=self.batch_size,
            device=self.device,
        )

    def _set_seed(self, seed):
        return seed


class NestedCountingEnv(CountingEnv):
    """CountingEnv with observation nested under ("data", "states")
    (reference mocking_classes.py:1492)."""

    def __init__(self, max_steps: int = 5, nest_obs: bool = True, batch_size=(), device=None):
        super().__init__(max_steps=max_steps, batch_size=batch_size, device=device)
        bs = self.batch_size
        self.observation_spec = Composite(
            {
                "data": Composite(
                    {
                        "states": Unbounded(
                            shape=(*bs, 1), device=self.device, dtype=torch.float32
                        )
                    },
                    shape=bs,
                    device=self.device,
                )
            },
            shape=bs,
            device=self.device,
        )

    def _reset(self, tensordict=None, **kwargs):
        td = super()._reset(tensordict, **kwargs)
        obs = td.pop("observation")
        td.set(("data", "states"), obs)
        return td

    def _step(self, tensordict):
        td = super()._step(tensordict)
        obs = td.pop("observation")
        td.set(("data", "states"), obs)
        return td


class ContinuousActionVecMockEnv(EnvBase):
    """Gaussian-dynamics continuous-control mock: obs' = obs + action-norm,
    reward = -|obs|.  Shapes mimic a MuJoCo-style env."""

    def __init__(
        self,
        obs_dim: int = 7,
        action_dim: int = 5,
        max_steps: int = 100,
        batch_size=(),
        device=None,
    ):
        super().__init__(device=device, batch_size=batch_size)
        bs = self.batch_size
        self.obs_dim = obs_dim
        self.action_dim = action_dim
        self.max_steps = max_steps
        self.observation_spec = Composite(
            {"observation": Unbounded(shape=(*bs, obs_dim), device=self.device)},
            shape=bs,
            device=self.device,
        )
        self.action_spec = Bounded(
            low=-1.0, high=1.0, shape=(*bs, action_dim), device=self.device
        )
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)
        self._obs = None
        self._t = None
        self._gen = torch.Generator(device="cpu")

    def _reset(self, tensordict=None, **kwargs):
        bs = self.batch_size
        self._obs = torch.randn(
            (*bs, self.obs_dim), generator=self._gen
        ).to(self.device)
        self._t = torch.zeros((*bs, 1), device=self.device)
        return TensorDict(
            {
                "observation": self._obs.clone(),
                "done": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
                "terminated": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
            },
            batch_size=bs,
            device=self.device,
        )

    def _step(self, tensordict):
        action = tensordict.get("action")
        self._obs = self._obs + action.norm(dim=-1, keepdim=True) * 0.1
        self._t = self._t + 1
        reward = -self._obs.norm(dim=-1, keepdim=True)
        done = self._t >= self.max_steps
        return TensorDict(
            {
                "observation": self._obs.clone(),
                "reward": reward,
                "done": done,
                "terminated": done,
            },
            batch_size=self.batch_size,
            device=self.device,
        )

    def _set_seed(self, seed):
        if seed is not None:
            self._gen.manual_seed(seed)
        return seed


class DiscreteActionVecMockEnv(EnvBase):
    """Discrete-action mock (CartPole-shaped): obs_dim floats, n_actions
    one-hot actions."""

    def __init__(
        self,
        obs_dim: int = 4,
        n_actions: int = 2,
        max_steps: int = 50,
        categorical: bool = False,
        batch_size=(),
        device=None,
    ):
        super().__init__(device=device, batch_size=batch_size)
        bs = self.batch_size
        self.obs_dim = obs_dim
        self.n_actions = n_actions
        self.max_steps = max_steps
        self.categorical = categorical
        self.observation_spec = Composite(
            {"observation": Unbounded(shape=(*bs, obs_dim), device=self.device)},
            shape=bs,
            device=self.device,
        )
        if categorical:
            self.action_spec = Categorical(n_actions, shape=bs, device=self.device)
        else:
            self.action_spec = OneHot(
                n_actions, shape=(*bs, n_actions), device=self.device
            )
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)
        self._obs = None
        self._t = None
        self._gen = torch.Generator(device="cpu")

    def _reset(self, tensordict=None, **kwargs):
        bs = self.batch_size
        self._obs = torch.randn((*bs, self.obs_dim), generator=self._gen).to(self.device) * 0.05
        self._t = torch.zeros((*bs, 1), device=self.device)
        return TensorDict(
            {
                "observation": self._obs.clone(),
                "done": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
                "terminated": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
            },
            batch_size=bs,
            device=self.device,
        )

    def _step(self, tensordict):
        action = tensordict.get("action")
        if not self.categorical:
            act_idx = action.to(torch.float32).argmax(-1, keepdim=True)
        else:
            act_idx = action.reshape(*self.batch_size, 1)
        self._obs = self._obs + (act_idx.float() - 0.5) * 0.01
        self._t = self._t + 1
        reward = torch.ones((*self.batch_size, 1), device=self.device)
        done = (self._t >= self.max_steps) | (self._obs.abs().max(-1, keepdim=True).values > 2.0)
        return TensorDict(
            {
                "observation": self._obs.clone(),
                "reward": reward,
                "done": done,
                "terminated": done,
            },
            batch_size=self.batch_size,
            device=self.device,
        )

    def _set_seed(self, seed):
        if seed is not None:
            self._gen.manual_seed(seed)
        return seed


class MockSerialEnv(CountingEnv):
    """Alias kept for reference-parity test naming."""


class EnvThatErrors(CountingEnv):
    """Raises after N steps — fault-injection fixture
    (reference mocking_classes.py:2486 ``EnvThatErrorsAfter10Iters``)."""

    def __init__(self, error_at: int = 10, **kwargs):
        super().__init__(**kwargs)
        self.error_at = error_at
        self._n = 0

    def _step(self, tensordict):
        self._n += 1
        if self._n >= self.error_at:
            raise RuntimeError("EnvThatErrors: deliberate failure")
        return super()._step(tensordict)


class MultiKeyCountingEnv(EnvBase):
    """Counting env with MULTIPLE observation/action/reward keys, some
    nested under an agent group (reference mocking_classes.py:1992
    MultiKeyCountingEnv): exercises key-selection machinery in
    collectors, step_mdp and losses."""

    _supports_masked_reset = True

    def __init__(self, max_steps: int = 5, batch_size=(), device=None):
        super().__init__(device=device, batch_size=batch_size)
        self.max_steps = max_steps
        bs = self.batch_size
        self.observation_spec = Composite(
            {
                "observation": Unbounded(shape=(*bs, 3), device=self.device),
                "observation_orig": Unbounded(shape=(*bs, 3), device=self.device),
                "nested_1": Composite(
                    {"observation": Unbounded(shape=(*bs, 2), device=self.device)},
                    shape=bs,
                    device=self.device,
                ),
                "nested_2": Composite(
                    {"observation": Unbounded(shape=(*bs, 4), device=self.device)},
                    shape=bs,
                    device=self.device,
                ),
            },
            shape=bs,
            device=self.device,
        )
        self.full_action_spec = Composite(
            {
                "action": Binary(shape=(*bs, 1), device=self.device, dtype=torch.bool),
                "nested_1": Composite(
                    {
                        "action": Categorical(
                            n=3, shape=(*bs,), device=self.device, dtype=torch.int64
                        )
                    },
                    shape=bs,
                    device=self.device,
                ),
                "nested_2": Composite(
                    {
                        "azione": Binary(
                            shape=(*bs, 1), device=self.device, dtype=torch.bool
                        )
                    },
                    shape=bs,
                    device=self.device,
                ),
            },
            shape=bs,
            device=self.device,
        )
        self.full_reward_spec = Composite(
            {
                "reward": Unbounded(shape=(*bs, 1), device=self.device),
                "nested_1": Composite(
                    {"gift": Unbounded(shape=(*bs, 1), device=self.device)},
                    shape=bs,
                    device=self.device,
                ),
            },
            shape=bs,
            device=self.device,
        )
        self.count = torch.zeros((*bs, 1), dtype=torch.float32, device=self.device)

    def _obs_td(self):
        bs = self.batch_size
        c = self.count
        return TensorDict(
            {
                "observation": c.expand(*bs, 3).clone(),
                "observation_orig": c.expand(*bs, 3).clone(),
                ("nested_1", "observation"): c.expand(*bs, 2).clone(),
                ("nested_2", "observation"): c.expand(*bs, 4).clone(),
            },
            batch_size=bs,
            device=self.device,
        )

    def _reset(self, tensordict=None, **kwargs):
        if tensordict is not None and "_reset" in tensordict:
            mask = tensordict.get("_reset").reshape(self.count.shape)
            self.count = torch.where(mask, torch.zeros_like(self.count), self.count)
        else:
            self.count.fill_(0.0)
        td = self._obs_td()
        td.set("done", torch.zeros_like(self.count, dtype=torch.bool))
        td.set("terminated", torch.zeros_like(self.count, dtype=torch.bool))
        return td

    def _step(self, tensordict):
        self.count = self.count + 1.0
        done = self.count >= self.max_steps
        td = self._obs_td()
        td.set("reward", torch.ones_like(self.count))
        td.set(("nested_1", "gift"), torch.full_like(self.count, 2.0))
        td.set("done", done)
        td.set("terminated", done)
        return td

    def _set_seed(self, seed):
        return seed


class HeterogeneousCountingEnv(EnvBase):
    """Multi-agent counting env whose agents have DIFFERENT observation
    widths — per-step data is a LazyStackedTensorDict under "agents"
    and the spec a LazyStackedComposite (reference
    mocking_classes.py:1787 HeterogeneousCountingEnv)."""

    def __init__(self, n_agents: int = 3, max_steps: int = 5, device=None):
        super().__init__(device=device, batch_size=())
        from ..data.tensor_specs import Stacked

        self.n_agents = n_agents
        self.max_steps = max_steps
        agent_specs = [
            Composite(
                {
                    "observation": Unbounded(shape=(i + 1,), device=self.device),
                },
                shape=(),
                device=self.device,
            )
            for i in range(n_agents)
        ]
        self.observation_spec = Composite(
            {},
            shape=(),
            device=self.device,
        )
        self.observation_spec.set("agents", Stacked(*agent_specs, dim=0))
        self.action_spec = Bounded(
            low=-1.0, high=1.0, shape=(n_agents,), device=self.device
        )
        self.reward_spec = Unbounded(shape=(1,), device=self.device)
        self.count = 0.0

    def _agents_td(self):
        from ..tensordict import lazy_stack

        return lazy_stack(
            [
                TensorDict(
                    {
                        "observation": torch.full(
                            (i + 1,), self.count, device=self.device
                        ),
                    },
                    batch_size=(),
                    device=self.device,
                )
                for i in range(self.n_agents)
            ],
            0,
        )

    def _reset(self, tensordict=None, **kwargs):
        self.count = 0.0
        td = TensorDict({}, batch_size=(), device=self.device)
        td.set("agents", self._agents_td())
        td.set("done", torch.zeros(1, dtype=torch.bool, device=self.device))
        td.set("terminated", torch.zeros(1, dtype=torch.bool, device=self.device))
        return td

    def _step(self, tensordict):
        self.count += 1.0
        done = torch.full(
            (1,), self.count >= self.max_steps, dtype=torch.bool, device=self.device
        )
        td = TensorDict({}, batch_size=(), device=self.device)
        td.set("agents", self._agents_td())
        td.set("reward", torch.ones(1, device=self.device))
        td.set("done", done)
        td.set("terminated", done)
        return td

    def _set_seed(self, seed):
        return seed


class EnvWithDynamicSpec(EnvBase):
    """Observation shape GROWS each step (reference
    mocking_classes.py:2307): rollouts cannot densify and must come back
    as LazyStackedTensorDicts (``return_contiguous=False``)."""

    def __init__(self, max_steps: int = 4, device=None):
        super().__init__(device=device, batch_size=())
        self.max_steps = max_steps
        self.observation_spec = Composite(
            {"observation": Unbounded(shape=(1, 2), device=self.device)},
            shape=(),
            device=self.device,
        )
        self.action_spec = Bounded(low=-1.0, high=1.0, shape=(2,), device=self.device)
        self.reward_spec = Unbounded(shape=(1,), device=self.device)
        self._t = 0

    def _obs(self):
        return torch.ones((self._t + 1, 2), device=self.device) * self._t

    def _reset(self, tensordict=None, **kwargs):
        self._t = 0
        return TensorDict(
            {
                "observation": self._obs(),
                "done": torch.zeros(1, dtype=torch.bool, device=self.device),
                "terminated": torch.zeros(1, dtype=torch.bool, device=self.device),
            },
            batch_size=(),
            device=self.device,
        )

    def _step(self, tensordict):
        self._t += 1
        done = torch.full(
            (1,), self._t >= self.max_steps, dtype=torch.bool, device=self.device
        )
        return TensorDict(
            {
                "observation": self._obs(),
                "reward": torch.ones(1, device=self.device),
                "done": done,
                "terminated": done,
            },
            batch_size=(),
            device=self.device,
        )

    def _set_seed(self, seed):
        return seed


__all__ += ["MultiKeyCountingEnv", "HeterogeneousCountingEnv", "EnvWithDynamicSpec"]


class StatelessCountingEnv(EnvBase):
    """Counter carried IN the tensordict (no hidden env state) —
    exercises state-in-data plumbing (reference stateless mocks;
    also the case GraphedRollout's entry snapshot covers trivially)."""

    def __init__(self, max_steps: int = 5, batch_size=(), device=None):
        super().__init__(device=device, batch_size=batch_size)
        self.max_steps = max_steps
        bs = self.batch_size
        self.observation_spec = Composite(
            {
                "count": Unbounded(shape=(*bs, 1), device=self.device),
            },
            shape=bs,
            device=self.device,
        )
        self.action_spec = Binary(shape=(*bs, 1), device=self.device, dtype=torch.bool)
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)

    def _reset(self, tensordict=None, **kwargs):
        bs = self.batch_size
        return TensorDict(
            {
                "count": torch.zeros(*bs, 1, device=self.device),
                "done": torch.zeros(*bs, 1, dtype=torch.bool, device=self.device),
                "terminated": torch.zeros(*bs, 1, dtype=torch.bool, device=self.device),
            },
            batch_size=bs,
            device=self.device,
        )

    def _step(self, tensordict):
        count = tensordict.get("count") + tensordict.get("action").float()
        done = count >= self.max_steps
        return TensorDict(
            {
                "count": count,
                "reward": torch.ones_like(count),
                "done": done,
                "terminated": done,
            },
            batch_size=self.batch_size,
            device=self.device,
        )

    def _set_seed(self, seed):
        return seed


class MockBatchedLockedEnv(CountingEnv):
    """Env whose batch size is FIXED at construction — rejects inputs of
    any other batch shape (reference mocking_classes MockBatchedLockedEnv)."""

    batch_locked = True

    def _step(self, tensordict):
        if tuple(tensordict.batch_size) != tuple(self.batch_size):
            raise RuntimeError(
                f"batch-locked env: expected {tuple(self.batch_size)}, got "
                f"{tuple(tensordict.batch_size)}"
            )
        return super()._step(tensordict)


class MockBatchedUnLockedEnv(CountingEnv):
    """Env accepting arbitrary leading batch shapes (reference
    MockBatchedUnLockedEnv)."""

    batch_locked = False

    def _step(self, tensordict):
        bs = tensordict.batch_size
        action = tensordict.get("action")
        count = tensordict.get("observation", torch.zeros(*bs, 1)) + action.float()
        done = count >= self.max_steps
        return TensorDict(
            {
                "observation": count,
                "reward": torch.ones_like(count),
                "done": done,
                "terminated": done,
            },
            batch_size=bs,
            device=self.device,
        )


class DiscreteActionConvMockEnv(EnvBase):
    """Pixel observation + categorical action (the DQN/Atari shape;
    reference DiscreteActionConvMockEnv)."""

    _supports_masked_reset = True

    def __init__(self, batch_size=(), device=None, pixel_shape=(1, 7, 7), n_actions: int = 7, max_steps: int = 10):
        super().__init__(device=device, batch_size=batch_size)
        self.max_steps = max_steps
        self.pixel_shape = tuple(pixel_shape)
        bs = self.batch_size
        self.observation_spec = Composite(
            {
                "pixels": Bounded(
                    low=0.0, high=1.0, shape=(*bs, *pixel_shape), device=self.device
                )
            },
            shape=bs,
            device=self.device,
        )
        self.action_spec = Categorical(n_actions, shape=bs, device=self.device, dtype=torch.int64)
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)
        self._t = torch.zeros(*bs, 1, device=self.device)

    def _pix(self):
        return torch.rand(*self.batch_size, *self.pixel_shape, device=self.device)

    def _reset(self, tensordict=None, **kwargs):
        self._t = torch.zeros(*self.batch_size, 1, device=self.device)
        bs = self.batch_size
        return TensorDict(
            {
                "pixels": self._pix(),
                "done": torch.zeros(*bs, 1, dtype=torch.bool, device=self.device),
                "terminated": torch.zeros(*bs, 1, dtype=torch.bool, device=self.device),
            },
            batch_size=bs,
            device=self.device,
        )

    def _step(self, tensordict):
        self._t = self._t + 1
        done = self._t >= self.max_steps
        act = tensordict.get("action").reshape(*self.batch_size, 1).float()
        return TensorDict(
            {
                "pixels": self._pix(),
                "reward": act / 10.0 + 1.0,
                "done": done,
                "terminated": done,
            },
            batch_size=self.batch_size,
            device=self.device,
        )

    def _set_seed(self, seed):
        return seed


class ContinuousActionConvMockEnv(DiscreteActionConvMockEnv):
    """Pixel observation + continuous action (reference
    ContinuousActionConvMockEnv)."""

    def __init__(self, batch_size=(), device=None, pixel_shape=(1, 7, 7), act_dim: int = 4, max_steps: int = 10):
        super().__init__(batch_size=batch_size, device=device, pixel_shape=pixel_shape, max_steps=max_steps)
        bs = self.batch_size
        self.action_spec = Bounded(low=-1.0, high=1.0, shape=(*bs, act_dim), device=self.device)

    def _step(self, tensordict):
        self._t = self._t + 1
        done = self._t >= self.max_steps
        act = tensordict.get("action")
        return TensorDict(
            {
                "pixels": self._pix(),
                "reward": act.pow(2).sum(-1, keepdim=True),
                "done": done,
                "terminated": done,
            },
            batch_size=self.batch_size,
            device=self.device,
        )


class MultiAgentCountingEnv(EnvBase):
    """Homogeneous n-agent counting env with ("agents", ...) grouping
    (reference MultiAgentCountingEnv — the dense-MARL complement of
    HeterogeneousCountingEnv)."""

    def __init__(self, n_agents: int = 3, max_steps: int = 5, batch_size=(), device=None):
        super().__init__(device=device, batch_size=batch_size)
        self.n_agents = n_agents
        self.max_steps = max_steps
        bs = self.batch_size
        self.observation_spec = Composite(
            {("agents", "observation"): Unbounded(shape=(*bs, n_agents, 3), device=self.device)},
            shape=bs,
            device=self.device,
        )
        self.full_action_spec = Composite(
            {("agents", "action"): Binary(shape=(*bs, n_agents, 1), device=self.device, dtype=torch.bool)},
            shape=bs,
            device=self.device,
        )
        self.full_reward_spec = Composite(
            {("agents", "reward"): Unbounded(shape=(*bs, n_agents, 1), device=self.device)},
            shape=bs,
            device=self.device,
        )
        self.count = torch.zeros(*bs, n_agents, 1, device=self.device)

    def _reset(self, tensordict=None, **kwargs):
        bs = self.batch_size
        self.count = torch.zeros(*bs, self.n_agents, 1, device=self.device)
        td = TensorDict({}, batch_size=bs, device=self.device)
        td.set(("agents", "observation"), self.count.expand(*bs, self.n_agents, 3).clone())
        td.set("done", torch.zeros(*bs, 1, dtype=torch.bool, device=self.device))
        td.set("terminated", torch.zeros(*bs, 1, dtype=torch.bool, device=self.device))
        return td

    def _step(self, tensordict):
        act = tensordict.get(("agents", "action")).float()
        self.count = self.count + act
        done = (self.count >= self.max_steps).any(-2)
        td = TensorDict({}, batch_size=self.batch_size, device=self.device)
        td.set(("agents", "observation"), self.count.expand(*self.batch_size, self.n_agents, 3).clone())
        td.set(("agents", "reward"), torch.ones_like(self.count))
        td.set("done", done)
        td.set("terminated", done.clone())
        return td

    def _set_seed(self, seed):
        return seed


class EnvWithMetadata(CountingEnv):
    """Counting env carrying a NON-TENSOR leaf through reset/step
    (reference EnvWithMetadata): exercises NonTensorData plumbing."""

    def _reset(self, tensordict=None, **kwargs):
        td = super()._reset(tensordict, **kwargs)
        td.set_non_tensor("info_str", "reset")
        return td

    def _step(self, tensordict):
        td = super()._step(tensordict)
        td.set_non_tensor("info_str", f"step{int(self.count.reshape(-1)[0])}")
        return td


class CountingPolicy:
    """Deterministic mock policy: always emits action=1 for counting
    envs (reference CountingEnvCountPolicy)."""

    def __init__(self, action_spec=None, action_key="action"):
        self.action_spec = action_spec
        self.action_key = action_key

    def __call__(self, td):
        shape = (
            self.action_spec.shape
            if self.action_spec is not None
            else (*td.batch_size, 1)
        )
        dtype = self.action_spec.dtype if self.action_spec is not None else torch.bool
        td.set(self.action_key, torch.ones(shape, dtype=dtype, device=td.device))
        return td


__all__ += [
    "StatelessCountingEnv",
    "MockBatchedLockedEnv",
    "MockBatchedUnLockedEnv",
    "DiscreteActionConvMockEnv",
    "ContinuousActionConvMockEnv",
    "MultiAgentCountingEnv",
    "EnvWithMetadata",
    "CountingPolicy",
]
