"""TradingEnv — synthetic OHLC trading environment.

Reference: pytorch/rl torchrl/envs/custom/ (TradingEnv): the agent holds
a position in {-1, 0, +1}; prices follow a batched geometric random walk
generated on-device; reward = position · log-return − transaction costs.
"""
from __future__ import annotations

from typing import Optional

import torch

from ...data.tensor_specs import Bounded, Categorical, Composite, Unbounded
from ...tensordict import TensorDict, TensorDictBase
from ..common import EnvBase

__all__ = ["TradingEnv"]


class TradingEnv(EnvBase):
    _supports_masked_reset = True

    def __init__(
        self,
        batch_size=(),
        device=None,
        window: int = 16,
        episode_len: int = 256,
        vol: float = 0.01,
        fee: float = 1e-4,
        seed: int = 0,
    ):
        super().__init__(device=device, batch_size=batch_size)
        self.window = window
        self.episode_len = episode_len
        self.vol = vol
        self.fee = fee
        bs = self.batch_size
        self.observation_spec = Composite(
            {
                "returns": Unbounded(shape=(*bs, window), device=self.device),
                "position": Unbounded(shape=(*bs, 1), device=self.device),
            },
            shape=bs,
            device=self.device,
        )
        # actions: 0 = short, 1 = flat, 2 = long
        self.action_spec = Categorical(3, shape=bs, device=self.device)
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)
        self._returns = None
        self._pos = None
        self._t = None

    def _randn(self, *shape):
        return torch.randn(*shape, device=self.device)

    def _reset(self, tensordict=None, **kwargs) -> TensorDictBase:
        bs = self.batch_size
        new_hist = self._randn(*bs, self.window) * self.vol
        new_pos = torch.zeros((*bs, 1), device=self.device)
        new_t = torch.zeros((*bs, 1), device=self.device)
        if tensordict is not None and "_reset" in tensordict and self._returns is not None:
            mask = tensordict.get("_reset").reshape(*bs, 1)
            self._returns = torch.where(mask.expand_as(new_hist), new_hist, self._returns)
            self._pos = torch.where(mask, new_pos, self._pos)
            self._t = torch.where(mask, new_t, self._t)
        else:
            self._returns = new_hist
            self._pos = new_pos
            self._t = new_t
        return TensorDict(
            {
                "returns": self._returns.clone(),
                "position": self._pos.clone(),
                "done": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
                "terminated": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
            },
            batch_size=bs,
            device=self.device,
        )

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        bs = self.batch_size
        action = tensordict.get("action").reshape(*bs, 1)
        new_pos = action.float() - 1.0  # {-1, 0, 1}
        ret = self._randn(*bs, 1) * self.vol
        cost = self.fee * (new_pos - self._pos).abs()
        reward = new_pos * ret - cost
        self._returns = torch.cat([self._returns[..., 1:], ret], -1)
        self._pos = new_pos
        self._t = self._t + 1
        trunc = self._t >= self.episode_len
        return TensorDict(
            {
                "returns": self._returns.clone(),
                "position": self._pos.clone(),
                "reward": reward,
                "done": trunc,
                "terminated": torch.zeros_like(trunc),
                "truncated": trunc,
            },
            batch_size=bs,
            device=self.device,
        )

    def _set_seed(self, seed):
        return seed
