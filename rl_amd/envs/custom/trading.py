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


class FinancialRegimeEnv(EnvBase):
    """Batched GBM trading env with holdings state (reference
    custom/trading.py:28): actions are 0=Hold / 1=Buy / 2=Sell; the
    observation is a rolling ``price_history`` window plus
    ``current_holdings`` / ``entry_price`` / ``step_count``; reward is
    realized P&L minus transaction costs.  All state is device-resident
    tensors, so thousands of markets step in a few fused kernels.
    """

    DEFAULT_WINDOW_SIZE = 50
    _supports_masked_reset = True

    def __init__(
        self,
        batch_size=(),
        device=None,
        window_size: int = 50,
        episode_len: int = 200,
        volatility: float = 0.2,
        drift: float = 0.05,
        transaction_cost: float = 0.001,
        dt: float = 1.0 / 252.0,
        seed: Optional[int] = None,
    ):
        super().__init__(device=device, batch_size=batch_size)
        self.window_size = window_size
        self.episode_len = episode_len
        self.volatility = volatility
        self.drift = drift
        self.transaction_cost = transaction_cost
        self.dt = dt
        bs = self.batch_size
        self.observation_spec = Composite(
            {
                "price_history": Unbounded(shape=(*bs, window_size), device=self.device),
                "current_holdings": Unbounded(shape=(*bs, 1), dtype=torch.bool, device=self.device),
                "entry_price": Unbounded(shape=(*bs, 1), device=self.device),
                "step_count": Unbounded(shape=(*bs,), dtype=torch.int64, device=self.device),
            },
            shape=bs,
            device=self.device,
        )
        self.action_spec = Categorical(3, shape=bs, device=self.device)
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)
        self._hist = None

    def _gbm_path(self, *shape):
        z = torch.randn(*shape, device=self.device)
        ret = self.drift * self.dt + z * self.volatility * (self.dt ** 0.5)
        return 100.0 * torch.exp(torch.cumsum(ret, dim=-1))

    def _fresh_state(self):
        bs = self.batch_size
        return (
            self._gbm_path(*bs, self.window_size),
            torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
            torch.zeros((*bs, 1), device=self.device),
            torch.zeros(bs, dtype=torch.int64, device=self.device),
        )

    def _reset(self, tensordict=None, **kwargs) -> TensorDictBase:
        bs = self.batch_size
        hist, hold, entry, t = self._fresh_state()
        if tensordict is not None and "_reset" in tensordict and self._hist is not None:
            mask = tensordict.get("_reset").reshape(*bs, 1)
            self._hist = torch.where(mask.expand_as(hist), hist, self._hist)
            self._hold = torch.where(mask, hold, self._hold)
            self._entry = torch.where(mask, entry, self._entry)
            self._t = torch.where(mask.squeeze(-1), t, self._t)
        else:
            self._hist, self._hold, self._entry, self._t = hist, hold, entry, t
        return TensorDict(
            {
                "price_history": self._hist.clone(),
                "current_holdings": self._hold.clone(),
                "entry_price": self._entry.clone(),
                "step_count": self._t.clone(),
                "done": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
                "terminated": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
            },
            batch_size=bs,
            device=self.device,
        )

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        bs = self.batch_size
        action = tensordict.get("action").reshape(bs).long()
        price = self._hist[..., -1:]
        z = torch.randn((*bs, 1), device=self.device)
        new_price = price * torch.exp(
            self.drift * self.dt + z * self.volatility * (self.dt ** 0.5)
        )
        holding = self._hold
        buy = (action == 1).unsqueeze(-1) & ~holding
        sell = (action == 2).unsqueeze(-1) & holding
        reward = torch.zeros((*bs, 1), device=self.device)
        # mark-to-market gain while holding
        reward = reward + holding.float() * (new_price - price)
        # entry/exit transaction costs
        reward = reward - (buy | sell).float() * self.transaction_cost * price
        self._entry = torch.where(buy, price, self._entry)
        self._hold = (holding | buy) & ~sell
        self._hist = torch.cat([self._hist[..., 1:], new_price], dim=-1)
        self._t = self._t + 1
        done = (self._t >= self.episode_len).reshape(*bs, 1)
        return TensorDict(
            {
                "price_history": self._hist.clone(),
                "current_holdings": self._hold.clone(),
                "entry_price": self._entry.clone(),
                "step_count": self._t.clone(),
                "reward": reward,
                "done": done,
                "terminated": done,
            },
            batch_size=bs,
            device=self.device,
        )

    def _set_seed(self, seed: Optional[int]):
        if seed is not None:
            torch.manual_seed(seed)


__all__.append("FinancialRegimeEnv")
