"""Hindsight experience replay.

Reference: pytorch/rl torchrl/data/replay_buffers/her.py:49
(HERReplayBuffer), :32 (HindsightStrategy): relabel sampled trajectories
with achieved goals ("future" strategy) and recompute rewards.
"""
from __future__ import annotations

from typing import Callable, Optional

import torch

from ..tensordict import TensorDict, TensorDictBase
from .replay_buffers.buffers import TensorDictReplayBuffer

__all__ = ["HERReplayBuffer", "HindsightStrategy"]


class HindsightStrategy:
    """Pick substitute goals from the same trajectory (strategy "future":
    a random achieved goal later in the trajectory; "final": the last)."""

    def __init__(self, strategy: str = "future", p: float = 0.8):
        if strategy not in ("future", "final", "episode"):
            raise ValueError(f"unknown strategy {strategy}")
        self.strategy = strategy
        self.p = p

    def pick_goal_index(self, t: torch.Tensor, T: int) -> torch.Tensor:
        """t: [N] current step indices within a [*, T] slice."""
        if self.strategy == "final":
            return torch.full_like(t, T - 1)
        if self.strategy == "episode":
            return torch.randint(0, T, t.shape, device=t.device)
        # future: uniform in [t, T)
        span = (T - t).clamp_min(1)
        return t + (torch.rand_like(t.float()) * span.float()).long()


class HERReplayBuffer(TensorDictReplayBuffer):
    """Replay buffer that relabels goals on sample
    (reference her.py:49).

    Expects trajectory slices of shape [num_slices, slice_len] from a
    SliceSampler (or stores whole trajectories): on sample, with
    probability ``p`` the ``desired_goal`` is replaced by an
    ``achieved_goal`` from later in the same slice and the reward is
    recomputed by ``reward_fn(achieved, desired)``.
    """

    def __init__(
        self,
        *,
        reward_fn: Optional[Callable] = None,
        achieved_goal_key: str = "achieved_goal",
        desired_goal_key: str = "desired_goal",
        strategy: str = "future",
        p: float = 0.8,
        **kwargs,
    ):
        super().__init__(**kwargs)
        self.reward_fn = reward_fn or self._default_reward
        self.achieved_goal_key = achieved_goal_key
        self.desired_goal_key = desired_goal_key
        self.hindsight = HindsightStrategy(strategy, p)

    @staticmethod
    def _default_reward(achieved: torch.Tensor, desired: torch.Tensor) -> torch.Tensor:
        """Sparse: 0 when within tolerance, −1 otherwise."""
        d = (achieved - desired).norm(dim=-1, keepdim=True)
        return -(d > 0.05).float()

    def sample(self, batch_size: Optional[int] = None, return_info: bool = False, **kwargs):
        out = super().sample(batch_size, return_info=True)
        data, info = out
        slice_len = info.get("slice_len")
        if slice_len is None:
            # no trajectory structure available: return as-is
            return (data, info) if return_info else data
        num_slices = info["num_slices"]
        shaped = data.reshape(num_slices, slice_len)
        achieved = shaped.get(("next", self.achieved_goal_key))
        # per-slice: pick a future index per timestep
        t_idx = (
            torch.arange(slice_len, device=achieved.device)
            .unsqueeze(0)
            .expand(num_slices, slice_len)
            .reshape(-1)
        )
        goal_idx = self.hindsight.pick_goal_index(t_idx, slice_len).reshape(
            num_slices, slice_len
        )
        relabel = (
            torch.rand(num_slices, 1, device=achieved.device) < self.hindsight.p
        )
        new_goals = achieved.gather(
            1,
            goal_idx.unsqueeze(-1).expand(*goal_idx.shape, achieved.shape[-1]),
        )
        old_goals = shaped.get(self.desired_goal_key)
        mix = torch.where(relabel.unsqueeze(-1), new_goals, old_goals)
        shaped.set(self.desired_goal_key, mix)
        nxt = shaped.get("next")
        nxt.set(self.desired_goal_key, mix)
        nxt.set("reward", self.reward_fn(achieved, mix))
        flat = shaped.reshape(-1)
        return (flat, info) if return_info else flat
