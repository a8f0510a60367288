"""MultiStep — n-step return folding on collector output.

Reference: pytorch/rl torchrl/data/postprocs/postprocs.py:85 (MultiStep),
:299 (DensifyReward).

Given a ``[B, T]`` batch, rewrites each transition to the n-step
transition: reward becomes the discounted n-step sum, the "next"
observation becomes the observation n steps ahead (clipped at trajectory
ends), and ``steps_to_next_obs`` records the actual lookahead.  All ops
are vectorized over [B, T] (the per-offset shifts are fused gathers —
GPU-friendly, no python loop over batch).
"""
from __future__ import annotations

from typing import Optional

import torch

from ..tensordict import TensorDict, TensorDictBase

__all__ = ["MultiStep", "DensifyReward"]


class MultiStep:
    def __init__(self, gamma: float, n_steps: int, done_key=("next", "done"), reward_key=("next", "reward")):
        if n_steps < 1:
            raise ValueError("n_steps must be >= 1")
        self.gamma = gamma
        self.n_steps = n_steps

    def __call__(self, td: TensorDictBase) -> TensorDictBase:
        if td.batch_dims < 2:
            raise RuntimeError("MultiStep expects a [B, T] batch")
        nxt = td.get("next")
        reward = nxt.get("reward")
        done = nxt.get("done")
        terminated = nxt.get("terminated", done)
        B_dims = td.batch_size[:-1]
        T = td.batch_size[-1]
        n = self.n_steps
        dtype = reward.dtype
        device = reward.device

        not_done = (~done).to(dtype)
        # within-trajectory mask: alive[t, k] = no done in steps t..t+k-1
        acc_reward = reward.clone()
        alive = torch.ones_like(reward)
        steps = torch.ones_like(reward)
        done_any = done.clone()
        for k in range(1, n):
            # shift the next-step quantities left by k along time
            r_k = _shift_left(reward, k)
            d_prev = _shift_left(done, k - 1) if k > 1 else done
            alive = alive * (~_shift_left(done, k - 1)).to(dtype) if k > 1 else alive * not_done
            acc_reward = acc_reward + (self.gamma**k) * alive * r_k
            steps = steps + alive
            done_any = done_any | (_shift_left(done, k) & alive.bool())
        # n-step next obs / done: gather at index t + steps - 1 (clipped)
        idx = (
            torch.arange(T, device=device).reshape(*(1,) * len(B_dims), T, 1)
            + steps.long()
            - 1
        ).clamp_max(T - 1)
        new_next = nxt.clone(False)
        for key in list(nxt.keys(True, True)):
            val = nxt.get(key)
            gather_idx = idx
            while gather_idx.dim() < val.dim():
                gather_idx = gather_idx.unsqueeze(-1)
            gather_idx = gather_idx.expand(*val.shape[: len(B_dims)], T, *val.shape[len(B_dims) + 1 :])
            new_next.set(key, val.gather(len(B_dims), gather_idx))
        new_next.set("reward", acc_reward)
        out = td.clone(False)
        out.set("next", new_next)
        out.set("steps_to_next_obs", steps.long())
        out.set(("next", "original_reward"), reward)
        return out


def _shift_left(x: torch.Tensor, k: int) -> torch.Tensor:
    """x[..., t, :] ← x[..., t+k, :] with zero/False padding at the end
    (along dim -2)."""
    if k == 0:
        return x
    T = x.shape[-2]
    pad = torch.zeros_like(x[..., :k, :])
    if k >= T:
        return pad[..., :T, :]
    return torch.cat([x[..., k:, :], pad], dim=-2)


class DensifyReward:
    """Spread a sparse terminal reward across the trajectory
    (reference postprocs.py:299)."""

    def __init__(self, reward_key=("next", "reward"), done_key=("next", "done")):
        self.reward_key = reward_key
        self.done_key = done_key

    def __call__(self, td: TensorDictBase) -> TensorDictBase:
        reward = td.get(self.reward_key)
        done = td.get(self.done_key)
        # backward-fill the terminal reward over each trajectory segment
        from ..objectives.value.functional import _reverse_scan

        a = (~done).to(reward.dtype)
        filled = _reverse_scan(reward, a)
        td.set(self.reward_key, filled)
        return td
