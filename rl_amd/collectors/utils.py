"""Collector utilities: trajectory splitting/padding.

Reference: pytorch/rl torchrl/collectors/utils.py (split_trajectories).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from ..tensordict import TensorDict, TensorDictBase, pad as td_pad, stack as td_stack

__all__ = ["split_trajectories", "RandomPolicy"]


def split_trajectories(
    rollout: TensorDictBase,
    *,
    prefix=("collector",),
    trajectory_key=None,
    done_key=("next", "done"),
    as_nested: bool = False,
) -> TensorDictBase:
    """Reshape a ``[*B, T]`` rollout into ``[n_traj, max_len]`` with a
    boolean ``mask`` marking valid steps (reference collectors/utils.py).
    """
    traj_key = trajectory_key if trajectory_key is not None else (*prefix, "traj_ids")
    try:
        traj_ids = rollout.get(traj_key)
    except KeyError:
        done = rollout.get(done_key).squeeze(-1)
        # ids from cumulated dones along time
        shifted = torch.cat(
            [torch.zeros_like(done[..., :1]), done[..., :-1]], dim=-1
        )
        traj_ids = shifted.cumsum(-1)
        if rollout.batch_dims > 1:
            # offset each row so ids are globally unique
            max_per_row = traj_ids.max(-1, keepdim=True).values + 1
            offsets = torch.cat(
                [torch.zeros_like(max_per_row[:1]), max_per_row[:-1].cumsum(0)], 0
            )
            traj_ids = traj_ids + offsets
    flat = rollout.reshape(-1) if rollout.batch_dims > 1 else rollout
    flat_ids = traj_ids.reshape(-1)
    uniq = torch.unique(flat_ids)
    pieces: List[TensorDictBase] = []
    max_len = 0
    for u in uniq.tolist():
        m = flat_ids == u
        piece = flat[m]
        max_len = max(max_len, piece.batch_size[0])
        pieces.append(piece)
    out = []
    for piece in pieces:
        L = piece.batch_size[0]
        mask = torch.zeros(max_len, dtype=torch.bool, device=piece.device)
        mask[:L] = True
        if L < max_len:
            piece = td_pad(piece, [0, max_len - L])
        piece.set("mask", mask)
        out.append(piece)
    return td_stack(out, 0)


class RandomPolicy:
    """Sample random actions from an action spec (reference
    collectors/collectors.py RandomPolicy) — the default policy when a
    collector is built without one."""

    def __init__(self, action_spec, action_key="action"):
        self.action_spec = action_spec
        self.action_key = action_key

    def __call__(self, td):
        td.set(self.action_key, self.action_spec.rand())
        return td
