"""Offline-to-online mixed replay.

Reference: pytorch/rl torchrl/data/replay_buffers/offline_to_online.py:276
(mixes a frozen offline dataset with an online buffer at a given ratio —
the AWAC / offline-bootstrapping pattern).
"""
from __future__ import annotations

from typing import Optional

import torch

from ..tensordict import TensorDictBase, cat as td_cat
from .replay_buffers.buffers import ReplayBuffer

__all__ = ["OfflineOnlineReplayBuffer"]


class OfflineOnlineReplayBuffer:
    """Sample a fixed fraction from an offline buffer, the rest online."""

    def __init__(self, offline_buffer: ReplayBuffer, online_buffer: ReplayBuffer, *, offline_fraction: float = 0.5, batch_size: Optional[int] = None):
        self.offline = offline_buffer
        self.online = online_buffer
        self.offline_fraction = offline_fraction
        self._batch_size = batch_size

    def add(self, data):
        return self.online.add(data)

    def extend(self, data):
        return self.online.extend(data)

    def sample(self, batch_size: Optional[int] = None) -> TensorDictBase:
        bs = batch_size or self._batch_size
        n_off = int(bs * self.offline_fraction)
        n_on = bs - n_off
        parts = []
        if n_off and len(self.offline):
            parts.append(self.offline.sample(n_off))
        if n_on and len(self.online):
            parts.append(self.online.sample(n_on))
        if not parts:
            raise RuntimeError("both buffers are empty")
        return parts[0] if len(parts) == 1 else td_cat(parts, 0)

    def __len__(self):
        return len(self.offline) + len(self.online)

    def update_priority(self, index, priority):
        # priorities route to the online buffer (offline is frozen)
        self.online.update_priority(index, priority)
