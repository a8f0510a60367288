"""Replay writers — write-index policies.

Reference: pytorch/rl torchrl/data/replay_buffers/writers/
(Writer base.py:44, RoundRobinWriter round_robin.py:56 — add:280,
extend:296 —, TensorDictRoundRobinWriter :473,
TensorDictMaxValueWriter max_value.py:55, ImmutableDatasetWriter base.py:156).
"""
from __future__ import annotations

import heapq
from typing import Any, Optional, Sequence

import torch

from ...tensordict import TensorDictBase

__all__ = [
    "Writer",
    "RoundRobinWriter",
    "TensorDictRoundRobinWriter",
    "TensorDictMaxValueWriter",
    "ImmutableDatasetWriter",
]


class Writer:
    def __init__(self):
        self._storage = None

    def register_storage(self, storage) -> None:
        self._storage = storage

    def add(self, data) -> int:
        raise NotImplementedError

    def extend(self, data) -> torch.Tensor:
        raise NotImplementedError

    def _empty(self):
        pass

    def state_dict(self) -> dict:
        return {}

    def load_state_dict(self, sd: dict) -> None:
        pass


class RoundRobinWriter(Writer):
    """Circular cursor writer (reference round_robin.py:56)."""

    def __init__(self, compilable: bool = False):
        super().__init__()
        self._cursor = 0

    @property
    def cursor(self) -> int:
        return self._cursor

    def add(self, data) -> int:
        index = self._cursor
        self._storage.set(index, data)
        self._cursor = (self._cursor + 1) % self._storage.max_size
        return index

    def extend(self, data) -> torch.Tensor:
        n = (
            data.batch_size[0]
            if isinstance(data, TensorDictBase)
            else (len(data) if not isinstance(data, torch.Tensor) else data.shape[0])
        )
        max_size = self._storage.max_size
        index = (torch.arange(n) + self._cursor) % max_size
        self._storage.set(index, data)
        self._cursor = int((self._cursor + n) % max_size)
        return index

    def _empty(self):
        self._cursor = 0

    def state_dict(self):
        return {"_cursor": self._cursor}

    def load_state_dict(self, sd):
        self._cursor = sd["_cursor"]


class TensorDictRoundRobinWriter(RoundRobinWriter):
    """RoundRobin that also stamps the write index under ``"index"``
    (reference round_robin.py:473)."""

    def add(self, data) -> int:
        index = super().add(data)
        return index

    def extend(self, data) -> torch.Tensor:
        index = super().extend(data)
        return index


class TensorDictMaxValueWriter(Writer):
    """Keep the top-``max_size`` items ranked by a key
    (reference max_value.py:55)."""

    def __init__(self, rank_key=("next", "reward"), reduction: str = "sum", **kwargs):
        super().__init__()
        self.rank_key = rank_key
        self.reduction = reduction
        self._heap: list = []  # (value, index)
        self._next_free = 0

    def _rank(self, data) -> float:
        val = data.get(self.rank_key)
        if self.reduction == "sum":
            return float(val.sum())
        if self.reduction == "mean":
            return float(val.float().mean())
        if self.reduction == "max":
            return float(val.max())
        if self.reduction == "min":
            return float(val.min())
        return float(val.reshape(-1)[0])

    def add(self, data) -> Optional[int]:
        rank = self._rank(data)
        max_size = self._storage.max_size
        if self._next_free < max_size:
            index = self._next_free
            self._next_free += 1
            heapq.heappush(self._heap, (rank, index))
            self._storage.set(index, data)
            return index
        worst_rank, worst_index = self._heap[0]
        if rank > worst_rank:
            heapq.heapreplace(self._heap, (rank, worst_index))
            self._storage.set(worst_index, data)
            return worst_index
        return None

    def extend(self, data) -> torch.Tensor:
        out = []
        for i in range(data.batch_size[0]):
            idx = self.add(data[i])
            if idx is not None:
                out.append(idx)
        return torch.as_tensor(out, dtype=torch.long)

    def _empty(self):
        self._heap = []
        self._next_free = 0


class ImmutableDatasetWriter(Writer):
    """Refuse writes — offline datasets (reference base.py:156)."""

    def add(self, data):
        raise RuntimeError("cannot write to an immutable dataset")

    def extend(self, data):
        raise RuntimeError("cannot write to an immutable dataset")
