"""Replay extras: parameter schedulers, consuming/staleness/prompt-group
samplers, compressed storage.

Reference: pytorch/rl torchrl/data/replay_buffers/
(scheduler.py ParameterScheduler; samplers/random.py ConsumingSampler:70,
staleness.py StalenessAwareSampler:23, llm.py PromptGroupSampler:24;
storages/list.py CompressedListStorage:309).
"""
from __future__ import annotations

import zlib
from typing import Any, Callable, List, Optional, Sequence

import torch

from ...tensordict import TensorDict, TensorDictBase, stack as td_stack
from .samplers import RandomSampler, Sampler
from .storages import ListStorage

__all__ = [
    "ParameterScheduler",
    "LinearScheduler",
    "StepScheduler",
    "ConsumingSampler",
    "StalenessAwareSampler",
    "PromptGroupSampler",
    "CompressedListStorage",
    "LambdaScheduler",
    "SchedulerList",
]


class ParameterScheduler:
    """Anneal a sampler/buffer hyper-parameter (β, α, ε) over steps
    (reference scheduler.py)."""

    def __init__(self, obj: Any, param_name: str, schedule: Callable[[int], float]):
        self.obj = obj
        self.param_name = param_name
        self.schedule = schedule
        self._step = 0

    def step(self, n: int = 1) -> float:
        self._step += n
        val = float(self.schedule(self._step))
        setattr(self.obj, self.param_name, val)
        return val

    @property
    def value(self) -> float:
        return getattr(self.obj, self.param_name)

    def state_dict(self):
        return {"step": self._step}

    def load_state_dict(self, sd):
        self._step = sd["step"]


def LinearScheduler(obj, param_name: str, init_val: float, final_val: float, num_steps: int) -> ParameterScheduler:
    def sched(t):
        frac = min(1.0, t / num_steps)
        return init_val + (final_val - init_val) * frac

    s = ParameterScheduler(obj, param_name, sched)
    setattr(obj, param_name, init_val)
    return s


def StepScheduler(obj, param_name: str, gamma: float = 0.9, n_steps: int = 1000) -> ParameterScheduler:
    init = getattr(obj, param_name)

    def sched(t):
        return init * (gamma ** (t // n_steps))

    return ParameterScheduler(obj, param_name, sched)


class ConsumingSampler(Sampler):
    """FIFO sampler: every item is returned exactly once, in write order
    (reference random.py:70) — on-policy consumption from a buffer."""

    def __init__(self):
        self._pending: List[int] = []

    def extend(self, index):
        idx = torch.as_tensor(index).reshape(-1).tolist()
        self._pending.extend(int(i) for i in idx)

    def add(self, index):
        self.extend([index] if isinstance(index, int) else index)

    def sample(self, storage, batch_size: int):
        if len(self._pending) < batch_size:
            raise RuntimeError(
                f"only {len(self._pending)} unconsumed items; wanted {batch_size}"
            )
        take = self._pending[:batch_size]
        self._pending = self._pending[batch_size:]
        return torch.as_tensor(take, dtype=torch.long), {}

    @property
    def ran_out(self):
        return not self._pending

    def _empty(self):
        self._pending = []

    def state_dict(self):
        return {"pending": list(self._pending)}

    def load_state_dict(self, sd):
        self._pending = list(sd["pending"])


class StalenessAwareSampler(Sampler):
    """Uniform sampling with rejection of too-stale items
    (reference staleness.py:23): items carry the policy version at write
    time; samples older than ``max_staleness`` versions are filtered."""

    def __init__(self, max_staleness: int, version_key: str = "policy_version"):
        self.max_staleness = max_staleness
        self.version_key = version_key
        self.current_version = 0
        self._versions: dict = {}

    def set_version(self, v: int) -> None:
        self.current_version = v

    def extend(self, index):
        for i in torch.as_tensor(index).reshape(-1).tolist():
            self._versions[int(i)] = self.current_version

    def add(self, index):
        self.extend([index])

    def sample(self, storage, batch_size: int):
        n = len(storage)
        fresh = [
            i
            for i in range(n)
            if self.current_version - self._versions.get(i, -(10**9))
            <= self.max_staleness
        ]
        if not fresh:
            raise RuntimeError("no sufficiently fresh samples in the buffer")
        pick = torch.randint(0, len(fresh), (batch_size,))
        fresh_t = torch.as_tensor(fresh, dtype=torch.long)
        return fresh_t[pick], {}


class PromptGroupSampler(Sampler):
    """Sample whole GRPO prompt groups (reference llm.py:24): the storage
    holds G consecutive responses per prompt; indices come out
    group-aligned."""

    def __init__(self, group_size: int):
        self.group_size = group_size

    def sample(self, storage, batch_size: int):
        if batch_size % self.group_size != 0:
            raise ValueError("batch_size must be a multiple of group_size")
        n_groups = len(storage) // self.group_size
        pick = torch.randint(0, n_groups, (batch_size // self.group_size,))
        index = (
            pick.unsqueeze(1) * self.group_size
            + torch.arange(self.group_size).unsqueeze(0)
        ).reshape(-1)
        return index, {"group_size": self.group_size}


class CompressedListStorage(ListStorage):
    """zlib-compressed pickled items (reference list.py:309) — trades CPU
    for memory on huge CPU-side buffers."""

    def __init__(self, max_size: int = 10_000, level: int = 3, **kwargs):
        super().__init__(max_size, **kwargs)
        self.level = level

    def set(self, index, data):
        import pickle

        if isinstance(index, int):
            blob = zlib.compress(pickle.dumps(data), self.level)
            if index >= len(self._storage):
                self._storage.extend([None] * (index + 1 - len(self._storage)))
            self._storage[index] = blob
        else:
            if isinstance(index, torch.Tensor):
                index = index.tolist()
            for i, d in zip(index, data):
                self.set(int(i), d)

    def get(self, index):
        import pickle

        if isinstance(index, int):
            return pickle.loads(zlib.decompress(self._storage[index]))
        if isinstance(index, torch.Tensor):
            index = index.tolist()
        out = [pickle.loads(zlib.decompress(self._storage[int(i)])) for i in index]
        if out and isinstance(out[0], TensorDictBase):
            return td_stack(out, 0)
        return out


def LambdaScheduler(obj, param_name: str, fn: Callable[[int], float]) -> ParameterScheduler:
    """Arbitrary-function schedule (reference scheduler.py LambdaScheduler)."""
    return ParameterScheduler(obj, param_name, fn)


class SchedulerList:
    """Step several parameter schedulers together (reference
    scheduler.py SchedulerList)."""

    def __init__(self, schedulers):
        self.schedulers = list(schedulers)

    def step(self, n: int = 1):
        return [s.step(n) for s in self.schedulers]

    def state_dict(self):
        return [s.state_dict() for s in self.schedulers]

    def load_state_dict(self, sds):
        for s, sd in zip(self.schedulers, sds):
            s.load_state_dict(sd)
