"""ReplayBuffer composition root + TensorDict variants.

Reference: pytorch/rl torchrl/data/replay_buffers/replay_buffers/
(ReplayBuffer base.py:158 — sample:2029, prefetch:2067-2080 —,
TensorDictReplayBuffer tensordict.py:57,
TensorDictPrioritizedReplayBuffer prioritized_tensordict.py:52,
PrioritizedReplayBuffer prioritized.py:55).

Composition: storage + sampler + writer + transforms.  On MI355X the
canonical configuration is HBM-resident ``LazyTensorStorage(device="cuda")``
with on-device samplers — `sample()` is a single gather kernel.
"""
from __future__ import annotations

import threading
from concurrent.futures import ThreadPoolExecutor
from typing import Any, Callable, List, Optional, Sequence, Union

import torch

from ...tensordict import TensorDict, TensorDictBase
from .samplers import PrioritizedSampler, RandomSampler, Sampler
from .storages import LazyTensorStorage, ListStorage, Storage
from .writers import RoundRobinWriter, TensorDictRoundRobinWriter, Writer

__all__ = [
    "ReplayBuffer",
    "PrioritizedReplayBuffer",
    "TensorDictReplayBuffer",
    "TensorDictPrioritizedReplayBuffer",
    "ReplayBufferEnsemble",
    "RemoteTensorDictReplayBuffer",
]


class ReplayBuffer:
    """Generic composable replay buffer (reference base.py:158)."""

    def __init__(
        self,
        *,
        storage: Optional[Storage] = None,
        sampler: Optional[Sampler] = None,
        writer: Optional[Writer] = None,
        collate_fn: Optional[Callable] = None,
        pin_memory: bool = False,
        prefetch: Optional[int] = None,
        transform: Optional[Callable] = None,
        batch_size: Optional[int] = None,
        dim_extend: Optional[int] = None,
        generator=None,
        shared: bool = False,
    ):
        self._storage = storage if storage is not None else ListStorage(max_size=1000)
        self._sampler = sampler if sampler is not None else RandomSampler()
        self._writer = writer if writer is not None else RoundRobinWriter()
        self._writer.register_storage(self._storage)
        self._storage.attach(self)
        self._collate_fn = collate_fn
        self._pin_memory = pin_memory
        self._prefetch_cap = prefetch or 0
        self._prefetch_executor = (
            ThreadPoolExecutor(max_workers=1) if self._prefetch_cap else None
        )
        self._prefetch_queue: List = []
        self._transforms: List[Callable] = []
        if transform is not None:
            self.append_transform(transform)
        self._batch_size = batch_size
        self._write_lock = threading.Lock()

    # -- properties -------------------------------------------------------- #
    @property
    def storage(self):
        return self._storage

    @property
    def sampler(self):
        return self._sampler

    @property
    def writer(self):
        return self._writer

    @property
    def batch_size(self):
        return self._batch_size

    def __len__(self):
        return len(self._storage)

    def __repr__(self):
        return (
            f"{type(self).__name__}(storage={type(self._storage).__name__}, "
            f"sampler={type(self._sampler).__name__}, "
            f"writer={type(self._writer).__name__}, size={len(self)})"
        )

    # -- transforms --------------------------------------------------------- #
    def append_transform(self, transform: Callable, *, invert: bool = False) -> "ReplayBuffer":
        self._transforms.append(transform)
        return self

    def insert_transform(self, index: int, transform: Callable) -> "ReplayBuffer":
        self._transforms.insert(index, transform)
        return self

    def _apply_transforms(self, data):
        for t in self._transforms:
            if hasattr(t, "forward"):
                data = t.forward(data)
            else:
                data = t(data)
        return data

    def _inv_transforms(self, data):
        for t in reversed(self._transforms):
            if hasattr(t, "inv"):
                data = t.inv(data)
        return data

    # -- write -------------------------------------------------------------- #
    def add(self, data) -> int:
        data = self._inv_transforms(data)
        with self._write_lock:
            index = self._writer.add(data)
            self._sampler.add(index)
        return index

    def extend(self, data) -> torch.Tensor:
        data = self._inv_transforms(data)
        with self._write_lock:
            index = self._writer.extend(data)
            self._sampler.extend(index)
        return index

    # -- read --------------------------------------------------------------- #
    def _sample(self, batch_size: int):
        index, info = self._sampler.sample(self._storage, batch_size)
        data = self._storage.get(index)
        if self._collate_fn is not None:
            data = self._collate_fn(data)
        if isinstance(data, TensorDictBase):
            if "_weight" in info and info["_weight"] is not None:
                w = info["_weight"]
                if w.numel() == data.batch_size[0] if data.batch_size else False:
                    pass
                data.set("_weight", w.reshape(data.batch_size[:1]) if data.batch_size else w)
            data.set("index", index.to(data.device) if data.device else index)
            trunc = info.get("truncated", None)
            if trunc is not None:
                # mark slice boundaries truncated (SliceSampler contract)
                tkey = info.get("truncated_key", ("next", "truncated"))
                cur = data.get(tkey, None)
                trunc = trunc.to(data.device) if data.device else trunc
                if cur is not None:
                    data.set(tkey, cur | trunc.reshape(cur.shape))
                else:
                    data.set(tkey, trunc.reshape(*data.batch_size, 1))
                dkey = ("next", "done") if not isinstance(tkey, str) else "done"
                done = data.get(dkey, None)
                if done is not None:
                    data.set(dkey, done | trunc.reshape(done.shape))
        if self._pin_memory and isinstance(data, TensorDictBase):
            data = data.pin_memory()
        data = self._apply_transforms(data)
        return data, info

    def sample(self, batch_size: Optional[int] = None, return_info: bool = False):
        if batch_size is None:
            if self._batch_size is None:
                raise ValueError("batch_size must be given here or at construction")
            batch_size = self._batch_size
        if self._prefetch_executor is not None:
            while len(self._prefetch_queue) < self._prefetch_cap:
                self._prefetch_queue.append(
                    self._prefetch_executor.submit(self._sample, batch_size)
                )
            data, info = self._prefetch_queue.pop(0).result()
            self._prefetch_queue.append(
                self._prefetch_executor.submit(self._sample, batch_size)
            )
        else:
            data, info = self._sample(batch_size)
        if return_info:
            return data, info
        return data

    def __iter__(self):
        while True:
            yield self.sample()
            if self._sampler.ran_out:
                break

    def update_priority(self, index, priority) -> None:
        self._sampler.update_priority(index, priority, storage=self._storage)

    def mark_update(self, index) -> None:
        self._sampler.mark_update(index, storage=self._storage)

    def empty(self) -> None:
        self._writer._empty()
        self._sampler._empty()
        self._storage._len = 0 if hasattr(self._storage, "_len") else None

    # -- checkpoint ---------------------------------------------------------- #
    def state_dict(self) -> dict:
        return {
            "storage": self._storage.state_dict(),
            "sampler": self._sampler.state_dict(),
            "writer": self._writer.state_dict(),
        }

    def load_state_dict(self, sd: dict) -> None:
        self._storage.load_state_dict(sd["storage"])
        self._sampler.load_state_dict(sd["sampler"])
        self._writer.load_state_dict(sd["writer"])

    def dumps(self, path) -> None:
        import os

        os.makedirs(path, exist_ok=True)
        self._storage.dumps(path)
        torch.save(
            {"sampler": self._sampler.state_dict(), "writer": self._writer.state_dict()},
            os.path.join(path, "rb_meta.pt"),
        )

    def loads(self, path) -> None:
        import os

        self._storage.loads(path)
        meta = torch.load(os.path.join(path, "rb_meta.pt"), weights_only=False)
        self._sampler.load_state_dict(meta["sampler"])
        self._writer.load_state_dict(meta["writer"])


class PrioritizedReplayBuffer(ReplayBuffer):
    """Non-TensorDict PER (reference prioritized.py:55)."""

    def __init__(
        self,
        *,
        alpha: float = 0.7,
        beta: float = 0.5,
        eps: float = 1e-8,
        storage: Optional[Storage] = None,
        collate_fn=None,
        pin_memory: bool = False,
        prefetch: Optional[int] = None,
        transform=None,
        batch_size: Optional[int] = None,
    ):
        storage = storage if storage is not None else ListStorage(max_size=1000)
        sampler = PrioritizedSampler(storage.max_size, alpha=alpha, beta=beta, eps=eps)
        super().__init__(
            storage=storage,
            sampler=sampler,
            collate_fn=collate_fn,
            pin_memory=pin_memory,
            prefetch=prefetch,
            transform=transform,
            batch_size=batch_size,
        )


class TensorDictReplayBuffer(ReplayBuffer):
    """TD-aware buffer: stamps ``index``, plumbs priorities through the
    ``priority_key`` (reference tensordict.py:57)."""

    def __init__(self, *, priority_key: str = "td_error", **kwargs):
        kwargs.setdefault("writer", TensorDictRoundRobinWriter())
        super().__init__(**kwargs)
        self.priority_key = priority_key

    def add(self, data: TensorDictBase) -> int:
        index = super().add(data)
        if isinstance(data, TensorDictBase):
            prio = data.get(self.priority_key, None)
            if prio is not None:
                self.update_priority(index, prio.reshape(-1))
        return index

    def extend(self, data: TensorDictBase) -> torch.Tensor:
        index = super().extend(data)
        if isinstance(data, TensorDictBase):
            prio = data.get(self.priority_key, None)
            if prio is not None:
                ntotal = index.numel()
                self.update_priority(index, prio.reshape(ntotal, -1).mean(-1))
        return index

    def update_tensordict_priority(self, data: TensorDictBase) -> None:
        index = data.get("index")
        prio = data.get(self.priority_key, None)
        if prio is None:
            return
        n = index.numel()
        self.update_priority(index.reshape(-1), prio.reshape(n, -1).max(-1).values)

    def sample(self, batch_size: Optional[int] = None, return_info: bool = False, include_info: bool = True):
        return super().sample(batch_size, return_info)


class TensorDictPrioritizedReplayBuffer(TensorDictReplayBuffer):
    """PER with TensorDict IO (reference prioritized_tensordict.py:52)."""

    def __init__(
        self,
        *,
        alpha: float = 0.7,
        beta: float = 0.5,
        eps: float = 1e-8,
        storage: Optional[Storage] = None,
        priority_key: str = "td_error",
        reduction: str = "max",
        **kwargs,
    ):
        storage = storage if storage is not None else ListStorage(max_size=1000)
        sampler = PrioritizedSampler(
            storage.max_size, alpha=alpha, beta=beta, eps=eps, reduction=reduction
        )
        super().__init__(
            storage=storage, sampler=sampler, priority_key=priority_key, **kwargs
        )


class ReplayBufferEnsemble:
    """Sample proportionally across several buffers (reference ensemble.py:48)."""

    def __init__(self, *buffers: ReplayBuffer, p: Optional[Sequence[float]] = None, sample_from_all: bool = False):
        self._buffers = list(buffers)
        self.p = list(p) if p is not None else None
        self.sample_from_all = sample_from_all

    def __getitem__(self, i):
        return self._buffers[i]

    def __len__(self):
        return len(self._buffers)

    def sample(self, batch_size: int, return_info: bool = False):
        from ...tensordict import stack as td_stack

        if self.sample_from_all:
            per = batch_size // len(self._buffers)
            out = [rb.sample(per) for rb in self._buffers]
            data = td_stack(out, 0)
            return (data, {}) if return_info else data
        if self.p is not None:
            buf_idx = int(torch.multinomial(torch.tensor(self.p, dtype=torch.float), 1))
        else:
            buf_idx = int(torch.randint(0, len(self._buffers), (1,)))
        data = self._buffers[buf_idx].sample(batch_size)
        return (data, {"buffer": buf_idx}) if return_info else data


class RemoteTensorDictReplayBuffer(TensorDictReplayBuffer):
    """Placeholder parity class: in the reference this wraps an RPC remote
    buffer (remote.py:36); in rl_amd cross-process buffers go through the
    shared-memory / RCCL paths in rl_amd.parallel."""
