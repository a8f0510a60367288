"""Replay storages: ListStorage, TensorStorage, LazyTensorStorage,
LazyMemmapStorage.

Reference: pytorch/rl torchrl/data/replay_buffers/storages/
(Storage base.py:23, ListStorage list.py:31, TensorStorage tensor.py:156,
LazyTensorStorage :913, LazyMemmapStorage :1165).

MI355X-first: ``LazyTensorStorage(device="cuda")`` is the primary path —
1M-transition buffers live in the 288 GB of HBM3E and sampling is an
on-device gather with zero host traffic.
"""
from __future__ import annotations

import os
from typing import Any, Optional, Sequence, Union

import torch

from ...tensordict import TensorDict, TensorDictBase, stack as td_stack

__all__ = [
    "Storage",
    "ListStorage",
    "LazyStackStorage",
    "TensorStorage",
    "LazyTensorStorage",
    "LazyMemmapStorage",
    "StorageEnsemble",
]


class Storage:
    """Index-addressed container ABC (reference storages/base.py:23)."""

    def __init__(self, max_size: int, ndim: int = 1):
        self.max_size = int(max_size)
        self.ndim = ndim
        self._attached_entities: list = []

    def attach(self, buffer) -> None:
        self._attached_entities.append(buffer)

    def set(self, index, data):
        raise NotImplementedError

    def get(self, index):
        raise NotImplementedError

    def __len__(self):
        raise NotImplementedError

    def __getitem__(self, index):
        return self.get(index)

    def __setitem__(self, index, value):
        return self.set(index, value)

    def dumps(self, path):
        raise NotImplementedError

    def loads(self, path):
        raise NotImplementedError

    def state_dict(self):
        return {}

    def load_state_dict(self, sd):
        pass


class ListStorage(Storage):
    """Python-list storage for arbitrary objects (reference list.py:31)."""

    def __init__(self, max_size: int = 10_000, **kwargs):
        super().__init__(max_size)
        self._storage: list = []

    def set(self, index, data):
        if isinstance(index, int):
            if index >= len(self._storage):
                self._storage.extend([None] * (index + 1 - len(self._storage)))
            self._storage[index] = data
        else:
            if isinstance(index, torch.Tensor):
                index = index.tolist()
            for i, d in zip(index, data):
                self.set(int(i), d)

    def get(self, index):
        if isinstance(index, int):
            return self._storage[index]
        if isinstance(index, slice):
            return self._storage[index]
        if isinstance(index, torch.Tensor):
            index = index.tolist()
        out = [self._storage[int(i)] for i in index]
        if out and isinstance(out[0], TensorDictBase):
            return td_stack(out, 0)
        return out

    def __len__(self):
        return len(self._storage)

    def state_dict(self):
        return {"_storage": self._storage}

    def load_state_dict(self, sd):
        self._storage = list(sd["_storage"])

    def dumps(self, path):
        torch.save(self._storage, os.path.join(path, "list_storage.pt"))

    def loads(self, path):
        self._storage = torch.load(
            os.path.join(path, "list_storage.pt"), weights_only=False
        )


class LazyStackStorage(ListStorage):
    """ListStorage that stacks TensorDicts on read (reference list.py:236)."""


class TensorStorage(Storage):
    """Contiguous pre-allocated tensor/TensorDict storage
    (reference tensor.py:156).  ``get``/``set`` are pure gather/scatter —
    on HIP devices they run as single kernels over HBM."""

    def __init__(
        self,
        storage: Union[torch.Tensor, TensorDictBase],
        max_size: Optional[int] = None,
        device=None,
        ndim: int = 1,
    ):
        if max_size is None:
            max_size = storage.shape[0]
        super().__init__(max_size, ndim=ndim)
        self._storage = storage
        self.device = torch.device(device) if device is not None else None
        self._len = 0
        self.initialized = storage is not None

    @property
    def _len_along_dim0(self):
        return self._len

    def _init(self, data):
        raise RuntimeError("TensorStorage requires a pre-allocated storage")

    def set(self, index, data):
        if not self.initialized:
            if isinstance(index, int):
                self._init(data)
            else:
                self._init(data[0] if data.batch_size else data)
        if isinstance(index, int):
            self._storage[index] = data
            self._len = max(self._len, index + 1)
        else:
            if isinstance(index, slice):
                start, stop, step = index.indices(self.max_size)
                top = stop
            else:
                if not isinstance(index, torch.Tensor):
                    index = torch.as_tensor(index)
                top = int(index.max().item()) + 1 if index.numel() else 0
            self._storage[index] = data
            self._len = max(self._len, top)

    def get(self, index):
        return self._storage[index]

    def __len__(self):
        return self._len

    def state_dict(self):
        return {"_storage": self._storage, "_len": self._len}

    def load_state_dict(self, sd):
        storage = sd["_storage"]
        if self.initialized and isinstance(self._storage, TensorDictBase):
            self._storage.update_(storage)
        else:
            self._storage = storage
            self.initialized = True
        self._len = sd["_len"]

    def dumps(self, path):
        os.makedirs(path, exist_ok=True)
        if isinstance(self._storage, TensorDictBase):
            self._storage.clone().memmap_(os.path.join(path, "storage"))
        else:
            torch.save(self._storage, os.path.join(path, "storage.pt"))
        torch.save({"_len": self._len}, os.path.join(path, "meta.pt"))

    def loads(self, path):
        meta = torch.load(os.path.join(path, "meta.pt"), weights_only=False)
        self._len = meta["_len"]
        td_path = os.path.join(path, "storage")
        if os.path.isdir(td_path):
            loaded = TensorDict.load_memmap(td_path)
            if self.initialized:
                dev = self._storage.device
                self._storage.update_(loaded.to(dev) if dev else loaded)
            else:
                self._storage = loaded
                self.initialized = True
        else:
            self._storage = torch.load(os.path.join(path, "storage.pt"), weights_only=False)
            self.initialized = True


class LazyTensorStorage(TensorStorage):
    """TensorStorage allocated on first write (reference tensor.py:913).
    ``device="cuda"`` keeps the whole buffer HBM-resident."""

    def __init__(self, max_size: int, device=None, ndim: int = 1, compilable: bool = False):
        Storage.__init__(self, max_size, ndim=ndim)
        self._storage = None
        self.device = torch.device(device) if device is not None else None
        self._len = 0
        self.initialized = False

    def _init(self, data):
        """Allocate [max_size, *data.shape] zeros matching data's structure."""
        def make(t):
            return torch.zeros(
                (self.max_size, *t.shape),
                dtype=t.dtype,
                device=self.device if self.device is not None else t.device,
            )

        if isinstance(data, TensorDictBase):
            out = TensorDict(
                {},
                batch_size=(self.max_size, *data.batch_size),
                device=self.device,
            )
            for k, v in data.items(True, True):
                if isinstance(v, torch.Tensor):
                    out.set(k, make(v))
            self._storage = out
        else:
            self._storage = make(data)
        self.initialized = True

    def set(self, index, data):
        if not self.initialized:
            example = data
            if not isinstance(index, int):
                example = data[0]
            self._init(example)
        super().set(index, data)


class LazyMemmapStorage(LazyTensorStorage):
    """Disk-backed memmap storage (reference tensor.py:1165)."""

    def __init__(self, max_size: int, scratch_dir: Optional[str] = None, device=None, ndim: int = 1):
        super().__init__(max_size, device=device, ndim=ndim)
        self.scratch_dir = scratch_dir

    def _init(self, data):
        import tempfile

        prefix = self.scratch_dir or tempfile.mkdtemp(prefix="rl_amd_storage_")

        def make(t):
            return torch.zeros((self.max_size, *t.shape), dtype=t.dtype)

        if isinstance(data, TensorDictBase):
            out = TensorDict({}, batch_size=(self.max_size, *data.batch_size))
            for k, v in data.items(True, True):
                if isinstance(v, torch.Tensor):
                    out.set(k, make(v))
            out.memmap_(prefix)
            self._storage = out
        else:
            td = TensorDict({"data": make(data)}, batch_size=[self.max_size])
            td.memmap_(prefix)
            self._storage = td.get("data")
        self.initialized = True


class StorageEnsemble(Storage):
    """Several storages behind one index space (reference ensemble.py:18)."""

    def __init__(self, *storages: Storage):
        super().__init__(sum(s.max_size for s in storages))
        self._storages = list(storages)

    def __getitem__(self, index):
        buf, idx = index
        return self._storages[buf].get(idx)

    def __len__(self):
        return sum(len(s) for s in self._storages)

    def get(self, index):
        return self.__getitem__(index)
