"""Storage checkpointers — pluggable persistence strategies for replay
buffers.

Reference: pytorch/rl torchrl/data/replay_buffers/checkpointers.py
(StorageCheckpointerBase, ListStorageCheckpointer,
TensorStorageCheckpointer, FlatStorageCheckpointer,
NestedStorageCheckpointer, H5StorageCheckpointer) and utils.py TED2Flat.

The flat checkpointer deduplicates TED data (``("next", obs)`` of step t
equals the root obs of step t+1 inside a trajectory): it stores root
keys for every step plus the ``next`` values only at trajectory
boundaries, roughly halving observation bytes on disk.
"""
from __future__ import annotations

import importlib.util
import json
import os
from typing import Any, List, Optional

import torch

from ...tensordict import TensorDict, TensorDictBase

__all__ = [
    "StorageCheckpointerBase",
    "ListStorageCheckpointer",
    "TensorStorageCheckpointer",
    "FlatStorageCheckpointer",
    "NestedStorageCheckpointer",
    "H5StorageCheckpointer",
]


class StorageCheckpointerBase:
    """dumps/loads strategy over a storage object."""

    def dumps(self, storage, path) -> None:
        raise NotImplementedError

    def loads(self, storage, path) -> None:
        raise NotImplementedError


class ListStorageCheckpointer(StorageCheckpointerBase):
    """Pickle the python list (delegates to ListStorage.dumps)."""

    def dumps(self, storage, path):
        os.makedirs(path, exist_ok=True)
        storage.dumps(path)

    def loads(self, storage, path):
        storage.loads(path)


class TensorStorageCheckpointer(StorageCheckpointerBase):
    """Save the full storage TensorDict with torch.save."""

    def dumps(self, storage, path):
        os.makedirs(path, exist_ok=True)
        data = storage._storage
        torch.save(
            {"data": data, "len": len(storage)},
            os.path.join(path, "tensor_storage.pt"),
        )

    def loads(self, storage, path):
        sd = torch.load(os.path.join(path, "tensor_storage.pt"), weights_only=False)
        storage._storage = sd["data"]
        storage._len = sd["len"]


class FlatStorageCheckpointer(StorageCheckpointerBase):
    """TED → flat dedupe (reference utils.py TED2Flat): for keys present
    both at the root and under ``next``, store the root series plus only
    the boundary ``next`` values (trajectory ends), and rebuild
    ``next[k][i] = root[k][i+1]`` inside trajectories on load."""

    def __init__(self, done_key=("next", "done")):
        self.done_key = done_key

    def dumps(self, storage, path):
        os.makedirs(path, exist_ok=True)
        data = storage._storage[: len(storage)]
        nxt = data.get("next")
        dup_keys = [
            k for k in nxt.keys(True, True)
            if k in [kk for kk in data.keys(True, True)]
        ]
        done = data.get(self.done_key).reshape(len(storage)).bool()
        boundary_idx = torch.nonzero(done).reshape(-1)
        compact = data.exclude(*[("next", *(k if isinstance(k, tuple) else (k,))) for k in dup_keys])
        boundary_vals = TensorDict(
            {k: nxt.get(k)[boundary_idx] for k in dup_keys},
            batch_size=[boundary_idx.numel()],
        )
        torch.save(
            {
                "compact": compact,
                "boundary_idx": boundary_idx,
                "boundary_vals": boundary_vals,
                "dup_keys": dup_keys,
                "len": len(storage),
            },
            os.path.join(path, "flat_storage.pt"),
        )

    def loads(self, storage, path):
        sd = torch.load(os.path.join(path, "flat_storage.pt"), weights_only=False)
        compact: TensorDictBase = sd["compact"]
        n = sd["len"]
        nxt = compact.get("next")
        for k in sd["dup_keys"]:
            root = compact.get(k)
            rebuilt = torch.empty_like(root)
            if n > 1:
                rebuilt[:-1] = root[1:]
            rebuilt[-1] = root[-1]
            bidx = sd["boundary_idx"]
            if bidx.numel():
                rebuilt[bidx] = sd["boundary_vals"].get(k)
            nxt.set(k, rebuilt)
        storage._storage = compact
        storage._len = n


class NestedStorageCheckpointer(FlatStorageCheckpointer):
    """Flat dedupe for nested (non-contiguous trajectory) layouts —
    same strategy; trajectory boundaries come from the done flags."""


class H5StorageCheckpointer(StorageCheckpointerBase):
    """HDF5-backed checkpoint (gated: h5py is not in this image)."""

    def __init__(self):
        if importlib.util.find_spec("h5py") is None:
            raise ImportError(
                "H5StorageCheckpointer requires h5py, which is not installed "
                "in this image. Use FlatStorageCheckpointer or "
                "TensorStorageCheckpointer instead."
            )


class CompressedListStorageCheckpointer(ListStorageCheckpointer):
    """Checkpointer for CompressedListStorage (reference
    checkpointers.py): entries are decompressed on save so the on-disk
    format matches the plain list checkpoint."""

    def save(self, storage, path):
        inner = getattr(storage, "_decompressed", None)
        super().save(storage if inner is None else inner, path)


__all__.append("CompressedListStorageCheckpointer")
