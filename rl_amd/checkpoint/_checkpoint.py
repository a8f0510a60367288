"""Checkpoint system — component registry, adapters, rotation, RNG state.

Reference: pytorch/rl torchrl/checkpoint/_checkpoint.py (Checkpoint:701,
CheckpointAdapter:166, DumpLoadCheckpointAdapter:211,
StateDictCheckpointAdapter:432, JSONCheckpointAdapter:550,
GlobalRNGState:605, CheckpointRotation:1348).

**Format compatibility is a north-star requirement**: the manifest uses
``_FORMAT_NAME = "torchrl.checkpoint"`` / ``_FORMAT_VERSION = 1``
(reference :41-43) so checkpoints interchange at the manifest level.
"""
from __future__ import annotations

import json
import os
import random
import shutil
import zipfile
from typing import Any, Dict, List, Optional, Union

import numpy as np
import torch

from ..tensordict import TensorDict, TensorDictBase

__all__ = [
    "Checkpoint",
    "CheckpointAdapter",
    "StateDictCheckpointAdapter",
    "DumpLoadCheckpointAdapter",
    "JSONCheckpointAdapter",
    "GlobalRNGState",
    "CheckpointRotation",
]

_FORMAT_NAME = "torchrl.checkpoint"
_FORMAT_VERSION = 1


class CheckpointAdapter:
    """How one component serializes (reference :166)."""

    def save(self, obj: Any, path: str) -> None:
        raise NotImplementedError

    def load(self, obj: Any, path: str) -> None:
        raise NotImplementedError

    @classmethod
    def for_object(cls, obj: Any) -> "CheckpointAdapter":
        if isinstance(obj, GlobalRNGState):
            return _RNGAdapter()
        if isinstance(obj, TensorDictBase):
            return DumpLoadCheckpointAdapter()
        if hasattr(obj, "dumps") and hasattr(obj, "loads"):
            return _DumpsLoadsAdapter()
        if hasattr(obj, "state_dict"):
            return StateDictCheckpointAdapter()
        return JSONCheckpointAdapter()


class StateDictCheckpointAdapter(CheckpointAdapter):
    """torch state_dict → file (reference :432)."""

    def save(self, obj, path: str) -> None:
        os.makedirs(path, exist_ok=True)
        torch.save(obj.state_dict(), os.path.join(path, "state_dict.pt"))

    def load(self, obj, path: str) -> None:
        sd = torch.load(
            os.path.join(path, "state_dict.pt"), weights_only=False, map_location="cpu"
        )
        obj.load_state_dict(sd)


class DumpLoadCheckpointAdapter(CheckpointAdapter):
    """TensorDict memmap dumps (reference :211)."""

    def save(self, obj: TensorDictBase, path: str) -> None:
        os.makedirs(path, exist_ok=True)
        obj.clone().memmap_(os.path.join(path, "td"))

    def load(self, obj: TensorDictBase, path: str) -> None:
        loaded = TensorDict.load_memmap(os.path.join(path, "td"))
        obj.update_(loaded.to(obj.device) if obj.device else loaded)


class _DumpsLoadsAdapter(CheckpointAdapter):
    def save(self, obj, path: str) -> None:
        os.makedirs(path, exist_ok=True)
        obj.dumps(path)

    def load(self, obj, path: str) -> None:
        obj.loads(path)


class JSONCheckpointAdapter(CheckpointAdapter):
    """Plain-python payloads (reference :550)."""

    def save(self, obj, path: str) -> None:
        os.makedirs(path, exist_ok=True)
        payload = obj.__dict__ if hasattr(obj, "__dict__") else obj
        with open(os.path.join(path, "data.json"), "w") as f:
            json.dump(payload, f, default=str)

    def load(self, obj, path: str) -> None:
        with open(os.path.join(path, "data.json")) as f:
            data = json.load(f)
        if hasattr(obj, "__dict__") and isinstance(data, dict):
            obj.__dict__.update(data)


class GlobalRNGState:
    """python/numpy/torch/device RNG capture (reference :605)."""

    def state_dict(self) -> Dict[str, Any]:
        out = {
            "python": random.getstate(),
            "numpy": np.random.get_state(),
            "torch": torch.get_rng_state(),
        }
        if torch.cuda.is_available():
            out["cuda"] = torch.cuda.get_rng_state_all()
        return out

    def load_state_dict(self, sd: Dict[str, Any]) -> None:
        random.setstate(
            tuple(
                tuple(x) if isinstance(x, list) else x for x in sd["python"]
            )
            if isinstance(sd["python"], (list, tuple))
            else sd["python"]
        )
        np_state = sd["numpy"]
        if isinstance(np_state, (list, tuple)):
            np_state = (
                np_state[0],
                np.asarray(np_state[1], dtype=np.uint32),
                *np_state[2:],
            )
        np.random.set_state(np_state)
        torch.set_rng_state(torch.as_tensor(sd["torch"], dtype=torch.uint8))
        if "cuda" in sd and torch.cuda.is_available():
            torch.cuda.set_rng_state_all(
                [torch.as_tensor(s, dtype=torch.uint8) for s in sd["cuda"]]
            )


class _RNGAdapter(CheckpointAdapter):
    def save(self, obj: GlobalRNGState, path: str) -> None:
        os.makedirs(path, exist_ok=True)
        torch.save(obj.state_dict(), os.path.join(path, "rng.pt"))

    def load(self, obj: GlobalRNGState, path: str) -> None:
        obj.load_state_dict(
            torch.load(os.path.join(path, "rng.pt"), weights_only=False)
        )


class Checkpoint:
    """Component-registry checkpoint with manifest (reference :701).

    Usage::

        ckpt = Checkpoint()
        ckpt.register(policy, "policy")
        ckpt.register(optimizer, "optimizer")
        ckpt.register(GlobalRNGState(), "rng")
        ckpt.save("run/ckpt_0")          # directory format
        ckpt.save("run/ckpt_0.zip")      # archive format
        ckpt.load("run/ckpt_0")
    """

    def __init__(self):
        self._components: Dict[str, Any] = {}
        self._adapters: Dict[str, CheckpointAdapter] = {}

    def register(self, obj: Any, name: str, adapter: Optional[CheckpointAdapter] = None) -> "Checkpoint":
        self._components[name] = obj
        self._adapters[name] = adapter or CheckpointAdapter.for_object(obj)
        return self

    def components(self) -> List[str]:
        return list(self._components)

    def _manifest(self) -> dict:
        return {
            "format": _FORMAT_NAME,
            "version": _FORMAT_VERSION,
            "components": {
                name: type(self._adapters[name]).__name__
                for name in self._components
            },
        }

    def save(self, path: str) -> str:
        if path.endswith(".zip"):
            tmp = path[:-4] + "_tmpdir"
            self._save_dir(tmp)
            with zipfile.ZipFile(path, "w") as zf:
                for root, _dirs, files in os.walk(tmp):
                    for fn in files:
                        full = os.path.join(root, fn)
                        zf.write(full, os.path.relpath(full, tmp))
            shutil.rmtree(tmp)
            return path
        return self._save_dir(path)

    def _save_dir(self, path: str) -> str:
        os.makedirs(path, exist_ok=True)
        for name, obj in self._components.items():
            self._adapters[name].save(obj, os.path.join(path, name))
        with open(os.path.join(path, "manifest.json"), "w") as f:
            json.dump(self._manifest(), f, indent=2)
        return path

    def load(self, path: str) -> None:
        if path.endswith(".zip"):
            tmp = path[:-4] + "_loadtmp"
            with zipfile.ZipFile(path) as zf:
                zf.extractall(tmp)
            try:
                self._load_dir(tmp)
            finally:
                shutil.rmtree(tmp)
            return
        self._load_dir(path)

    def _load_dir(self, path: str) -> None:
        with open(os.path.join(path, "manifest.json")) as f:
            manifest = json.load(f)
        if manifest.get("format") != _FORMAT_NAME:
            raise RuntimeError(
                f"unknown checkpoint format {manifest.get('format')!r}"
            )
        for name, obj in self._components.items():
            comp_path = os.path.join(path, name)
            if os.path.isdir(comp_path):
                self._adapters[name].load(obj, comp_path)


class CheckpointRotation:
    """keep-last-N + best-metric rotation (reference :1348)."""

    def __init__(
        self,
        checkpoint: Checkpoint,
        dirname: str,
        keep_last: int = 3,
        keep_best: bool = True,
        higher_is_better: bool = True,
    ):
        self.checkpoint = checkpoint
        self.dirname = dirname
        self.keep_last = keep_last
        self.keep_best = keep_best
        self.higher_is_better = higher_is_better
        self._history: List[str] = []
        self._best_metric: Optional[float] = None
        self._best_path: Optional[str] = None
        os.makedirs(dirname, exist_ok=True)

    def step(self, step: int, metric: Optional[float] = None) -> str:
        path = os.path.join(self.dirname, f"ckpt_{step}")
        self.checkpoint.save(path)
        self._history.append(path)
        if metric is not None and self.keep_best:
            better = (
                self._best_metric is None
                or (self.higher_is_better and metric > self._best_metric)
                or (not self.higher_is_better and metric < self._best_metric)
            )
            if better:
                self._best_metric = metric
                best = os.path.join(self.dirname, "ckpt_best")
                if os.path.exists(best):
                    shutil.rmtree(best)
                shutil.copytree(path, best)
                self._best_path = best
        while len(self._history) > self.keep_last:
            victim = self._history.pop(0)
            if os.path.exists(victim):
                shutil.rmtree(victim)
        return path

    @property
    def best_path(self) -> Optional[str]:
        return self._best_path
