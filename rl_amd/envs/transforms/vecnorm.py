"""VecNorm — online observation/reward normalization with shareable stats.

Reference: pytorch/rl torchrl/envs/transforms/vecnorm.py:34 (VecNormV2) and
_normalization.py:466 (legacy VecNorm).  Keeps decayed running sum/sum-sq
per key; ``share_memory_()`` moves the stat buffers to shared memory so
ParallelEnv workers update one copy (mp.Lock-guarded on CPU; on a single
GPU the updates are naturally serialized on the HIP stream).
"""
from __future__ import annotations

import multiprocessing as mp
from typing import Optional, Sequence

import torch

from ...tensordict import TensorDict, TensorDictBase
from ._base import Transform

__all__ = ["VecNorm", "VecNormV2"]


class VecNorm(Transform):
    def __init__(
        self,
        in_keys: Sequence = ("observation", "reward"),
        out_keys: Optional[Sequence] = None,
        decay: float = 0.9999,
        eps: float = 1e-4,
        shapes: Optional[Sequence] = None,
        lock=None,
    ):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.decay = decay
        self.eps = eps
        self.lock = lock
        self._stats = TensorDict({}, batch_size=[])
        self.frozen = False

    def freeze(self) -> "VecNorm":
        self.frozen = True
        return self

    def unfreeze(self) -> "VecNorm":
        self.frozen = False
        return self

    def _init_key(self, flat: str, val: torch.Tensor):
        feat = val.shape[-1:]
        self._stats.set(flat + "_sum", torch.zeros(feat, device=val.device))
        self._stats.set(flat + "_ssq", torch.zeros(feat, device=val.device))
        self._stats.set(flat + "_count", torch.zeros(1, device=val.device))

    def _norm(self, key, val):
        flat = key if isinstance(key, str) else ".".join(key)
        if flat + "_sum" not in self._stats:
            self._init_key(flat, val)
        s = self._stats.get(flat + "_sum")
        ssq = self._stats.get(flat + "_ssq")
        count = self._stats.get(flat + "_count")
        if not self.frozen:
            x = val.detach().reshape(-1, val.shape[-1])
            n = x.shape[0]
            if self.lock is not None:
                self.lock.acquire()
            try:
                decay = self.decay**n
                s.mul_(decay).add_(x.sum(0))
                ssq.mul_(decay).add_((x * x).sum(0))
                count.mul_(decay).add_(float(n))
            finally:
                if self.lock is not None:
                    self.lock.release()
        mean = s / count.clamp_min(1.0)
        var = (ssq / count.clamp_min(1.0) - mean * mean).clamp_min(0.0)
        return (val - mean) / (var.sqrt() + self.eps)

    def _call(self, td):
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            val = td.get(in_key, None)
            if val is not None:
                td.set(out_key, self._norm(in_key, val))
        return td

    def share_memory_(self) -> "VecNorm":
        # pre-materialize nothing: stats buffers are created lazily, so call
        # after a first dummy _call, or pass shapes at construction.
        self._stats.share_memory_()
        if self.lock is None:
            self.lock = mp.Lock()
        return self

    def state_dict(self, *args, **kwargs):
        return {"stats": self._stats.to_dict(), "decay": self.decay}

    def load_state_dict(self, sd, strict: bool = True):
        for k, v in sd["stats"].items():
            self._stats.set(k, torch.as_tensor(v))

    def loc_scale(self, key):
        flat = key if isinstance(key, str) else ".".join(key)
        s = self._stats.get(flat + "_sum")
        ssq = self._stats.get(flat + "_ssq")
        count = self._stats.get(flat + "_count")
        mean = s / count.clamp_min(1.0)
        var = (ssq / count.clamp_min(1.0) - mean * mean).clamp_min(0.0)
        return mean, var.sqrt() + self.eps

    def transform_observation_spec(self, spec):
        from ..._utils import logger  # noqa: F401  (import check)
        from ...data.tensor_specs import Unbounded as _U

        for in_key, out_key in zip(self.in_keys, self.out_keys):
            if in_key in spec:
                sp = spec[in_key]
                spec[out_key] = _U(shape=sp.shape, dtype=sp.dtype, device=sp.device)
        return spec


VecNormV2 = VecNorm
