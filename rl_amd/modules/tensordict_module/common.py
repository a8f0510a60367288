"""Safe modules — spec-projected outputs.

Reference: pytorch/rl torchrl/modules/tensordict_module/common.py
(SafeModule), sequence.py (SafeSequential), probabilistic.py
(SafeProbabilisticModule): when ``safe=True`` the module's outputs are
projected back into the declared spec after every call.
"""
from __future__ import annotations

from typing import Optional, Sequence

from ...data.tensor_specs import TensorSpec
from ...tensordict import (
    ProbabilisticTensorDictModule,
    TensorDictBase,
    TensorDictModule,
    TensorDictSequential,
)

__all__ = ["SafeModule", "SafeSequential", "SafeProbabilisticModule"]


class SafeModule(TensorDictModule):
    def __init__(self, module, in_keys, out_keys, spec: Optional[TensorSpec] = None, safe: bool = False):
        super().__init__(module, in_keys, out_keys)
        self.spec = spec
        self.safe = safe

    def forward(self, td=None, *args, **kwargs):
        out = super().forward(td, *args, **kwargs)
        if self.safe and self.spec is not None and isinstance(out, TensorDictBase):
            key = self.out_keys[0]
            out.set(key, self.spec.project(out.get(key)))
        return out


class SafeSequential(TensorDictSequential):
    """partial_tolerant sequential (reference sequence.py)."""

    def __init__(self, *modules, partial_tolerant: bool = True):
        super().__init__(*modules, partial_tolerant=partial_tolerant)


class SafeProbabilisticModule(ProbabilisticTensorDictModule):
    def __init__(self, *args, spec: Optional[TensorSpec] = None, safe: bool = False, **kwargs):
        super().__init__(*args, **kwargs)
        self.spec = spec
        self.safe = safe

    def forward(self, td):
        out = super().forward(td)
        if self.safe and self.spec is not None:
            key = self.out_keys[0]
            out.set(key, self.spec.project(out.get(key)))
        return out
