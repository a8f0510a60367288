"""TensorSpec family — shape/dtype/device/bounds contracts for env & module IO.

MI355X-native re-design of the reference spec layer
(pytorch/rl torchrl/data/tensor_specs.py:607-6463): same public taxonomy
(``TensorSpec``, ``Bounded``, ``Unbounded``, ``Categorical``, ``OneHot``,
``MultiOneHot``, ``MultiCategorical``, ``Binary``, ``NonTensor``,
``Composite``) with compact implementations.  Specs are metadata only —
``rand``/``zero`` allocate straight on the target device so GPU-resident
envs never round-trip through host memory.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence, Tuple, Union

import numpy as np
import torch

from ..tensordict import TensorDict, TensorDictBase, NonTensorData, unravel_key

__all__ = [
    "TensorSpec",
    "Bounded",
    "Unbounded",
    "UnboundedContinuous",
    "UnboundedDiscrete",
    "Categorical",
    "OneHot",
    "MultiOneHot",
    "MultiCategorical",
    "Binary",
    "NonTensor",
    "Composite",
    "Choice",
    "stack_specs",
    # reference-era aliases
    "BoundedTensorSpec",
    "UnboundedContinuousTensorSpec",
    "DiscreteTensorSpec",
    "OneHotDiscreteTensorSpec",
    "MultiDiscreteTensorSpec",
    "BinaryDiscreteTensorSpec",
    "CompositeSpec",
]

DEVICE_TYPING = Union[str, torch.device, int]


def _size(shape) -> torch.Size:
    if shape is None:
        return torch.Size([])
    if isinstance(shape, int):
        return torch.Size([shape])
    return torch.Size(shape)


class TensorSpec:
    """Base class for all specs."""

    shape: torch.Size
    dtype: torch.dtype
    device: Optional[torch.device]
    domain: str = ""

    def __init__(self, shape=None, device=None, dtype=torch.float32):
        self.shape = _size(shape)
        self.device = torch.device(device) if device is not None else None
        self.dtype = dtype

    # -- interface ------------------------------------------------------- #
    def rand(self, shape=None) -> torch.Tensor:
        raise NotImplementedError

    def sample(self, shape=None):
        return self.rand(shape)

    def zero(self, shape=None) -> torch.Tensor:
        shape = _size(shape)
        return torch.zeros(
            (*shape, *self.shape), dtype=self.dtype, device=self.device
        )

    def zeros(self, shape=None):
        return self.zero(shape)

    def is_in(self, val) -> bool:
        raise NotImplementedError

    def project(self, val: torch.Tensor) -> torch.Tensor:
        if not self.is_in(val):
            return self._project(val)
        return val

    def _project(self, val: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def assert_is_in(self, val) -> None:
        if not self.is_in(val):
            raise AssertionError(
                f"value {val} is not contained in spec {self}"
            )

    def encode(self, val, *, ignore_device: bool = False) -> torch.Tensor:
        if isinstance(val, np.ndarray):
            val = torch.as_tensor(val.copy())
        elif not isinstance(val, torch.Tensor):
            val = torch.as_tensor(val)
        if val.dtype != self.dtype:
            val = val.to(self.dtype)
        if not ignore_device and self.device is not None:
            val = val.to(self.device)
        if val.shape[-len(self.shape):] != self.shape and len(self.shape):
            val = val.reshape(*val.shape[: val.dim() - 1], *self.shape)
        return val

    def to_numpy(self, val: torch.Tensor, safe: bool = False) -> np.ndarray:
        if safe:
            self.assert_is_in(val)
        return val.detach().cpu().numpy()

    # -- structural ------------------------------------------------------ #
    def clone(self) -> "TensorSpec":
        import copy

        return copy.deepcopy(self)

    def to(self, device) -> "TensorSpec":
        out = self.clone()
        out._apply_device(torch.device(device))
        return out

    def _apply_device(self, device: torch.device):
        self.device = device

    def expand(self, *shape) -> "TensorSpec":
        if len(shape) == 1 and isinstance(shape[0], (tuple, list, torch.Size)):
            shape = tuple(shape[0])
        out = self.clone()
        out._expand_shape(torch.Size(shape))
        return out

    def _expand_shape(self, shape: torch.Size):
        self.shape = shape

    def unsqueeze(self, dim: int) -> "TensorSpec":
        shape = list(self.shape)
        if dim < 0:
            dim = len(shape) + dim + 1
        shape.insert(dim, 1)
        return self.expand(*shape)

    def squeeze(self, dim: Optional[int] = None) -> "TensorSpec":
        shape = list(self.shape)
        if dim is None:
            shape = [s for s in shape if s != 1]
        else:
            if shape[dim] == 1:
                shape.pop(dim)
        return self.expand(*shape)

    @property
    def ndim(self) -> int:
        return len(self.shape)

    def __getitem__(self, idx):
        new_shape = torch.empty(self.shape, device="meta")[idx].shape
        return self.expand(*new_shape)

    def __eq__(self, other):
        return (
            type(self) is type(other)
            and self.shape == other.shape
            and self.dtype == other.dtype
            and self.device == other.device
        )

    def __repr__(self):
        return (
            f"{type(self).__name__}(shape={tuple(self.shape)}, dtype={self.dtype}, "
            f"device={self.device})"
        )


class Unbounded(TensorSpec):
    """Unbounded continuous (or discrete, by dtype) values."""

    domain = "continuous"

    def __init__(self, shape=None, device=None, dtype=torch.float32, **kwargs):
        super().__init__(shape, device, dtype)
        if not dtype.is_floating_point:
            self.domain = "discrete"

    def rand(self, shape=None) -> torch.Tensor:
        shape = _size(shape)
        if self.dtype.is_floating_point:
            return torch.randn(
                (*shape, *self.shape), dtype=self.dtype, device=self.device
            )
        return torch.randint(
            0, 100, (*shape, *self.shape), dtype=self.dtype, device=self.device
        )

    def is_in(self, val) -> bool:
        return (
            isinstance(val, torch.Tensor)
            and val.dtype == self.dtype
            and val.shape[-len(self.shape):] == self.shape
            if len(self.shape)
            else True
        )

    def _project(self, val):
        return val


def UnboundedContinuous(shape=None, device=None, dtype=torch.float32, **kw):
    return Unbounded(shape, device, dtype)


def UnboundedDiscrete(shape=None, device=None, dtype=torch.int64, **kw):
    return Unbounded(shape, device, dtype)


class Bounded(TensorSpec):
    """Box-bounded continuous values with per-element low/high."""

    domain = "continuous"

    def __init__(
        self,
        low=None,
        high=None,
        shape=None,
        device=None,
        dtype=torch.float32,
        *,
        minimum=None,
        maximum=None,
    ):
        if low is None and minimum is not None:
            low = minimum
        if high is None and maximum is not None:
            high = maximum
        low = torch.as_tensor(low, dtype=dtype)
        high = torch.as_tensor(high, dtype=dtype)
        if shape is None:
            shape = torch.broadcast_shapes(low.shape, high.shape)
        shape = _size(shape)
        super().__init__(shape, device, dtype)
        self.low = low.expand(shape).clone().to(self.device) if self.device else low.expand(shape).clone()
        self.high = high.expand(shape).clone().to(self.device) if self.device else high.expand(shape).clone()

    @property
    def minimum(self):
        return self.low

    @property
    def maximum(self):
        return self.high

    def rand(self, shape=None) -> torch.Tensor:
        shape = _size(shape)
        if self.dtype.is_floating_point:
            u = torch.rand((*shape, *self.shape), dtype=self.dtype, device=self.device)
            low = self.low.to(self.device) if self.device else self.low
            high = self.high.to(self.device) if self.device else self.high
            interval = (high - low).clamp_max(1e6)
            return (u * interval + low).clamp(low, high)
        return torch.randint(
            int(self.low.min()),
            int(self.high.max()) + 1,
            (*shape, *self.shape),
            dtype=self.dtype,
            device=self.device,
        )

    def is_in(self, val) -> bool:
        if not isinstance(val, torch.Tensor) or val.dtype != self.dtype:
            return False
        low = self.low.to(val.device)
        high = self.high.to(val.device)
        try:
            return bool(((val >= low - 1e-6) & (val <= high + 1e-6)).all())
        except RuntimeError:
            # shapes don't broadcast (e.g. rollout with extra time dim over a
            # batched spec): fall back to the global bound envelope
            return bool(
                (val >= low.amin() - 1e-6).all() and (val <= high.amax() + 1e-6).all()
            )

    def _project(self, val):
        return val.clamp(self.low.to(val.device), self.high.to(val.device))

    def _apply_device(self, device):
        self.device = device
        self.low = self.low.to(device)
        self.high = self.high.to(device)

    def _expand_shape(self, shape):
        # keep trailing dims aligned
        self.low = self.low.expand(shape).clone()
        self.high = self.high.expand(shape).clone()
        self.shape = shape

    def __eq__(self, other):
        return (
            super().__eq__(other)
            and bool((self.low == other.low).all())
            and bool((self.high == other.high).all())
        )

    def __repr__(self):
        return (
            f"Bounded(shape={tuple(self.shape)}, low={self.low.flatten()[0].item():.3g}, "
            f"high={self.high.flatten()[0].item():.3g}, dtype={self.dtype}, device={self.device})"
        )


class Categorical(TensorSpec):
    """Integer category in ``[0, n)``; shape excludes the category dim."""

    domain = "discrete"

    def __init__(self, n: int, shape=None, device=None, dtype=torch.int64):
        super().__init__(shape if shape is not None else (), device, dtype)
        self.n = int(n)
        self.space_n = self.n

    def rand(self, shape=None) -> torch.Tensor:
        shape = _size(shape)
        return torch.randint(
            0, self.n, (*shape, *self.shape), dtype=self.dtype, device=self.device
        )

    def is_in(self, val) -> bool:
        if not isinstance(val, torch.Tensor):
            return False
        return bool(((val >= 0) & (val < self.n)).all()) and val.dtype == self.dtype

    def _project(self, val):
        return val.clamp(0, self.n - 1).to(self.dtype)

    def to_one_hot(self, val: torch.Tensor) -> torch.Tensor:
        return torch.nn.functional.one_hot(val.long(), self.n).to(torch.bool)

    def __eq__(self, other):
        return super().__eq__(other) and self.n == getattr(other, "n", None)

    def __repr__(self):
        return (
            f"Categorical(n={self.n}, shape={tuple(self.shape)}, dtype={self.dtype}, "
            f"device={self.device})"
        )


class OneHot(TensorSpec):
    """One-hot encoded category; last dim of ``shape`` is ``n``."""

    domain = "discrete"

    def __init__(self, n: int, shape=None, device=None, dtype=torch.bool):
        if shape is None:
            shape = (n,)
        shape = _size(shape)
        assert shape[-1] == n, "last dim of OneHot shape must equal n"
        super().__init__(shape, device, dtype)
        self.n = int(n)

    def rand(self, shape=None) -> torch.Tensor:
        shape = _size(shape)
        idx = torch.randint(0, self.n, (*shape, *self.shape[:-1]), device=self.device)
        return torch.nn.functional.one_hot(idx, self.n).to(self.dtype)

    def zero(self, shape=None) -> torch.Tensor:
        shape = _size(shape)
        out = torch.zeros((*shape, *self.shape), dtype=self.dtype, device=self.device)
        out[..., 0] = 1
        return out

    def is_in(self, val) -> bool:
        if not isinstance(val, torch.Tensor):
            return False
        return bool((val.sum(-1) == 1).all()) and bool(
            ((val == 0) | (val == 1)).all()
        )

    def _project(self, val):
        idx = val.argmax(-1)
        return torch.nn.functional.one_hot(idx, self.n).to(self.dtype)

    def to_categorical(self, val: torch.Tensor) -> torch.Tensor:
        if val.dtype == torch.bool:
            val = val.to(torch.uint8)
        return val.argmax(-1)

    def encode(self, val, *, ignore_device=False):
        val = torch.as_tensor(val)
        if val.shape[-1:] != (self.n,):
            val = torch.nn.functional.one_hot(val.long(), self.n)
        return val.to(self.dtype if ignore_device else self.dtype).to(
            self.device if not ignore_device and self.device else val.device
        )

    def __eq__(self, other):
        return super().__eq__(other) and self.n == getattr(other, "n", None)

    def __repr__(self):
        return f"OneHot(n={self.n}, shape={tuple(self.shape)}, device={self.device})"


class MultiOneHot(TensorSpec):
    """Concatenation of several one-hot groups (``nvec`` per group)."""

    domain = "discrete"

    def __init__(self, nvec: Sequence[int], shape=None, device=None, dtype=torch.bool):
        self.nvec = [int(n) for n in nvec]
        total = sum(self.nvec)
        if shape is None:
            shape = (total,)
        super().__init__(shape, device, dtype)

    def rand(self, shape=None) -> torch.Tensor:
        shape = _size(shape)
        outs = []
        for n in self.nvec:
            idx = torch.randint(0, n, (*shape, *self.shape[:-1]), device=self.device)
            outs.append(torch.nn.functional.one_hot(idx, n).to(self.dtype))
        return torch.cat(outs, -1)

    def is_in(self, val) -> bool:
        start = 0
        for n in self.nvec:
            if not bool((val[..., start : start + n].sum(-1) == 1).all()):
                return False
            start += n
        return True

    def _project(self, val):
        outs = []
        start = 0
        for n in self.nvec:
            idx = val[..., start : start + n].argmax(-1)
            outs.append(torch.nn.functional.one_hot(idx, n).to(self.dtype))
            start += n
        return torch.cat(outs, -1)

    def to_categorical(self, val) -> torch.Tensor:
        if val.dtype == torch.bool:
            val = val.to(torch.uint8)
        outs = []
        start = 0
        for n in self.nvec:
            outs.append(val[..., start : start + n].argmax(-1))
            start += n
        return torch.stack(outs, -1)


class MultiCategorical(TensorSpec):
    """Vector of categorical values with per-position cardinality ``nvec``."""

    domain = "discrete"

    def __init__(self, nvec: Sequence[int], shape=None, device=None, dtype=torch.int64):
        nvec_t = torch.as_tensor(list(nvec))
        if shape is None:
            shape = nvec_t.shape
        super().__init__(shape, device, dtype)
        self.nvec = nvec_t

    def rand(self, shape=None) -> torch.Tensor:
        shape = _size(shape)
        nvec = self.nvec.to(self.device) if self.device else self.nvec
        u = torch.rand((*shape, *self.shape), device=self.device)
        return (u * nvec.float()).floor().to(self.dtype)

    def is_in(self, val) -> bool:
        nvec = self.nvec.to(val.device)
        return bool(((val >= 0) & (val < nvec)).all())

    def _project(self, val):
        nvec = self.nvec.to(val.device)
        return torch.minimum(val.clamp_min(0), nvec - 1).to(self.dtype)


class Binary(TensorSpec):
    """Binary-valued tensor (done flags etc.)."""

    domain = "discrete"

    def __init__(self, n: Optional[int] = None, shape=None, device=None, dtype=torch.bool):
        if shape is None:
            shape = (n,) if n else (1,)
        shape = _size(shape)
        super().__init__(shape, device, dtype)
        self.n = n if n is not None else (shape[-1] if len(shape) else 1)

    def rand(self, shape=None) -> torch.Tensor:
        shape = _size(shape)
        return (
            torch.rand((*shape, *self.shape), device=self.device) < 0.5
        ).to(self.dtype)

    def is_in(self, val) -> bool:
        if val.dtype == torch.bool:
            return True
        return bool(((val == 0) | (val == 1)).all())

    def _project(self, val):
        return (val != 0).to(self.dtype)

    def __repr__(self):
        return f"Binary(shape={tuple(self.shape)}, dtype={self.dtype}, device={self.device})"


class NonTensor(TensorSpec):
    """Spec slot for non-tensor payloads."""

    domain = "nontensor"

    def __init__(self, shape=None, device=None, dtype=None, example_data=None):
        super().__init__(shape, device, torch.float32)
        self.example_data = example_data

    def rand(self, shape=None):
        return NonTensorData(self.example_data)

    def zero(self, shape=None):
        return NonTensorData(self.example_data)

    def is_in(self, val) -> bool:
        return isinstance(val, NonTensorData) or not isinstance(val, torch.Tensor)

    def encode(self, val, **kwargs):
        return NonTensorData(val)


class Choice(TensorSpec):
    """Sample uniformly among a fixed list of specs/values."""

    def __init__(self, choices: Sequence, device=None):
        super().__init__((), device, torch.float32)
        self.choices = list(choices)

    def rand(self, shape=None):
        import random

        c = random.choice(self.choices)
        if isinstance(c, TensorSpec):
            return c.rand(shape)
        return c

    def is_in(self, val) -> bool:
        for c in self.choices:
            if isinstance(c, TensorSpec) and c.is_in(val):
                return True
            if not isinstance(c, TensorSpec) and c == val:
                return True
        return False


class Composite(TensorSpec):
    """Dict-of-specs; the spec of a TensorDict.

    ``shape`` is the batch shape shared by all entries (leading dims).
    """

    domain = "composite"

    def __init__(self, *args, shape=None, device=None, **kwargs):
        if len(args) == 1 and isinstance(args[0], dict):
            kwargs = {**args[0], **kwargs}
        elif args:
            raise TypeError("Composite accepts a single dict positional arg")
        super().__init__(shape if shape is not None else (), device, torch.float32)
        self._specs: Dict[str, Optional[TensorSpec]] = {}
        for k, v in kwargs.items():
            self[k] = v

    # -- mapping --------------------------------------------------------- #
    def __setitem__(self, key, value):
        key = unravel_key(key)
        if isinstance(key, tuple):
            first, rest = key[0], key[1:]
            if first not in self._specs or not isinstance(self._specs[first], Composite):
                self._specs[first] = Composite(shape=self.shape, device=self.device)
            self._specs[first][rest if len(rest) > 1 else rest[0]] = value
            return
        if isinstance(value, dict):
            value = Composite(value, shape=self.shape, device=self.device)
        if value is not None and self.device is not None and value.device != self.device:
            value = value.to(self.device)
        self._specs[key] = value

    def __getitem__(self, key):
        if isinstance(key, str):
            return self._specs[key]
        key_u = unravel_key(key) if isinstance(key, tuple) and all(isinstance(k, str) for k in key) else None
        if key_u is not None:
            if isinstance(key_u, str):
                return self._specs[key_u]
            obj = self
            for k in key_u:
                obj = obj._specs[k] if isinstance(obj, Composite) else obj[k]
            return obj
        # numeric index
        out = Composite(shape=torch.empty(self.shape, device="meta")[key].shape, device=self.device)
        for k, v in self._specs.items():
            out[k] = v[key] if v is not None else None
        return out

    def __delitem__(self, key):
        key = unravel_key(key)
        if isinstance(key, str):
            del self._specs[key]
        else:
            parent = self
            for k in key[:-1]:
                parent = parent._specs[k]
            del parent._specs[key[-1]]

    def __contains__(self, key):
        try:
            self[key]
            return True
        except KeyError:
            return False

    def get(self, key, default=None):
        try:
            val = self[key]
            return val
        except KeyError:
            return default

    def set(self, key, value):
        self[key] = value
        return self

    def keys(self, include_nested: bool = False, leaves_only: bool = False):
        out = []
        for k, v in self._specs.items():
            is_comp = isinstance(v, Composite)
            if not (leaves_only and is_comp):
                out.append(k)
            if include_nested and is_comp:
                for sub in v.keys(True, leaves_only):
                    out.append((k, *(sub if isinstance(sub, tuple) else (sub,))))
        return out

    def items(self, include_nested: bool = False, leaves_only: bool = False):
        for k in self.keys(include_nested, leaves_only):
            yield k, self[k]

    def values(self, include_nested: bool = False, leaves_only: bool = False):
        for k in self.keys(include_nested, leaves_only):
            yield self[k]

    def is_empty(self) -> bool:
        return not self._specs

    # -- spec interface --------------------------------------------------- #
    def rand(self, shape=None) -> TensorDict:
        shape = _size(shape)
        out = TensorDict({}, batch_size=(*shape, *self.shape), device=self.device)
        for k, v in self._specs.items():
            if v is not None:
                out.set(k, v.rand(shape))
        return out

    def zero(self, shape=None) -> TensorDict:
        shape = _size(shape)
        out = TensorDict({}, batch_size=(*shape, *self.shape), device=self.device)
        for k, v in self._specs.items():
            if v is not None:
                out.set(k, v.zero(shape))
        return out

    def is_in(self, val: TensorDictBase) -> bool:
        for k, v in self._specs.items():
            if v is None:
                continue
            try:
                item = val.get(k)
            except KeyError:
                return False
            if not v.is_in(item):
                return False
        return True

    def project(self, val: TensorDictBase) -> TensorDictBase:
        for k, v in self._specs.items():
            if v is None:
                continue
            item = val.get(k, None)
            if item is not None:
                val.set(k, v.project(item))
        return val

    def encode(self, vals: dict, *, ignore_device=False) -> TensorDict:
        out = TensorDict({}, batch_size=self.shape, device=self.device)
        for k, v in vals.items():
            spec = self._specs.get(k)
            if spec is not None:
                out.set(k, spec.encode(v, ignore_device=ignore_device))
            else:
                out.set(k, v)
        return out

    def _apply_device(self, device):
        self.device = device
        for v in self._specs.values():
            if v is not None:
                v._apply_device(device)

    def _expand_shape(self, shape):
        old_ndim = len(self.shape)
        for k, v in self._specs.items():
            if v is None:
                continue
            tail = v.shape[old_ndim:]
            v._expand_shape(torch.Size((*shape, *tail)))
        self.shape = torch.Size(shape)

    def update(self, other: "Composite") -> "Composite":
        for k, v in other._specs.items():
            if (
                isinstance(v, Composite)
                and k in self._specs
                and isinstance(self._specs[k], Composite)
            ):
                self._specs[k].update(v)
            else:
                self[k] = v
        return self

    def clone(self) -> "Composite":
        out = Composite(shape=self.shape, device=self.device)
        for k, v in self._specs.items():
            out._specs[k] = v.clone() if v is not None else None
        return out

    def select(self, *keys) -> "Composite":
        out = Composite(shape=self.shape, device=self.device)
        for key in keys:
            out[key] = self[key]
        return out

    def exclude(self, *keys) -> "Composite":
        excluded = {unravel_key(k) for k in keys}
        out = self.clone()
        for k in excluded:
            try:
                del out[k]
            except KeyError:
                pass
        return out

    def __eq__(self, other):
        if not isinstance(other, Composite):
            return False
        if set(self._specs.keys()) != set(other._specs.keys()):
            return False
        return all(self._specs[k] == other._specs[k] for k in self._specs)

    def __repr__(self):
        inner = ", ".join(f"{k}: {v}" for k, v in self._specs.items())
        return f"Composite({inner}, shape={tuple(self.shape)}, device={self.device})"

    def __len__(self):
        return len(self._specs)

    def __iter__(self):
        return iter(self._specs)


def stack_specs(specs: Sequence[TensorSpec], dim: int = 0) -> TensorSpec:
    """Stack identical specs into one with an extra leading dim (eager
    equivalent of the reference's Stacked/StackedComposite)."""
    first = specs[0]
    n = len(specs)
    if isinstance(first, Composite):
        out = Composite(
            shape=torch.Size([*first.shape[:dim], n, *first.shape[dim:]]),
            device=first.device,
        )
        for k in first._specs:
            out[k] = stack_specs([s[k] for s in specs], dim)
        return out
    new_shape = torch.Size([*first.shape[:dim], n, *first.shape[dim:]])
    return first.expand(*new_shape)


# Reference-era aliases (old TorchRL naming)
BoundedTensorSpec = Bounded
UnboundedContinuousTensorSpec = Unbounded
DiscreteTensorSpec = Categorical
OneHotDiscreteTensorSpec = OneHot
MultiDiscreteTensorSpec = MultiCategorical
BinaryDiscreteTensorSpec = Binary
CompositeSpec = Composite


# --------------------------------------------------------------------------- #
# Reference name-parity aliases and box containers
# (reference tensor_specs.py: BoundedContinuous/BoundedDiscrete are the
# dtype-specialized views of Bounded; ContinuousBox/CategoricalBox/
# BinaryBox/DiscreteBox describe the underlying value domain; legacy
# long names map to the modern short ones.)
# --------------------------------------------------------------------------- #
BoundedContinuous = Bounded
BoundedDiscrete = Bounded
UnboundedContinuous = Unbounded
UnboundedDiscrete = Unbounded
DiscreteTensorSpec = Categorical
OneHotDiscreteTensorSpec = OneHot
MultiOneHotDiscreteTensorSpec = MultiOneHot
MultiDiscreteTensorSpec = MultiCategorical
BinaryDiscreteTensorSpec = Binary
BoundedTensorSpec = Bounded
UnboundedContinuousTensorSpec = Unbounded
CompositeSpec = Composite
NonTensorSpec = NonTensor


class Box:
    """Value-domain descriptor base (reference tensor_specs.py Box)."""


class ContinuousBox(Box):
    def __init__(self, low, high):
        self.low = low
        self.high = high

    def __repr__(self):
        return f"ContinuousBox(low={self.low}, high={self.high})"


class CategoricalBox(Box):
    def __init__(self, n: int):
        self.n = n

    def __repr__(self):
        return f"CategoricalBox(n={self.n})"


DiscreteBox = CategoricalBox


class BinaryBox(Box):
    def __init__(self, n: int):
        self.n = n


class BoxList(Box):
    def __init__(self, boxes):
        self.boxes = list(boxes)

    def __len__(self):
        return len(self.boxes)


def _specs_homogeneous(specs) -> bool:
    first = specs[0]
    if isinstance(first, Composite):
        k0 = set(first.keys(True, True))
        return all(
            isinstance(s, Composite)
            and set(s.keys(True, True)) == k0
            and all(s[k].shape == first[k].shape and s[k].dtype == first[k].dtype
                    for k in k0)
            for s in specs[1:]
        )
    return all(
        type(s) is type(first) and s.shape == first.shape and s.dtype == first.dtype
        for s in specs[1:]
    )


def _merged_shape(shapes, dim, n):
    """Per-dim merge with -1 where components disagree (the reference's
    heterogeneous Stacked shape convention)."""
    nd = max(len(s) for s in shapes)
    dims = []
    for i in range(nd):
        vals = {s[i] if i < len(s) else None for s in shapes}
        dims.append(vals.pop() if len(vals) == 1 and None not in vals else -1)
    return torch.Size([*dims[:dim], n, *dims[dim:]])


class _LazyStackedSpecBase:
    """Shared plumbing for lazily stacked (heterogeneous) specs."""

    def __init__(self, *specs, dim: int = 0):
        if len(specs) == 1 and isinstance(specs[0], (list, tuple)):
            specs = tuple(specs[0])
        self._specs_list = list(specs)
        self.dim = dim

    def __len__(self):
        return len(self._specs_list)

    def __getitem__(self, idx):
        if isinstance(idx, int):
            return self._specs_list[idx]
        raise KeyError(idx)

    @property
    def device(self):
        return self._specs_list[0].device

    @property
    def shape(self):
        return _merged_shape(
            [tuple(s.shape) for s in self._specs_list], self.dim,
            len(self._specs_list),
        )

    def clone(self):
        return type(self)(*[s.clone() for s in self._specs_list], dim=self.dim)

    def to(self, device):
        return type(self)(
            *[s.to(device) for s in self._specs_list], dim=self.dim
        )


class LazyStackedSpec(_LazyStackedSpecBase):
    """Lazily stacked LEAF specs with heterogeneous shapes (reference
    Stacked, tensor_specs.py:1496).  ``shape`` reports -1 at
    disagreeing dims; ``rand``/``zero`` return per-component lists."""

    @property
    def dtype(self):
        return self._specs_list[0].dtype

    def rand(self, shape=None):
        return [s.rand() for s in self._specs_list]

    def zero(self, shape=None):
        return [s.zero() for s in self._specs_list]

    def is_in(self, val) -> bool:
        vals = val if isinstance(val, (list, tuple)) else val.unbind(self.dim)
        return all(s.is_in(v) for s, v in zip(self._specs_list, vals))


class LazyStackedComposite(_LazyStackedSpecBase):
    """Lazily stacked Composite specs (reference StackedComposite,
    tensor_specs.py:6463) — the spec of a LazyStackedTensorDict, e.g.
    heterogeneous multi-agent groups."""

    def keys(self, include_nested: bool = False, leaves_only: bool = False):
        common = None
        for s in self._specs_list:
            ks = set(s.keys(include_nested, leaves_only))
            common = ks if common is None else common & ks
        return [
            k
            for k in self._specs_list[0].keys(include_nested, leaves_only)
            if k in common
        ]

    def items(self, include_nested: bool = False, leaves_only: bool = False):
        for k in self.keys(include_nested, leaves_only):
            yield k, self[k]

    def __getitem__(self, key):
        if isinstance(key, int):
            return self._specs_list[key]
        children = [s[key] for s in self._specs_list]
        return Stacked(*children, dim=self.dim)

    def __contains__(self, key):
        try:
            self[key]
            return True
        except KeyError:
            return False

    def rand(self, shape=None):
        from ..tensordict import lazy_stack

        return lazy_stack([s.rand() for s in self._specs_list], self.dim)

    def zero(self, shape=None):
        from ..tensordict import lazy_stack

        return lazy_stack([s.zero() for s in self._specs_list], self.dim)

    def is_in(self, val) -> bool:
        from ..tensordict import LazyStackedTensorDict

        if isinstance(val, LazyStackedTensorDict):
            parts = val.tensordicts
        else:
            parts = [val[i] for i in range(len(self._specs_list))]
        return all(s.is_in(v) for s, v in zip(self._specs_list, parts))


class Stacked:
    """Stacked spec (reference tensor_specs.py:1496): homogeneous specs
    stack EAGERLY into a dense spec; heterogeneous specs return a
    :class:`LazyStackedSpec` / :class:`LazyStackedComposite` that keeps
    the components (shape reports -1 at disagreeing dims)."""

    def __new__(cls, *specs, dim: int = 0):
        if len(specs) == 1 and isinstance(specs[0], (list, tuple)):
            specs = tuple(specs[0])
        if _specs_homogeneous(list(specs)):
            return stack_specs(list(specs), dim)
        if isinstance(specs[0], Composite):
            return LazyStackedComposite(*specs, dim=dim)
        return LazyStackedSpec(*specs, dim=dim)


class StackedComposite:
    """Stacked composite spec (reference tensor_specs.py:6463)."""

    def __new__(cls, *specs, dim: int = 0):
        if _specs_homogeneous(list(specs)):
            return stack_specs(list(specs), dim)
        return LazyStackedComposite(*specs, dim=dim)


__all__ += ["Stacked", "StackedComposite", "LazyStackedSpec", "LazyStackedComposite"]
