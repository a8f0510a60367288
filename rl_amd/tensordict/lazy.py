"""LazyStackedTensorDict — heterogeneous stacks without densification.

Reference capability: the external ``tensordict`` package's
LazyStackedTensorDict (used by pytorch/rl for heterogeneous multi-agent
data, e.g. torchrl/testing/mocking_classes.py:1787
HeterogeneousCountingEnv).  rl_amd stacks homogeneous data eagerly
(HBM3E is big, dense layouts feed kernels); THIS class exists for data
that cannot densify — per-element tensors with different trailing
shapes or different key sets.  Component tensordicts are kept as-is;
keys common to all components with matching shapes read as dense
stacks, mismatched keys read as nested tensors or per-component access.
"""
from __future__ import annotations

from typing import Any, Callable, List, Optional, Sequence, Tuple

import torch

from .tensordict import (
    NestedKey,
    NonTensorData,
    TensorDict,
    TensorDictBase,
    _NO_DEFAULT,
    unravel_key,
)

__all__ = ["LazyStackedTensorDict"]


class LazyStackedTensorDict(TensorDictBase):
    """A stack of TensorDicts materialized per-key on access.

    Args:
        *tds: component tensordicts.  They must share ``batch_size``
            (the stack dim is inserted into it) but may disagree on key
            sets and on per-key trailing shapes.
        stack_dim: position of the stacked dimension (default 0).
    """

    def __init__(self, *tds: TensorDictBase, stack_dim: int = 0):
        if len(tds) == 1 and isinstance(tds[0], (list, tuple)):
            tds = tuple(tds[0])
        if not tds:
            raise ValueError("LazyStackedTensorDict needs at least one element")
        bs = tds[0].batch_size
        for td in tds[1:]:
            if td.batch_size != bs:
                raise ValueError(
                    f"stacked components must share batch_size; got {bs} vs "
                    f"{td.batch_size}"
                )
        if stack_dim < 0:
            stack_dim = len(bs) + 1 + stack_dim
        if not 0 <= stack_dim <= len(bs):
            raise ValueError(f"invalid stack_dim {stack_dim} for batch {bs}")
        self.tensordicts: List[TensorDictBase] = list(tds)
        self.stack_dim = stack_dim

    # -- shape ---------------------------------------------------------- #
    @property
    def batch_size(self) -> torch.Size:
        bs = self.tensordicts[0].batch_size
        return torch.Size(
            [*bs[: self.stack_dim], len(self.tensordicts), *bs[self.stack_dim :]]
        )

    @property
    def batch_dims(self) -> int:
        return len(self.batch_size)

    @property
    def shape(self) -> torch.Size:
        return self.batch_size

    def dim(self) -> int:
        return self.batch_dims

    @property
    def ndim(self) -> int:
        return self.batch_dims

    def __len__(self) -> int:
        return self.batch_size[0] if self.batch_size else 0

    @property
    def device(self):
        return self.tensordicts[0].device

    # -- keys ----------------------------------------------------------- #
    def keys(self, include_nested: bool = False, leaves_only: bool = False, **kw):
        common = None
        for td in self.tensordicts:
            ks = set(td.keys(include_nested, leaves_only))
            common = ks if common is None else (common & ks)
        order = [
            k
            for k in self.tensordicts[0].keys(include_nested, leaves_only)
            if k in common
        ]
        return order

    def __contains__(self, key) -> bool:
        return all(key in td for td in self.tensordicts)

    def items(self, include_nested: bool = False, leaves_only: bool = False):
        for k in self.keys(include_nested, leaves_only):
            yield k, self.get(k)

    def values(self, include_nested: bool = False, leaves_only: bool = False):
        for k in self.keys(include_nested, leaves_only):
            yield self.get(k)

    # -- access --------------------------------------------------------- #
    def get(self, key: NestedKey, default=_NO_DEFAULT):
        vals = []
        for td in self.tensordicts:
            v = td.get(key, None)
            if v is None:
                if default is _NO_DEFAULT:
                    raise KeyError(key)
                return default
            vals.append(v)
        if isinstance(vals[0], TensorDictBase):
            return LazyStackedTensorDict(*vals, stack_dim=self.stack_dim)
        if isinstance(vals[0], NonTensorData):
            return vals[0]
        shapes = {tuple(v.shape) for v in vals}
        if len(shapes) == 1:
            return torch.stack(vals, dim=self.stack_dim)
        raise RuntimeError(
            f"key {key!r} has heterogeneous shapes {sorted(shapes)}; use "
            "get_nestedtensor(key) or index a single element"
        )

    def get_nestedtensor(self, key: NestedKey):
        """Nested-tensor view of a (possibly heterogeneous) key."""
        return torch.nested.nested_tensor(
            [td.get(key) for td in self.tensordicts]
        )

    def set(self, key: NestedKey, value, **kwargs) -> "LazyStackedTensorDict":
        n = len(self.tensordicts)
        if isinstance(value, (list, tuple)):
            assert len(value) == n
            parts = value
        elif isinstance(value, LazyStackedTensorDict):
            parts = value.tensordicts
        elif isinstance(value, (torch.Tensor, TensorDictBase)):
            parts = value.unbind(self.stack_dim)
        else:
            for td in self.tensordicts:
                td.set(key, value)
            return self
        for td, part in zip(self.tensordicts, parts):
            td.set(key, part)
        return self

    def set_(self, key, value, **kw):
        parts = (
            value.unbind(self.stack_dim)
            if isinstance(value, torch.Tensor)
            else value
        )
        for td, part in zip(self.tensordicts, parts):
            td.set_(key, part)
        return self

    def __getitem__(self, index):
        if isinstance(index, str) or (
            isinstance(index, tuple) and index and isinstance(index[0], str)
        ):
            return self.get(index)
        if self.stack_dim == 0:
            if isinstance(index, int):
                return self.tensordicts[index]
            if isinstance(index, slice):
                return LazyStackedTensorDict(
                    *self.tensordicts[index], stack_dim=0
                )
            if isinstance(index, (list, torch.Tensor)):
                idx = (
                    index.tolist() if isinstance(index, torch.Tensor) else index
                )
                return LazyStackedTensorDict(
                    *[self.tensordicts[int(i)] for i in idx], stack_dim=0
                )
            if isinstance(index, tuple) and index:
                head, rest = index[0], index[1:]
                sub = self[head]
                return sub[rest] if rest else sub
        return self.to_tensordict()[index]

    def __setitem__(self, index, value):
        if isinstance(index, (str, tuple)) and (
            isinstance(index, str) or isinstance(index[0], str)
        ):
            self.set(index, value)
            return
        if self.stack_dim == 0 and isinstance(index, int):
            self.tensordicts[index] = value
            return
        raise NotImplementedError("complex indexing assignment on lazy stacks")

    def unbind(self, dim: int = 0):
        if dim == self.stack_dim:
            return tuple(self.tensordicts)
        return self.to_tensordict().unbind(dim)

    # -- transforms ----------------------------------------------------- #
    def clone(self, recurse: bool = True) -> "LazyStackedTensorDict":
        return LazyStackedTensorDict(
            *[td.clone(recurse) for td in self.tensordicts],
            stack_dim=self.stack_dim,
        )

    def to(self, *args, **kwargs) -> "LazyStackedTensorDict":
        return LazyStackedTensorDict(
            *[td.to(*args, **kwargs) for td in self.tensordicts],
            stack_dim=self.stack_dim,
        )

    def detach(self) -> "LazyStackedTensorDict":
        return LazyStackedTensorDict(
            *[td.detach() for td in self.tensordicts], stack_dim=self.stack_dim
        )

    def cpu(self):
        return self.to("cpu")

    def select(self, *keys, strict: bool = True):
        return LazyStackedTensorDict(
            *[td.select(*keys, strict=strict) for td in self.tensordicts],
            stack_dim=self.stack_dim,
        )

    def exclude(self, *keys):
        return LazyStackedTensorDict(
            *[td.exclude(*keys) for td in self.tensordicts],
            stack_dim=self.stack_dim,
        )

    def apply(self, fn: Callable, *others, **kwargs):
        return LazyStackedTensorDict(
            *[td.apply(fn) for td in self.tensordicts], stack_dim=self.stack_dim
        )

    def update(self, other, **kwargs):
        if isinstance(other, LazyStackedTensorDict):
            for td, o in zip(self.tensordicts, other.tensordicts):
                td.update(o, **kwargs)
            return self
        if isinstance(other, TensorDictBase):
            for i, td in enumerate(self.tensordicts):
                td.update(other[i] if self.stack_dim == 0 else other, **kwargs)
            return self
        for k, v in other.items():
            self.set(k, v)
        return self

    def to_tensordict(self) -> TensorDict:
        """Densify (raises if any key is heterogeneous)."""
        out = TensorDict({}, batch_size=self.batch_size, device=self.device)
        for k in self.keys(True, True):
            out.set(k, self.get(k))
        return out

    def contiguous(self) -> TensorDict:
        return self.to_tensordict()

    def empty(self) -> "LazyStackedTensorDict":
        return LazyStackedTensorDict(
            *[td.empty() for td in self.tensordicts], stack_dim=self.stack_dim
        )

    @property
    def is_heterogeneous(self) -> bool:
        for k in self.keys(True, True):
            try:
                self.get(k)
            except RuntimeError:
                return True
        return False

    def __repr__(self):
        return (
            f"LazyStackedTensorDict(n={len(self.tensordicts)}, "
            f"stack_dim={self.stack_dim}, batch_size={tuple(self.batch_size)})"
        )
