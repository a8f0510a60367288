"""TensorDict — the batched, nested tensor container that is rl_amd's data model.

The reference framework (pytorch/rl) builds on the external ``tensordict``
package (pyproject.toml:46 ``tensordict>=0.14``).  rl_amd ships its own
implementation, designed for the MI355X execution model:

* leaves live wherever the user puts them (ideally resident in the 288 GB of
  HBM3E) and every structural op (indexing, stacking, ``to``) is a thin loop
  over flat leaves — no graph, no lazy views that hide H2D traffic;
* ``share_memory_`` (POSIX shm via torch) and ``memmap_`` back the
  ParallelEnv / replay-storage IPC paths;
* nested keys are tuples of strings, e.g. ``("next", "observation")``.

Only tensor leaves and nested TensorDicts are stored; non-tensor payloads go
through :class:`NonTensorData`.
"""
from __future__ import annotations

import json
import os
from typing import (
    Any,
    Callable,
    Dict,
    Iterable,
    Iterator,
    List,
    Mapping,
    Optional,
    Sequence,
    Tuple,
    Union,
)

import numpy as np
import torch

NestedKey = Union[str, Tuple[str, ...]]

_NO_DEFAULT = object()


class NonTensorData:
    """Wrapper carrying arbitrary python payloads through a TensorDict."""

    __slots__ = ("data",)

    def __init__(self, data: Any, batch_size=None, device=None):
        if isinstance(data, NonTensorData):
            data = data.data
        self.data = data

    def __repr__(self):
        return f"NonTensorData({self.data!r})"

    def __eq__(self, other):
        if isinstance(other, NonTensorData):
            return self.data == other.data
        return self.data == other

    def __hash__(self):
        try:
            return hash(self.data)
        except TypeError:
            return id(self.data)

    def clone(self, recurse: bool = True):
        return NonTensorData(self.data)

    def to(self, *args, **kwargs):
        return self


def _unravel_key(key: NestedKey) -> Tuple[str, ...]:
    if isinstance(key, str):
        return (key,)
    out: List[str] = []
    for k in key:
        if isinstance(k, str):
            out.append(k)
        else:
            out.extend(_unravel_key(k))
    return tuple(out)


def unravel_key(key: NestedKey) -> NestedKey:
    """Normalize a nested key: plain str stays str, tuples flatten; a
    1-element tuple collapses to str."""
    if isinstance(key, str):
        return key
    flat = _unravel_key(key)
    if len(flat) == 1:
        return flat[0]
    return flat


def _normalize_index(idx, nb: int):
    """Anchor an index to the BATCH dims: expand Ellipsis against nb dims
    so leaf tensors (which have extra feature dims) are indexed only on
    their leading batch dims."""
    if idx is Ellipsis:
        return tuple(slice(None) for _ in range(nb))
    if isinstance(idx, tuple) and any(i is Ellipsis for i in idx):
        pos = idx.index(Ellipsis)
        n_named = len(idx) - 1
        fill = tuple(slice(None) for _ in range(nb - n_named))
        return idx[:pos] + fill + idx[pos + 1 :]
    return idx


def _shape_of_index(batch_size: torch.Size, idx) -> torch.Size:
    """Shape of ``empty(batch_size)[idx]`` without allocating (meta tensor)."""
    if isinstance(idx, int):
        return batch_size[1:]
    if isinstance(idx, slice):
        start, stop, step = idx.indices(batch_size[0])
        n = max(0, (stop - start + (step - 1 if step > 0 else step + 1)) // step)
        return torch.Size([n, *batch_size[1:]])
    if isinstance(idx, torch.Tensor) and idx.dtype == torch.bool:
        # boolean mask over the leading dims: result is [n_true, *rest]
        n = int(idx.sum())
        return torch.Size([n, *batch_size[idx.dim():]])
    if isinstance(idx, tuple) and any(
        isinstance(i, torch.Tensor) and i.dtype == torch.bool for i in idx
    ):
        # rare path: bool mask inside a tuple — use a real (1-byte) dummy
        return torch.zeros(batch_size, dtype=torch.bool, device="cpu")[
            tuple(i.cpu() if isinstance(i, torch.Tensor) else i for i in idx)
        ].shape
    # general path: meta-tensor indexing
    t = torch.empty(batch_size, device="meta")
    if isinstance(idx, tuple):
        idx = tuple(
            i.to("meta") if isinstance(i, torch.Tensor) else i for i in idx
        )
    elif isinstance(idx, torch.Tensor):
        idx = idx.to("meta")
    return t[idx].shape


def _index_for_leaf(idx, leaf_device=None, orig_idx_tensors=None):
    return idx


def is_tensor_collection(obj) -> bool:
    return isinstance(obj, TensorDictBase)


class _TensorDictKeysView:
    def __init__(self, td: "TensorDict", include_nested: bool, leaves_only: bool):
        self._td = td
        self._include_nested = include_nested
        self._leaves_only = leaves_only

    def _iter(self, td, prefix):
        if not hasattr(td, "_data"):
            # lazy-stacked child: delegate to its own keys view
            for sub in td.keys(self._include_nested, self._leaves_only):
                yield prefix + (sub if isinstance(sub, tuple) else (sub,)) if prefix else sub
            return
        for k, v in td._data.items():
            is_td = isinstance(v, TensorDictBase)
            key = prefix + (k,) if prefix else k
            if is_td:
                if not self._leaves_only:
                    yield key
                if self._include_nested:
                    sub_prefix = prefix + (k,) if prefix else (k,)
                    yield from self._iter(v, sub_prefix)
            else:
                yield key

    def __iter__(self):
        return self._iter(self._td, ())

    def __len__(self):
        return sum(1 for _ in self)

    def __contains__(self, key):
        key = unravel_key(key)
        try:
            self._td.get(key)
            return True
        except KeyError:
            return False

    def __repr__(self):
        return f"keys({list(self)})"


class TensorDictBase:
    """Abstract base so ``isinstance`` checks cover future lazy variants."""


class TensorDict(TensorDictBase):
    __slots__ = ("_data", "_batch_size", "_device", "_names")

    def __init__(
        self,
        source: Optional[Mapping] = None,
        batch_size: Union[Sequence[int], torch.Size, int, None] = None,
        device: Union[str, torch.device, None] = None,
        names: Optional[Sequence[str]] = None,
        **kwargs,
    ):
        if source is None:
            source = kwargs if kwargs else {}
        if batch_size is None:
            batch_size = torch.Size([])
        elif isinstance(batch_size, int):
            batch_size = torch.Size([batch_size])
        else:
            batch_size = torch.Size(batch_size)
        self._batch_size = batch_size
        self._device = torch.device(device) if device is not None else None
        self._names = list(names) if names is not None else None
        self._data: Dict[str, Any] = {}
        for k, v in source.items():
            self.set(k, v)

    # ------------------------------------------------------------------ #
    # Construction helpers
    # ------------------------------------------------------------------ #
    @classmethod
    def _new_unsafe(cls, data: dict, batch_size: torch.Size, device=None) -> "TensorDict":
        td = cls.__new__(cls)
        td._data = data
        td._batch_size = batch_size
        td._device = device
        td._names = None
        return td

    @classmethod
    def from_dict(cls, d: Mapping, batch_size=None, device=None) -> "TensorDict":
        out = {}
        for k, v in d.items():
            if isinstance(v, Mapping) and not isinstance(v, TensorDictBase):
                out[k] = cls.from_dict(v, batch_size=batch_size, device=device)
            else:
                out[k] = v
        return cls(out, batch_size=batch_size, device=device)

    @classmethod
    def from_module(cls, module: torch.nn.Module, as_module: bool = False) -> "TensorDict":
        """Extract a module's parameters/buffers into a nested TensorDict
        (reference: ``TensorDict.from_module``)."""
        td = cls({}, batch_size=torch.Size([]))
        for name, param in module.named_parameters(recurse=True):
            td.set(tuple(name.split(".")), param)
        for name, buf in module.named_buffers(recurse=True):
            td.set(tuple(name.split(".")), buf)
        return td

    def to_module(self, module: torch.nn.Module) -> None:
        """Write this TensorDict's leaves into a module's params/buffers
        (in-place ``copy_``)."""
        with torch.no_grad():
            for name, param in module.named_parameters(recurse=True):
                try:
                    val = self.get(tuple(name.split(".")))
                except KeyError:
                    continue
                param.data.copy_(val)
            for name, buf in module.named_buffers(recurse=True):
                try:
                    val = self.get(tuple(name.split(".")))
                except KeyError:
                    continue
                buf.copy_(val)

    # ------------------------------------------------------------------ #
    # Properties
    # ------------------------------------------------------------------ #
    @property
    def batch_size(self) -> torch.Size:
        return self._batch_size

    @batch_size.setter
    def batch_size(self, value):
        self._batch_size = torch.Size(value)

    shape = batch_size

    @property
    def batch_dims(self) -> int:
        return len(self._batch_size)

    def dim(self) -> int:
        return len(self._batch_size)

    @property
    def ndim(self) -> int:
        return len(self._batch_size)

    def ndimension(self) -> int:
        return len(self._batch_size)

    def numel(self) -> int:
        n = 1
        for s in self._batch_size:
            n *= s
        return n

    @property
    def device(self) -> Optional[torch.device]:
        return self._device

    @device.setter
    def device(self, value):
        self._device = torch.device(value) if value is not None else None

    @property
    def requires_grad(self) -> bool:
        return any(
            v.requires_grad for v in self.values(True, True) if isinstance(v, torch.Tensor)
        )

    @property
    def names(self):
        return self._names

    def is_empty(self) -> bool:
        return len(self._data) == 0

    @property
    def sorted_keys(self):
        return sorted(self.keys(True, True), key=str)

    # ------------------------------------------------------------------ #
    # get / set
    # ------------------------------------------------------------------ #
    def _validate_value(self, value, check_shape: bool = True):
        if isinstance(value, TensorDictBase) and not hasattr(value, "_batch_size"):
            # lazy-stacked child: batch_size is computed, not stored
            if check_shape:
                bs = self._batch_size
                if value.batch_size[: len(bs)] != bs:
                    raise RuntimeError(
                        f"nested lazy stack batch_size {value.batch_size} "
                        f"incompatible with parent batch_size {bs}"
                    )
            return value
        if isinstance(value, TensorDictBase):
            if check_shape:
                bs = self._batch_size
                if value._batch_size[: len(bs)] != bs:
                    value = value.clone(False)
                    if value.is_empty():
                        value._batch_size = bs
                    else:
                        raise RuntimeError(
                            f"nested TensorDict batch_size {value._batch_size} "
                            f"incompatible with parent batch_size {bs}"
                        )
            if self._device is not None and value._device != self._device:
                value = value.to(self._device)
            return value
        if isinstance(value, NonTensorData):
            return value
        if not isinstance(value, torch.Tensor):
            if isinstance(value, Mapping):
                return self._validate_value(
                    TensorDict(value, batch_size=self._batch_size, device=self._device),
                    check_shape=check_shape,
                )
            if isinstance(value, (str, bytes)) or value is None:
                return NonTensorData(value)
            if isinstance(value, np.ndarray):
                value = torch.as_tensor(value)
            elif isinstance(value, (bool, int, float, list, np.number)):
                value = torch.as_tensor(value)
            else:
                return NonTensorData(value)
        if check_shape:
            bs = self._batch_size
            if value.shape[: len(bs)] != bs:
                raise RuntimeError(
                    f"tensor shape {tuple(value.shape)} incompatible with "
                    f"batch_size {tuple(bs)}"
                )
        if self._device is not None and value.device != self._device:
            value = value.to(self._device)
        return value

    def set(self, key: NestedKey, value, inplace: bool = False, non_blocking: bool = False) -> "TensorDict":
        key = unravel_key(key)
        if isinstance(key, str):
            if inplace and key in self._data:
                return self.set_(key, value)
            self._data[key] = self._validate_value(value)
            return self
        # nested
        first, rest = key[0], key[1:]
        if first not in self._data or not isinstance(self._data[first], TensorDictBase):
            self._data[first] = TensorDict(
                {}, batch_size=self._batch_size, device=self._device
            )
        self._data[first].set(rest if len(rest) > 1 else rest[0], value, inplace=inplace)
        return self

    def set_(self, key: NestedKey, value, non_blocking: bool = False) -> "TensorDict":
        """In-place copy into an existing entry."""
        key = unravel_key(key)
        dest = self.get(key)
        if isinstance(dest, TensorDictBase):
            dest.update_(value, non_blocking=non_blocking)
        elif isinstance(dest, NonTensorData):
            self.set(key, value)
        else:
            if not isinstance(value, torch.Tensor):
                value = torch.as_tensor(value, device=dest.device, dtype=dest.dtype)
            dest.copy_(value, non_blocking=non_blocking)
        return self

    def set_at_(self, key: NestedKey, value, index) -> "TensorDict":
        dest = self.get(key)
        if isinstance(dest, TensorDictBase):
            dest[index] = value
        else:
            if not isinstance(value, torch.Tensor):
                value = torch.as_tensor(value, device=dest.device, dtype=dest.dtype)
            dest[index] = value
        return self

    def get(self, key: NestedKey, default=_NO_DEFAULT):
        key = unravel_key(key)
        try:
            if isinstance(key, str):
                return self._data[key]
            obj = self
            for i, k in enumerate(key):
                if hasattr(obj, "_data"):
                    obj = obj._data[k]
                else:  # lazy-stacked child: delegate the remaining path
                    return obj.get(key[i:], default)
            return obj
        except (KeyError, AttributeError):
            if default is _NO_DEFAULT:
                raise KeyError(f"key {key!r} not found in TensorDict with keys {list(self.keys())}")
            return default

    def get_at(self, key: NestedKey, index, default=_NO_DEFAULT):
        try:
            val = self.get(key)
        except KeyError:
            if default is _NO_DEFAULT:
                raise
            return default
        return val[index]

    def get_non_tensor(self, key: NestedKey, default=_NO_DEFAULT):
        val = self.get(key, default=default)
        if isinstance(val, NonTensorData):
            return val.data
        return val

    def set_non_tensor(self, key: NestedKey, value):
        return self.set(key, NonTensorData(value))

    def pop(self, key: NestedKey, default=_NO_DEFAULT):
        key = unravel_key(key)
        try:
            val = self.get(key)
        except KeyError:
            if default is _NO_DEFAULT:
                raise
            return default
        self.del_(key)
        return val

    def setdefault(self, key: NestedKey, default):
        try:
            return self.get(key)
        except KeyError:
            self.set(key, default)
            return self.get(key)

    def del_(self, key: NestedKey) -> "TensorDict":
        key = unravel_key(key)
        if isinstance(key, str):
            del self._data[key]
            return self
        parent = self.get(key[:-1]) if len(key) > 2 else self._data[key[0]]
        if len(key) == 2:
            parent = self._data[key[0]]
        del parent._data[key[-1]]
        return self

    def __delitem__(self, key):
        self.del_(key)

    def rename_key_(self, old_key: NestedKey, new_key: NestedKey) -> "TensorDict":
        val = self.get(old_key)
        self.del_(old_key)
        self.set(new_key, val)
        return self

    # ------------------------------------------------------------------ #
    # Mapping protocol / indexing
    # ------------------------------------------------------------------ #
    def __contains__(self, key) -> bool:
        if isinstance(key, (str, tuple)):
            try:
                self.get(key)
                return True
            except KeyError:
                return False
        raise TypeError(f"membership check not supported for {type(key)}")

    def __getitem__(self, index):
        if isinstance(index, str):
            return self.get(index)
        if isinstance(index, tuple) and len(index) and all(
            isinstance(i, str) for i in index
        ):
            return self.get(index)
        return self._index(index)

    def _index(self, index) -> "TensorDict":
        index = _normalize_index(index, len(self._batch_size))
        new_bs = _shape_of_index(self._batch_size, index)
        out = {}
        for k, v in self._data.items():
            if isinstance(v, NonTensorData):
                out[k] = v
            else:
                out[k] = v[index]
        return TensorDict._new_unsafe(out, new_bs, self._device)

    def __setitem__(self, index, value):
        if isinstance(index, str) or (
            isinstance(index, tuple) and len(index) and all(isinstance(i, str) for i in index)
        ):
            self.set(index, value)
            return
        if isinstance(value, TensorDictBase):
            index = _normalize_index(index, len(self._batch_size))
            for k in value.keys(True, True):
                v = value.get(k)
                if isinstance(v, NonTensorData):
                    continue
                try:
                    dest = self.get(k)
                except KeyError:
                    # allocate on first write
                    new_shape = torch.Size(
                        [*self._batch_size, *v.shape[len(value.batch_size):]]
                    )
                    dest = torch.zeros(new_shape, dtype=v.dtype, device=v.device)
                    self.set(k, dest)
                if not isinstance(v, torch.Tensor):
                    v = torch.as_tensor(v, device=dest.device, dtype=dest.dtype)
                dest[index] = v
        else:
            raise TypeError(
                f"cannot assign {type(value)} at a tensor index; expected TensorDict"
            )

    def keys(self, include_nested: bool = False, leaves_only: bool = False, *, is_leaf=None):
        return _TensorDictKeysView(self, include_nested, leaves_only)

    def values(self, include_nested: bool = False, leaves_only: bool = False):
        for k in self.keys(include_nested, leaves_only):
            yield self.get(k)

    def items(self, include_nested: bool = False, leaves_only: bool = False):
        for k in self.keys(include_nested, leaves_only):
            yield k, self.get(k)

    def __len__(self) -> int:
        return self._batch_size[0] if len(self._batch_size) else 0

    def __iter__(self):
        if not len(self._batch_size):
            raise StopIteration("cannot iterate over a 0-dim TensorDict")
        for i in range(self._batch_size[0]):
            yield self._index(i)

    def __bool__(self):
        return True

    # ------------------------------------------------------------------ #
    # Structural ops
    # ------------------------------------------------------------------ #
    def clone(self, recurse: bool = True) -> "TensorDict":
        out = {}
        for k, v in self._data.items():
            if isinstance(v, TensorDictBase):
                out[k] = v.clone(recurse)
            elif isinstance(v, NonTensorData):
                out[k] = v.clone()
            else:
                out[k] = v.clone() if recurse else v
        return TensorDict._new_unsafe(out, self._batch_size, self._device)

    def copy(self) -> "TensorDict":
        return self.clone(False)

    def empty(self) -> "TensorDict":
        return TensorDict._new_unsafe({}, self._batch_size, self._device)

    def to_dict(self) -> dict:
        out = {}
        for k, v in self._data.items():
            if isinstance(v, TensorDictBase):
                out[k] = v.to_dict()
            elif isinstance(v, NonTensorData):
                out[k] = v.data
            else:
                out[k] = v
        return out

    def detach(self) -> "TensorDict":
        return self._fast_apply(lambda t: t.detach())

    def detach_(self) -> "TensorDict":
        for v in self.values(True, True):
            if isinstance(v, torch.Tensor):
                v.detach_()
        return self

    def contiguous(self) -> "TensorDict":
        return self._fast_apply(lambda t: t.contiguous())

    def to(self, *args, non_blocking: bool = False, **kwargs) -> "TensorDict":
        device = None
        dtype = None
        for a in args:
            if isinstance(a, (str, torch.device)):
                device = torch.device(a)
            elif isinstance(a, torch.dtype):
                dtype = a
        device = kwargs.get("device", device)
        dtype = kwargs.get("dtype", dtype)
        if device is not None:
            device = torch.device(device)
        if device is None and dtype is None:
            return self
        if device is not None and dtype is None and self._device == device:
            return self

        def conv(t):
            if dtype is not None and t.is_floating_point():
                return t.to(device=device, dtype=dtype, non_blocking=non_blocking)
            return t.to(device=device, non_blocking=non_blocking)

        out = self._fast_apply(conv)
        out._device = device if device is not None else self._device
        return out

    def cpu(self) -> "TensorDict":
        return self.to("cpu")

    def cuda(self, device: int = 0) -> "TensorDict":
        return self.to(f"cuda:{device}")

    def pin_memory(self) -> "TensorDict":
        return self._fast_apply(
            lambda t: t.pin_memory() if t.device.type == "cpu" else t
        )

    def _fast_apply(self, fn: Callable) -> "TensorDict":
        out = {}
        for k, v in self._data.items():
            if isinstance(v, TensorDictBase):
                out[k] = v._fast_apply(fn)
            elif isinstance(v, NonTensorData):
                out[k] = v
            else:
                out[k] = fn(v)
        return TensorDict._new_unsafe(out, self._batch_size, self._device)

    def apply(
        self,
        fn: Callable,
        *others: "TensorDict",
        batch_size=None,
        device=None,
        inplace: bool = False,
        default=_NO_DEFAULT,
        filter_empty: bool = False,
        **kwargs,
    ) -> Optional["TensorDict"]:
        out = {}
        for k, v in self._data.items():
            ovs = []
            for o in others:
                try:
                    ovs.append(o.get(k))
                except KeyError:
                    if default is _NO_DEFAULT:
                        raise
                    ovs.append(default)
            if isinstance(v, TensorDictBase):
                sub = v.apply(
                    fn,
                    *[
                        ov
                        if isinstance(ov, TensorDictBase)
                        else TensorDict({}, batch_size=v.batch_size)
                        for ov in ovs
                    ],
                    batch_size=batch_size,
                    device=device,
                    default=default,
                )
                if sub is not None:
                    out[k] = sub
            elif isinstance(v, NonTensorData):
                out[k] = v
            else:
                res = fn(v, *ovs)
                if res is not None:
                    out[k] = res
        if inplace:
            self._data = out
            return self
        bs = torch.Size(batch_size) if batch_size is not None else self._batch_size
        dev = device if device is not None else self._device
        return TensorDict._new_unsafe(out, bs, dev)

    def named_apply(self, fn: Callable, nested_keys: bool = True, **kwargs) -> "TensorDict":
        out = self.empty()
        for k, v in self.items(True, True):
            if isinstance(v, NonTensorData):
                out.set(k, v)
                continue
            res = fn(k, v)
            if res is not None:
                out.set(k, res)
        return out

    def update(
        self,
        other: Union[Mapping, "TensorDict"],
        inplace: bool = False,
        clone: bool = False,
        *,
        keys_to_update: Optional[Sequence[NestedKey]] = None,
        non_blocking: bool = False,
    ) -> "TensorDict":
        if other is None:
            return self
        items = (
            other.items() if isinstance(other, TensorDictBase) else other.items()
        )
        for k, v in items:
            if keys_to_update is not None and unravel_key(k) not in [
                unravel_key(kk) for kk in keys_to_update
            ]:
                continue
            if clone and isinstance(v, (torch.Tensor, TensorDictBase)):
                v = v.clone()
            if (
                isinstance(v, (TensorDictBase, Mapping))
                and k in self._data
                and isinstance(self._data[k], TensorDictBase)
            ):
                self._data[k].update(v, inplace=inplace, clone=clone)
            else:
                self.set(k, v, inplace=inplace)
        return self

    def update_(
        self,
        other: Union[Mapping, "TensorDict"],
        *,
        non_blocking: bool = False,
        keys_to_update: Optional[Sequence[NestedKey]] = None,
    ) -> "TensorDict":
        """In-place-copy update of existing keys; missing keys are added."""
        if isinstance(other, Mapping) and not isinstance(other, TensorDictBase):
            other = TensorDict(other, batch_size=self._batch_size)
        keys_norm = (
            None
            if keys_to_update is None
            else {unravel_key(k) for k in keys_to_update}
        )
        for k in other.keys(True, True):
            if keys_norm is not None:
                kn = unravel_key(k)
                root = kn if isinstance(kn, str) else kn[0]
                if kn not in keys_norm and root not in keys_norm:
                    continue
            v = other.get(k)
            if isinstance(v, NonTensorData):
                self.set(k, v)
                continue
            try:
                dest = self.get(k)
            except KeyError:
                self.set(k, v)
                continue
            if isinstance(dest, NonTensorData):
                self.set(k, v)
            else:
                dest.copy_(v, non_blocking=non_blocking)
        return self

    def update_at_(self, other: "TensorDict", index) -> "TensorDict":
        for k in other.keys(True, True):
            v = other.get(k)
            if isinstance(v, NonTensorData):
                continue
            self.get(k)[index] = v
        return self

    def select(self, *keys: NestedKey, inplace: bool = False, strict: bool = True) -> "TensorDict":
        out = TensorDict._new_unsafe({}, self._batch_size, self._device)
        for key in keys:
            try:
                out.set(key, self.get(key))
            except KeyError:
                if strict:
                    raise
        if inplace:
            self._data = out._data
            return self
        return out

    def exclude(self, *keys: NestedKey, inplace: bool = False) -> "TensorDict":
        excluded = {unravel_key(k) for k in keys}
        out = TensorDict._new_unsafe({}, self._batch_size, self._device)
        for k in self.keys(True, True):
            kn = unravel_key(k)
            if kn in excluded:
                continue
            # also exclude whole subtrees
            skip = False
            if isinstance(kn, tuple):
                for i in range(1, len(kn)):
                    prefix = kn[:i]
                    if (prefix if len(prefix) > 1 else prefix[0]) in excluded:
                        skip = True
                        break
            if not skip:
                out.set(k, self.get(k))
        if inplace:
            self._data = out._data
            return self
        return out

    def flatten_keys(self, separator: str = ".") -> "TensorDict":
        out = TensorDict._new_unsafe({}, self._batch_size, self._device)
        for k in self.keys(True, True):
            flat = k if isinstance(k, str) else separator.join(k)
            out._data[flat] = self.get(k)
        return out

    def unflatten_keys(self, separator: str = ".") -> "TensorDict":
        out = TensorDict._new_unsafe({}, self._batch_size, self._device)
        for k, v in self._data.items():
            parts = tuple(k.split(separator))
            out.set(parts if len(parts) > 1 else parts[0], v)
        return out

    # ------------------------------------------------------------------ #
    # Shape ops
    # ------------------------------------------------------------------ #
    def _leaf_shape_op(self, fn: Callable, new_bs: torch.Size) -> "TensorDict":
        out = {}
        for k, v in self._data.items():
            if isinstance(v, TensorDictBase):
                out[k] = v._leaf_shape_op(fn, new_bs)
            elif isinstance(v, NonTensorData):
                out[k] = v
            else:
                out[k] = fn(v)
        return TensorDict._new_unsafe(out, new_bs, self._device)

    def expand(self, *shape) -> "TensorDict":
        if len(shape) == 1 and isinstance(shape[0], (tuple, list, torch.Size)):
            shape = tuple(shape[0])
        nb = len(self._batch_size)
        new_bs = torch.Size(shape)

        def fn(t):
            feat = t.shape[nb:]
            return t.expand(*shape, *feat)

        return self._leaf_shape_op(fn, new_bs)

    def reshape(self, *shape) -> "TensorDict":
        if len(shape) == 1 and isinstance(shape[0], (tuple, list, torch.Size)):
            shape = tuple(shape[0])
        nb = len(self._batch_size)
        new_bs = torch.Size(torch.empty(self._batch_size, device="meta").reshape(shape).shape)

        def fn(t):
            feat = t.shape[nb:]
            return t.reshape(*new_bs, *feat)

        return self._leaf_shape_op(fn, new_bs)

    def view(self, *shape) -> "TensorDict":
        return self.reshape(*shape)

    def flatten(self, start_dim: int = 0, end_dim: int = -1) -> "TensorDict":
        nb = len(self._batch_size)
        if end_dim < 0:
            end_dim = nb + end_dim
        new_bs = torch.Size(
            [
                *self._batch_size[:start_dim],
                int(np.prod(self._batch_size[start_dim : end_dim + 1])),
                *self._batch_size[end_dim + 1 :],
            ]
        )

        def fn(t):
            return t.flatten(start_dim, end_dim) if t.dim() > end_dim else t.flatten(start_dim)

        return self._leaf_shape_op(fn, new_bs)

    def unflatten(self, dim: int, sizes) -> "TensorDict":
        if dim < 0:
            dim = len(self._batch_size) + dim
        new_bs = torch.Size(
            [*self._batch_size[:dim], *sizes, *self._batch_size[dim + 1 :]]
        )
        return self._leaf_shape_op(lambda t: t.unflatten(dim, sizes), new_bs)

    def squeeze(self, dim: Optional[int] = None) -> "TensorDict":
        nb = len(self._batch_size)
        if dim is None:
            for i, s in enumerate(self._batch_size):
                if s == 1:
                    return self.squeeze(i)
            return self
        if dim < 0:
            dim = nb + dim
        if self._batch_size[dim] != 1:
            return self
        new_bs = torch.Size([s for i, s in enumerate(self._batch_size) if i != dim])
        return self._leaf_shape_op(lambda t: t.squeeze(dim), new_bs)

    def unsqueeze(self, dim: int) -> "TensorDict":
        nb = len(self._batch_size)
        if dim < 0:
            dim = nb + dim + 1
        new_bs = torch.Size([*self._batch_size[:dim], 1, *self._batch_size[dim:]])
        return self._leaf_shape_op(lambda t: t.unsqueeze(dim), new_bs)

    def permute(self, *dims) -> "TensorDict":
        if len(dims) == 1 and isinstance(dims[0], (tuple, list)):
            dims = tuple(dims[0])
        nb = len(self._batch_size)
        new_bs = torch.Size([self._batch_size[d] for d in dims])

        def fn(t):
            extra = list(range(nb, t.dim()))
            return t.permute(*dims, *extra)

        return self._leaf_shape_op(fn, new_bs)

    def transpose(self, dim0: int, dim1: int) -> "TensorDict":
        nb = len(self._batch_size)
        dims = list(range(nb))
        dims[dim0], dims[dim1] = dims[dim1], dims[dim0]
        return self.permute(*dims)

    def repeat_interleave(self, repeats: int, dim: int = 0) -> "TensorDict":
        new_bs = list(self._batch_size)
        new_bs[dim] = new_bs[dim] * repeats
        return self._leaf_shape_op(
            lambda t: t.repeat_interleave(repeats, dim=dim), torch.Size(new_bs)
        )

    def split(self, split_size, dim: int = 0) -> List["TensorDict"]:
        if isinstance(split_size, int):
            idx = list(range(0, self._batch_size[dim], split_size))
            sizes = [
                min(split_size, self._batch_size[dim] - i) for i in idx
            ]
        else:
            sizes = list(split_size)
        outs = []
        start = 0
        for s in sizes:
            sl = [slice(None)] * dim + [slice(start, start + s)]
            outs.append(self[tuple(sl)])
            start += s
        return outs

    def chunk(self, chunks: int, dim: int = 0) -> List["TensorDict"]:
        size = self._batch_size[dim]
        split = -(-size // chunks)
        return self.split(split, dim=dim)

    def gather(self, dim: int, index: torch.Tensor) -> "TensorDict":
        nb = len(self._batch_size)
        new_bs = torch.Size(index.shape[:nb])

        def fn(t):
            idx = index
            while idx.dim() < t.dim():
                idx = idx.unsqueeze(-1)
            idx = idx.expand(*index.shape, *t.shape[nb:])
            return t.gather(dim, idx)

        return self._leaf_shape_op(fn, new_bs)

    def roll(self, shifts: int, dims: int = 0) -> "TensorDict":
        return self._leaf_shape_op(
            lambda t: t.roll(shifts, dims=dims), self._batch_size
        )

    # ------------------------------------------------------------------ #
    # Memory
    # ------------------------------------------------------------------ #
    def share_memory_(self) -> "TensorDict":
        for v in self.values(True, True):
            if isinstance(v, torch.Tensor):
                v.share_memory_()
        return self

    def is_shared(self) -> bool:
        return all(
            v.is_shared()
            for v in self.values(True, True)
            if isinstance(v, torch.Tensor)
        ) and any(True for _ in self.values(True, True))

    def memmap_(self, prefix: Optional[str] = None, copy_existing: bool = False) -> "TensorDict":
        """Move all leaves to disk-backed memory-mapped tensors under
        ``prefix`` (a directory).  Layout: one ``.memmap`` file per leaf,
        plus ``meta.json`` (reference: tensordict memmap format)."""
        if prefix is None:
            import tempfile

            prefix = tempfile.mkdtemp(prefix="rl_amd_memmap_")
        os.makedirs(prefix, exist_ok=True)
        meta = {"batch_size": list(self._batch_size), "leaves": {}}
        for k in list(self.keys(True, True)):
            v = self.get(k)
            if isinstance(v, NonTensorData):
                continue
            flat = k if isinstance(k, str) else ".".join(k)
            path = os.path.join(prefix, flat + ".memmap")
            mm = np.memmap(
                path,
                dtype=_torch_to_np_dtype(v.dtype),
                mode="w+",
                shape=tuple(v.shape) if v.numel() else (0,),
            )
            t = torch.from_numpy(mm)
            if v.dtype in (torch.bfloat16, torch.float16):
                t = t.view(v.dtype)
            t = t.reshape(v.shape)
            t.copy_(v.detach().cpu())
            self.set(k, t)
            meta["leaves"][flat] = {
                "shape": list(v.shape),
                "dtype": str(v.dtype),
            }
        with open(os.path.join(prefix, "meta.json"), "w") as f:
            json.dump(meta, f)
        self._memmap_prefix = prefix
        return self

    @classmethod
    def load_memmap(cls, prefix: str) -> "TensorDict":
        with open(os.path.join(prefix, "meta.json")) as f:
            meta = json.load(f)
        td = cls({}, batch_size=meta["batch_size"])
        for flat, info in meta["leaves"].items():
            dtype = _np_view_dtype(info["dtype"])
            shape = tuple(info["shape"])
            path = os.path.join(prefix, flat + ".memmap")
            mm = np.memmap(path, dtype=dtype, mode="r+", shape=shape if np.prod(shape) else (0,))
            t = torch.from_numpy(mm)
            torch_dtype = _str_to_torch_dtype(info["dtype"])
            if torch_dtype in (torch.bfloat16, torch.float16):
                t = t.view(torch_dtype)
            t = t.reshape(shape)
            key = tuple(flat.split("."))
            td.set(key if len(key) > 1 else key[0], t)
        return td

    def consolidate(self) -> "TensorDict":
        """Pack all leaves into one flat storage (fast pickling / IPC).

        On MI355X the payoff is on the CPU/pipe path — a single contiguous
        buffer crosses a mp.Pipe in one write."""
        leaves = [
            (k, v)
            for k, v in self.items(True, True)
            if isinstance(v, torch.Tensor)
        ]
        if not leaves:
            return self
        total = sum(v.numel() * v.element_size() for _, v in leaves)
        buf = torch.empty(total, dtype=torch.uint8, device=leaves[0][1].device)
        offset = 0
        out = self.empty()
        for k, v in leaves:
            n = v.numel() * v.element_size()
            chunk = buf[offset : offset + n].view(v.dtype).view(v.shape)
            chunk.copy_(v.reshape(-1).view(v.dtype).reshape(v.shape))
            out.set(k, chunk)
            offset += n
        out._consolidated_buffer = buf
        return out

    def zero_(self) -> "TensorDict":
        for v in self.values(True, True):
            if isinstance(v, torch.Tensor):
                v.zero_()
        return self

    def fill_(self, key: NestedKey, value) -> "TensorDict":
        self.get(key).fill_(value)
        return self

    # ------------------------------------------------------------------ #
    # Arithmetic-ish helpers used by losses
    # ------------------------------------------------------------------ #
    def abs(self) -> "TensorDict":
        return self._fast_apply(torch.abs)

    def mean(self, dim=None):
        if dim is None:
            vals = [
                v.float().mean()
                for v in self.values(True, True)
                if isinstance(v, torch.Tensor)
            ]
            return torch.stack(vals).mean() if vals else torch.tensor(float("nan"))
        return self._fast_apply(lambda t: t.mean(dim))

    def sum(self, dim=None):
        if dim is None:
            vals = [
                v.sum() for v in self.values(True, True) if isinstance(v, torch.Tensor)
            ]
            return torch.stack(vals).sum() if vals else torch.tensor(0.0)
        return self._fast_apply(lambda t: t.sum(dim))

    def masked_fill_(self, mask: torch.Tensor, value) -> "TensorDict":
        nb = len(self._batch_size)
        for v in self.values(True, True):
            if isinstance(v, torch.Tensor):
                m = mask
                while m.dim() < v.dim():
                    m = m.unsqueeze(-1)
                v.masked_fill_(m.expand_as(v), value)
        return self

    def where(self, condition: torch.Tensor, other: "TensorDict") -> "TensorDict":
        nb = len(self._batch_size)

        def fn(t, o):
            c = condition
            while c.dim() < t.dim():
                c = c.unsqueeze(-1)
            return torch.where(c.expand_as(t), t, o)

        return self.apply(fn, other)

    def isnan(self) -> "TensorDict":
        return self._fast_apply(torch.isnan)

    def bool(self) -> "TensorDict":
        return self._fast_apply(lambda t: t.bool())

    def float(self) -> "TensorDict":
        return self._fast_apply(lambda t: t.float() if t.is_floating_point() else t)

    def any(self) -> bool:
        for v in self.values(True, True):
            if isinstance(v, torch.Tensor) and v.any():
                return True
        return False

    def all(self) -> bool:
        for v in self.values(True, True):
            if isinstance(v, torch.Tensor) and not v.all():
                return False
        return True

    def requires_grad_(self, mode: bool = True) -> "TensorDict":
        for v in self.values(True, True):
            if isinstance(v, torch.Tensor) and v.is_floating_point():
                v.requires_grad_(mode)
        return self

    # ------------------------------------------------------------------ #
    # Repr / comparison
    # ------------------------------------------------------------------ #
    def __repr__(self):
        def fmt(td, indent):
            pad = " " * indent
            lines = []
            for k, v in sorted(td._data.items()):
                if isinstance(v, TensorDictBase):
                    lines.append(f"{pad}{k}: TensorDict(")
                    lines.append(fmt(v, indent + 4))
                    lines.append(f"{pad}    batch_size={tuple(v.batch_size)})")
                elif isinstance(v, NonTensorData):
                    lines.append(f"{pad}{k}: {v!r}")
                else:
                    lines.append(
                        f"{pad}{k}: Tensor(shape={tuple(v.shape)}, dtype={v.dtype}, "
                        f"device={v.device})"
                    )
            return "\n".join(lines)

        body = fmt(self, 4)
        return (
            f"TensorDict(\n{body}\n    batch_size={tuple(self._batch_size)}, "
            f"device={self._device})"
        )

    def __eq__(self, other):
        if isinstance(other, TensorDictBase):
            out = self.empty()
            for k in self.keys(True, True):
                a, b = self.get(k), other.get(k)
                if isinstance(a, NonTensorData):
                    continue
                out.set(k, a == b)
            return out
        if isinstance(other, (int, float, bool, torch.Tensor)):
            return self._fast_apply(lambda t: t == other)
        return NotImplemented

    def __ne__(self, other):
        eq = self.__eq__(other)
        if eq is NotImplemented:
            return eq
        return eq._fast_apply(lambda t: ~t)

    # state for pickling
    def __getstate__(self):
        return {
            "_data": self._data,
            "_batch_size": tuple(self._batch_size),
            "_device": str(self._device) if self._device is not None else None,
        }

    def __setstate__(self, state):
        self._data = state["_data"]
        self._batch_size = torch.Size(state["_batch_size"])
        self._device = (
            torch.device(state["_device"]) if state["_device"] is not None else None
        )
        self._names = None

    # save / load
    def save(self, path: str) -> None:
        torch.save(self, path)

    @classmethod
    def load(cls, path: str) -> "TensorDict":
        return torch.load(path, weights_only=False)

    # classmethod constructors over collections
    @classmethod
    def stack(cls, tds: Sequence["TensorDict"], dim: int = 0) -> "TensorDict":
        return stack(tds, dim)

    @classmethod
    def cat(cls, tds: Sequence["TensorDict"], dim: int = 0) -> "TensorDict":
        return cat(tds, dim)

    @classmethod
    def zeros(cls, *shape, device=None) -> "TensorDict":
        if len(shape) == 1 and isinstance(shape[0], (tuple, list, torch.Size)):
            shape = tuple(shape[0])
        return cls({}, batch_size=shape, device=device)

    @classmethod
    def fromkeys(cls, keys, value=0.0, batch_size=(), device=None):
        td = cls({}, batch_size=batch_size, device=device)
        for k in keys:
            td.set(k, torch.full(tuple(batch_size), value))
        return td


# --------------------------------------------------------------------------- #
# Module-level functional API
# --------------------------------------------------------------------------- #
def _common_keys(tds: Sequence[TensorDict]) -> List[NestedKey]:
    keys = list(tds[0].keys(True, True))
    keyset = set(keys)
    for td in tds[1:]:
        keyset &= set(td.keys(True, True))
    return [k for k in keys if k in keyset]


def stack(tds: Sequence[TensorDict], dim: int = 0) -> TensorDict:
    """Stack TensorDicts along a new batch dim.

    Homogeneous inputs stack EAGERLY (dense layouts feed kernels; HBM3E
    is big).  Heterogeneous inputs — per-element shape or key-set
    mismatches — return a :class:`~rl_amd.tensordict.LazyStackedTensorDict`
    instead (the reference's lazy-stack behavior)."""
    tds = list(tds)
    if not tds:
        raise ValueError("cannot stack an empty sequence of TensorDicts")
    if _is_heterogeneous(tds):
        from .lazy import LazyStackedTensorDict

        return LazyStackedTensorDict(*tds, stack_dim=dim)
    bs = tds[0].batch_size
    if dim < 0:
        dim = len(bs) + 1 + dim
    new_bs = torch.Size([*bs[:dim], len(tds), *bs[dim:]])
    out = TensorDict._new_unsafe({}, new_bs, tds[0].device)
    for k in _common_keys(tds):
        vals = [td.get(k) for td in tds]
        if isinstance(vals[0], NonTensorData):
            out.set(k, _merge_non_tensor(vals))
        elif isinstance(vals[0], torch.Tensor):
            out.set(k, torch.stack(vals, dim=dim))
    return out


def _merge_non_tensor(vals: Sequence[NonTensorData]) -> NonTensorData:
    """Combine non-tensor leaves across stacked/concatenated tds:
    identical values collapse to one; dicts of batch-lists (the History
    convention: {"roles": [[...]], "contents": [[...]]}) concatenate
    their outer lists; anything else becomes a list of values."""
    datas = [v.data if isinstance(v, NonTensorData) else v for v in vals]
    try:
        if all(d == datas[0] for d in datas[1:]):
            return NonTensorData(datas[0])
    except Exception:
        pass
    if all(isinstance(d, dict) for d in datas) and all(
        set(d.keys()) == set(datas[0].keys()) for d in datas[1:]
    ) and all(
        isinstance(v, list) for d in datas for v in d.values()
    ):
        merged = {
            key: [item for d in datas for item in d[key]] for key in datas[0]
        }
        return NonTensorData(merged)
    return NonTensorData(datas)


def _is_heterogeneous(tds: Sequence[TensorDict]) -> bool:
    """True when elements disagree on batch size, leaf key sets or leaf
    shapes (then only a lazy stack can represent them)."""
    first = tds[0]
    bs = first.batch_size
    keys0 = set(first.keys(True, True))
    shapes0 = None
    for td in tds[1:]:
        if td.batch_size != bs:
            return True
        if set(td.keys(True, True)) != keys0:
            return True
        if shapes0 is None:
            try:
                shapes0 = {k: tuple(first.get(k).shape)
                           for k in keys0
                           if isinstance(first.get(k), torch.Tensor)}
            except RuntimeError:
                # a component holds a heterogeneous lazy stack itself
                return True
        for k, shp in shapes0.items():
            v = td.get(k, None)
            if isinstance(v, torch.Tensor) and tuple(v.shape) != shp:
                return True
    return False


def lazy_stack(tds: Sequence[TensorDict], dim: int = 0):
    """Always-lazy stack (reference LazyStackedTensorDict constructor)."""
    from .lazy import LazyStackedTensorDict

    return LazyStackedTensorDict(*list(tds), stack_dim=dim)


def cat(tds: Sequence[TensorDict], dim: int = 0) -> TensorDict:
    tds = list(tds)
    if not tds:
        raise ValueError("cannot cat an empty sequence of TensorDicts")
    bs = tds[0].batch_size
    if dim < 0:
        dim = len(bs) + dim
    new_bs = torch.Size(
        [
            *bs[:dim],
            sum(td.batch_size[dim] for td in tds),
            *bs[dim + 1 :],
        ]
    )
    out = TensorDict._new_unsafe({}, new_bs, tds[0].device)
    for k in _common_keys(tds):
        vals = [td.get(k) for td in tds]
        if isinstance(vals[0], NonTensorData):
            out.set(k, _merge_non_tensor(vals))
        elif isinstance(vals[0], torch.Tensor):
            out.set(k, torch.cat(vals, dim=dim))
    return out


def pad(td: TensorDict, pad_size: Sequence[int], value: float = 0.0) -> TensorDict:
    """Pad batch dims: ``pad_size`` is (before_0, after_0, before_1, ...)."""
    nb = len(td.batch_size)
    new_bs = list(td.batch_size)
    for i in range(len(pad_size) // 2):
        new_bs[i] += pad_size[2 * i] + pad_size[2 * i + 1]

    def fn(t):
        n_feat = t.dim() - nb
        pads = []
        for i in reversed(range(len(pad_size) // 2)):
            pads.extend([pad_size[2 * i], pad_size[2 * i + 1]])
        pads = [0, 0] * n_feat + pads
        return torch.nn.functional.pad(t, pads, value=value)

    return td._leaf_shape_op(fn, torch.Size(new_bs))


def where(cond: torch.Tensor, a: TensorDict, b: TensorDict) -> TensorDict:
    return a.where(cond, b)


def _torch_to_np_dtype(dtype: torch.dtype):
    if dtype == torch.bfloat16:
        return np.uint16
    if dtype == torch.float16:
        return np.uint16
    if dtype == torch.bool:
        return np.bool_
    return torch.empty(0, dtype=dtype).numpy().dtype


def _np_view_dtype(dtype_str: str):
    mapping = {
        "torch.bfloat16": np.uint16,
        "torch.float16": np.uint16,
        "torch.float32": np.float32,
        "torch.float64": np.float64,
        "torch.int64": np.int64,
        "torch.int32": np.int32,
        "torch.int16": np.int16,
        "torch.int8": np.int8,
        "torch.uint8": np.uint8,
        "torch.bool": np.bool_,
    }
    return mapping[dtype_str]


def _str_to_torch_dtype(dtype_str: str) -> torch.dtype:
    return getattr(torch, dtype_str.split(".")[-1])


class TensorClass:
    """Minimal dataclass-over-TensorDict base (reference: tensorclass).

    Subclasses declare annotated fields; instances carry a TensorDict."""

    def __init__(self, batch_size=(), device=None, **fields):
        object.__setattr__(
            self, "_td", TensorDict({}, batch_size=batch_size, device=device)
        )
        for k, v in fields.items():
            setattr(self, k, v)

    def __getattr__(self, name):
        td = object.__getattribute__(self, "_td")
        try:
            val = td.get(name)
        except KeyError:
            raise AttributeError(name)
        if isinstance(val, NonTensorData):
            return val.data
        return val

    def __setattr__(self, name, value):
        if name.startswith("_"):
            object.__setattr__(self, name, value)
        else:
            self._td.set(name, value)

    @property
    def batch_size(self):
        return self._td.batch_size

    @property
    def device(self):
        return self._td.device

    def to_tensordict(self) -> TensorDict:
        return self._td

    def clone(self):
        out = type(self).__new__(type(self))
        object.__setattr__(out, "_td", self._td.clone())
        return out
