"""TensorDictModule & friends — the nn-module wrappers over TensorDict.

Re-implements (MI355X-first, no external tensordict dep) the API of
``tensordict.nn``: ``TensorDictModule``, ``TensorDictSequential``,
``ProbabilisticTensorDictModule``, ``set_interaction_type``.
Reference behavior: pytorch/rl uses these as the universal policy/model
wrappers (torchrl/modules/tensordict_module/*).
"""
from __future__ import annotations

import contextlib
import enum
import threading
from typing import Any, Callable, Dict, Iterable, List, Optional, Sequence, Tuple, Union

import torch
from torch import nn

from .tensordict import NestedKey, TensorDict, TensorDictBase, unravel_key

__all__ = [
    "TensorDictModuleBase",
    "TensorDictModule",
    "TensorDictSequential",
    "ProbabilisticTensorDictModule",
    "ProbabilisticTensorDictSequential",
    "InteractionType",
    "set_interaction_type",
    "interaction_type",
    "make_functional",
    "TensorDictParams",
    "WrapModule",
]


class InteractionType(enum.Enum):
    MODE = "mode"
    MEAN = "mean"
    MEDIAN = "median"
    RANDOM = "random"
    DETERMINISTIC = "deterministic"


_INTERACTION = threading.local()


def interaction_type() -> Optional[InteractionType]:
    return getattr(_INTERACTION, "type", None)


@contextlib.contextmanager
def set_interaction_type(type: Optional[InteractionType] = InteractionType.DETERMINISTIC):
    prev = getattr(_INTERACTION, "type", None)
    _INTERACTION.type = type
    try:
        yield
    finally:
        _INTERACTION.type = prev


def _norm_keys(keys) -> List[NestedKey]:
    if keys is None:
        return []
    if isinstance(keys, (str, tuple)) and (
        isinstance(keys, str) or all(isinstance(k, str) for k in keys)
    ):
        if isinstance(keys, tuple):
            # ambiguous: tuple of strings could be a nested key or a list
            return [unravel_key(keys)] if any("." not in k for k in keys) and False else [unravel_key(k) for k in keys]
        return [keys]
    return [unravel_key(k) for k in keys]


class TensorDictModuleBase(nn.Module):
    """Base class: a Module that reads ``in_keys`` from and writes
    ``out_keys`` to a TensorDict."""

    in_keys: List[NestedKey]
    out_keys: List[NestedKey]

    def reset_parameters_recursive(self):
        for m in self.modules():
            if hasattr(m, "reset_parameters") and m is not self:
                m.reset_parameters()


class TensorDictModule(TensorDictModuleBase):
    """Wrap a callable/Module: gathers ``in_keys`` as positional args, runs
    the module, scatters result(s) into ``out_keys``.

    ``in_keys`` may be a dict mapping td-keys to kwarg names.
    """

    def __init__(
        self,
        module: Union[Callable, nn.Module],
        in_keys: Union[NestedKey, Sequence[NestedKey], Dict[NestedKey, str]],
        out_keys: Union[NestedKey, Sequence[NestedKey]],
        inplace: bool = True,
    ):
        super().__init__()
        if isinstance(in_keys, dict):
            self._kwargs_map = {unravel_key(k): v for k, v in in_keys.items()}
            self.in_keys = list(self._kwargs_map.keys())
        else:
            self._kwargs_map = None
            if isinstance(in_keys, str):
                in_keys = [in_keys]
            self.in_keys = [unravel_key(k) for k in in_keys]
        if isinstance(out_keys, str):
            out_keys = [out_keys]
        self.out_keys = [unravel_key(k) for k in out_keys]
        if isinstance(module, nn.Module):
            self.module = module
        else:
            self.module = None
            self._fn = module
        self._inplace = inplace

    @property
    def _callable(self):
        return self.module if self.module is not None else self._fn

    def forward(self, tensordict: TensorDictBase = None, *args, **kwargs) -> TensorDictBase:
        if tensordict is None and args:
            tensordict, args = args[0], args[1:]
        if not isinstance(tensordict, TensorDictBase):
            # dispatch mode: raw tensors in the in_keys order
            tensors = (tensordict, *args)
            td = TensorDict({}, batch_size=tensors[0].shape[:1])
            for k, t in zip(self.in_keys, tensors):
                td.set(k, t)
            out_td = self.forward(td)
            outs = tuple(out_td.get(k) for k in self.out_keys)
            return outs if len(outs) > 1 else outs[0]
        if self._kwargs_map is not None:
            fkwargs = {
                name: tensordict.get(k) for k, name in self._kwargs_map.items()
            }
            out = self._callable(**fkwargs)
        else:
            inputs = tuple(tensordict.get(k, None) for k in self.in_keys)
            out = self._callable(*inputs)
        if isinstance(out, TensorDictBase):
            tensordict.update(out)
            return tensordict
        if not isinstance(out, tuple):
            out = (out,)
        for k, v in zip(self.out_keys, out):
            if k != "_" and v is not None:
                tensordict.set(k, v)
        return tensordict

    def __repr__(self):
        return (
            f"{type(self).__name__}(module={self._callable}, "
            f"in_keys={self.in_keys}, out_keys={self.out_keys})"
        )


class WrapModule(TensorDictModuleBase):
    """Wrap a TensorDict→TensorDict callable with declared keys."""

    def __init__(self, fn: Callable, in_keys=(), out_keys=()):
        super().__init__()
        self.fn = fn
        self.in_keys = _norm_keys(list(in_keys))
        self.out_keys = _norm_keys(list(out_keys))

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        return self.fn(td)


class TensorDictSequential(TensorDictModuleBase):
    """Chain of TensorDictModules; ``partial_tolerant`` skips modules whose
    inputs are missing (reference: SafeSequential)."""

    def __init__(self, *modules: TensorDictModuleBase, partial_tolerant: bool = False):
        super().__init__()
        if len(modules) == 1 and isinstance(modules[0], (list, tuple)):
            modules = tuple(modules[0])
        self.module = nn.ModuleList(modules)
        self.partial_tolerant = partial_tolerant
        in_keys: List[NestedKey] = []
        produced: set = set()
        out_keys: List[NestedKey] = []
        for m in modules:
            for k in getattr(m, "in_keys", []):
                if k not in produced and k not in in_keys:
                    in_keys.append(k)
            for k in getattr(m, "out_keys", []):
                produced.add(k)
                if k not in out_keys:
                    out_keys.append(k)
        self.in_keys = in_keys
        self.out_keys = out_keys

    def __iter__(self):
        return iter(self.module)

    def __len__(self):
        return len(self.module)

    def __getitem__(self, idx):
        if isinstance(idx, slice):
            return TensorDictSequential(*list(self.module)[idx])
        return self.module[idx]

    def forward(self, tensordict: TensorDictBase = None, *args, **kwargs) -> TensorDictBase:
        if tensordict is not None and not isinstance(tensordict, TensorDictBase):
            tensors = (tensordict, *args)
            td = TensorDict({}, batch_size=tensors[0].shape[:1])
            for k, t in zip(self.in_keys, tensors):
                td.set(k, t)
            out_td = self.forward(td)
            outs = tuple(out_td.get(k) for k in self.out_keys)
            return outs if len(outs) > 1 else outs[0]
        for m in self.module:
            if self.partial_tolerant:
                missing = [
                    k for k in getattr(m, "in_keys", []) if k not in tensordict
                ]
                if missing:
                    continue
            tensordict = m(tensordict)
        return tensordict

    def select_subsequence(self, in_keys=None, out_keys=None) -> "TensorDictSequential":
        mods = list(self.module)
        if out_keys is not None:
            out_keys = set(_norm_keys(out_keys))
            keep: List[TensorDictModuleBase] = []
            needed = set(out_keys)
            for m in reversed(mods):
                if any(k in needed for k in m.out_keys):
                    keep.insert(0, m)
                    needed |= set(m.in_keys)
            mods = keep
        return TensorDictSequential(*mods)


class ProbabilisticTensorDictModule(TensorDictModuleBase):
    """Builds a distribution from ``in_keys`` and writes a sample plus
    (optionally) its log-prob.

    * ``in_keys``: list, or dict mapping td-keys → dist kwarg names
      (e.g. ``{"loc": "loc", "scale": "scale"}``).
    * ``default_interaction_type`` governs sampling when no
      :func:`set_interaction_type` context is active.
    """

    def __init__(
        self,
        in_keys,
        out_keys,
        distribution_class,
        distribution_kwargs: Optional[dict] = None,
        default_interaction_type: InteractionType = InteractionType.MODE,
        return_log_prob: bool = False,
        log_prob_key: NestedKey = "sample_log_prob",
        cache_dist: bool = False,
        n_empirical_estimate: int = 1000,
        num_samples: Optional[int] = None,
    ):
        super().__init__()
        if isinstance(in_keys, dict):
            self.dist_keys = {unravel_key(k): v for k, v in in_keys.items()}
        else:
            if isinstance(in_keys, str):
                in_keys = [in_keys]
            self.dist_keys = {
                unravel_key(k): (k if isinstance(k, str) else k[-1]) for k in in_keys
            }
        self.in_keys = list(self.dist_keys.keys())
        if isinstance(out_keys, str):
            out_keys = [out_keys]
        self.out_keys = [unravel_key(k) for k in out_keys]
        self.distribution_class = distribution_class
        self.distribution_kwargs = distribution_kwargs or {}
        self.default_interaction_type = default_interaction_type
        self.return_log_prob = return_log_prob
        self.log_prob_key = log_prob_key
        if return_log_prob and log_prob_key not in self.out_keys:
            self.out_keys.append(log_prob_key)

    def get_dist(self, tensordict: TensorDictBase) -> torch.distributions.Distribution:
        kwargs = {name: tensordict.get(k) for k, name in self.dist_keys.items()}
        kwargs.update(self.distribution_kwargs)
        return self.distribution_class(**kwargs)

    def _draw(self, dist) -> torch.Tensor:
        itype = interaction_type()
        if itype is None:
            itype = self.default_interaction_type
        if itype == InteractionType.RANDOM:
            if dist.has_rsample:
                return dist.rsample()
            return dist.sample()
        if itype in (InteractionType.MODE, InteractionType.DETERMINISTIC):
            mode = getattr(dist, "mode", None)
            if mode is not None:
                try:
                    return mode() if callable(mode) else mode
                except NotImplementedError:
                    pass
            mean = getattr(dist, "deterministic_sample", None)
            if mean is not None:
                return mean
            return dist.mean
        if itype == InteractionType.MEAN:
            return dist.mean
        if itype == InteractionType.MEDIAN:
            med = getattr(dist, "median", None)
            if med is not None:
                return med() if callable(med) else med
            return dist.mean
        raise ValueError(f"unknown interaction type {itype}")

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        dist = self.get_dist(tensordict)
        sample = self._draw(dist)
        tensordict.set(self.out_keys[0], sample)
        if self.return_log_prob:
            lp = dist.log_prob(sample)
            tensordict.set(self.log_prob_key, lp)
        return tensordict

    def log_prob(self, tensordict: TensorDictBase) -> torch.Tensor:
        dist = self.get_dist(tensordict)
        return dist.log_prob(tensordict.get(self.out_keys[0]))


class ProbabilisticTensorDictSequential(TensorDictSequential):
    """Sequential whose last module is probabilistic; exposes ``get_dist``/
    ``log_prob`` over the whole chain."""

    def __init__(self, *modules, partial_tolerant: bool = False, return_composite: bool = False):
        super().__init__(*modules, partial_tolerant=partial_tolerant)

    @property
    def _prob_module(self) -> ProbabilisticTensorDictModule:
        for m in reversed(list(self.module)):
            if isinstance(m, (ProbabilisticTensorDictModule, ProbabilisticTensorDictSequential)):
                return m
        raise RuntimeError("no probabilistic module in sequence")

    def get_dist(self, tensordict: TensorDictBase) -> torch.distributions.Distribution:
        td = tensordict
        for m in list(self.module)[:-1]:
            td = m(td)
        last = self.module[-1]
        if isinstance(last, (ProbabilisticTensorDictModule, ProbabilisticTensorDictSequential)):
            return last.get_dist(td)
        raise RuntimeError("last module is not probabilistic")

    def log_prob(self, tensordict: TensorDictBase) -> torch.Tensor:
        dist = self.get_dist(tensordict)
        return dist.log_prob(tensordict.get(self._prob_module.out_keys[0]))

    @property
    def dist_sample_key(self) -> NestedKey:
        return self._prob_module.out_keys[0]


# --------------------------------------------------------------------------- #
# Functional utilities
# --------------------------------------------------------------------------- #
class TensorDictParams(nn.Module):
    """Hold a TensorDict of parameters as proper nn.Parameters so they are
    visible to optimizers (reference: tensordict.nn.TensorDictParams)."""

    def __init__(self, td: TensorDict, no_convert: bool = False):
        super().__init__()
        self._td_structure = td
        self._flat_params = nn.ParameterDict()
        for i, (k, v) in enumerate(td.items(True, True)):
            if isinstance(v, torch.Tensor) and v.is_floating_point() and not no_convert:
                name = k if isinstance(k, str) else "․".join(k)
                p = nn.Parameter(v.detach().clone()) if not isinstance(v, nn.Parameter) else v
                self._flat_params[name.replace(".", "․")] = p
                td.set(k, p)

    def to_tensordict(self) -> TensorDict:
        return self._td_structure

    def forward(self):
        return self._td_structure


def make_functional(module: nn.Module) -> TensorDict:
    """Extract params as a TensorDict (reference convert_to_functional's
    extraction step).  The module keeps its params; use
    ``torch.func.functional_call`` with the returned dict for stateless
    evaluation."""
    return TensorDict.from_module(module)


def functional_call(module: nn.Module, params: TensorDict, *args, **kwargs):
    flat = {}
    for k in params.keys(True, True):
        name = k if isinstance(k, str) else ".".join(k)
        flat[name] = params.get(k)
    return torch.func.functional_call(module, flat, args, kwargs)
