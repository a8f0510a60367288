"""Core runtime utilities for rl_amd.

MI355X-native re-design of the reference's runtime layer
(cf. pytorch/rl torchrl/_utils.py:48-1838): logger, ``timeit`` accumulation,
seeding, env-var flags, multiprocessing helpers and profiling gates.
"""
from __future__ import annotations

import collections
import contextlib
import logging
import math
import os
import sys
import time
import traceback
from typing import Any, Callable, Dict, Iterator, Optional

import numpy as np
import torch

__all__ = [
    "logger",
    "timeit",
    "seed_everything",
    "strtobool",
    "get_binary_env_var",
    "VERBOSE",
    "set_profiling_enabled",
    "profiling_enabled",
    "record_function",
    "prod",
    "_ProcessNoWarn",
    "_check_for_faulty_process",
    "hip_is_available",
    "device_of",
]


# --------------------------------------------------------------------------- #
# Logging
# --------------------------------------------------------------------------- #
def _make_logger() -> logging.Logger:
    log = logging.getLogger("rl_amd")
    if not log.handlers:
        handler = logging.StreamHandler(sys.stderr)
        handler.setFormatter(
            logging.Formatter("%(asctime)s [%(name)s][%(levelname)s] %(message)s")
        )
        log.addHandler(handler)
    log.setLevel(os.environ.get("RL_AMD_LOGLEVEL", "INFO").upper())
    return log


logger = _make_logger()


def strtobool(val: str) -> bool:
    """Convert a string representation of truth to ``True`` or ``False``."""
    val = val.lower()
    if val in ("y", "yes", "t", "true", "on", "1"):
        return True
    if val in ("n", "no", "f", "false", "off", "0"):
        return False
    raise ValueError(f"invalid truth value {val!r}")


def get_binary_env_var(key: str, default: bool = False) -> bool:
    val = os.environ.get(key)
    if val is None:
        return default
    try:
        return strtobool(val)
    except ValueError:
        return default


VERBOSE = get_binary_env_var("VERBOSE", False)


# --------------------------------------------------------------------------- #
# timeit — accumulating timer (reference torchrl/_utils.py:221)
# --------------------------------------------------------------------------- #
class timeit:
    """Context-manager / decorator that accumulates wall-clock timings globally.

    Usage::

        with timeit("collect"):
            ...
        timeit.print()
    """

    _REG: Dict[str, list] = collections.defaultdict(lambda: [0.0, 0.0, 0])

    def __init__(self, name: str):
        self.name = name

    def __call__(self, fn: Callable) -> Callable:
        def wrapper(*args, **kwargs):
            with timeit(self.name):
                return fn(*args, **kwargs)

        wrapper.__name__ = getattr(fn, "__name__", "wrapped")
        return wrapper

    def __enter__(self):
        self.t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        dt = time.perf_counter() - self.t0
        entry = self._REG[self.name]
        entry[0] = entry[0] + dt  # total
        entry[2] = entry[2] + 1  # count
        entry[1] = entry[0] / entry[2]  # mean

    @classmethod
    def todict(cls, percall: bool = True) -> Dict[str, float]:
        if percall:
            return {k: v[1] for k, v in cls._REG.items()}
        return {k: v[0] for k, v in cls._REG.items()}

    @classmethod
    def erase(cls) -> None:
        cls._REG.clear()

    @classmethod
    def print(cls, prefix: str = "") -> None:  # noqa: A003
        total = sum(v[0] for v in cls._REG.values())
        for k, (tot, mean, count) in sorted(
            cls._REG.items(), key=lambda kv: -kv[1][0]
        ):
            pct = 100.0 * tot / total if total else 0.0
            logger.info(
                f"{prefix}{k}: total={tot:.4f}s mean={mean * 1e3:.3f}ms "
                f"count={count} ({pct:.1f}%)"
            )


# --------------------------------------------------------------------------- #
# Profiling gates (reference torchrl/_utils.py:433-515)
# --------------------------------------------------------------------------- #
_PROFILING_ENABLED = get_binary_env_var("RL_AMD_PROFILING", False)


def set_profiling_enabled(mode: bool) -> None:
    global _PROFILING_ENABLED
    _PROFILING_ENABLED = bool(mode)


def profiling_enabled() -> bool:
    return _PROFILING_ENABLED


@contextlib.contextmanager
def record_function(name: str) -> Iterator[None]:
    """``torch.profiler.record_function`` range, emitted only when profiling
    is enabled so the hot path stays free of profiler overhead."""
    if _PROFILING_ENABLED:
        with torch.profiler.record_function(name):
            yield
    else:
        yield


# --------------------------------------------------------------------------- #
# Seeding
# --------------------------------------------------------------------------- #
def seed_everything(seed: int) -> int:
    import random

    random.seed(seed)
    np.random.seed(seed % (2**32))
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    return seed


def seed_generator(seed: int) -> int:
    """Next seed in a deterministic chain (for per-worker seeding)."""
    max_seed_val = (2**32) - 1
    rng = np.random.default_rng(seed)
    return int(rng.integers(0, max_seed_val))


# --------------------------------------------------------------------------- #
# Math / misc
# --------------------------------------------------------------------------- #
def prod(seq) -> int:
    return int(math.prod(seq))


def hip_is_available() -> bool:
    """True when a ROCm GPU is visible (torch.cuda IS HIP on ROCm builds)."""
    return torch.cuda.is_available()


def device_of(x: Any) -> Optional[torch.device]:
    if isinstance(x, torch.Tensor):
        return x.device
    dev = getattr(x, "device", None)
    if dev is not None:
        return torch.device(dev)
    return None


def expand_as_right(t: torch.Tensor, dest: torch.Tensor) -> torch.Tensor:
    """Expand ``t`` on the right to match ``dest``'s ndim, then broadcast."""
    while t.dim() < dest.dim():
        t = t.unsqueeze(-1)
    return t.expand_as(dest)


def expand_right(t: torch.Tensor, shape) -> torch.Tensor:
    while t.dim() < len(shape):
        t = t.unsqueeze(-1)
    return t.expand(shape)


# --------------------------------------------------------------------------- #
# Multiprocessing helpers (reference torchrl/_utils.py:48-124, :520)
# --------------------------------------------------------------------------- #
import multiprocessing as _mp  # noqa: E402


class _ProcessNoWarn(_mp.get_context("spawn").Process):
    """Spawn-context Process that silences warnings in the child and
    forwards the parent's relevant env vars."""

    def __init__(self, *args, num_threads: int | None = None, **kwargs):
        self.filter_warnings_subprocess = True
        self.num_threads = num_threads
        super().__init__(*args, **kwargs)

    def run(self):
        if self.num_threads is not None:
            torch.set_num_threads(self.num_threads)
        if self.filter_warnings_subprocess:
            import warnings

            with warnings.catch_warnings():
                warnings.simplefilter("ignore")
                return super().run()
        return super().run()


def _check_for_faulty_process(processes) -> None:
    """Raise if any worker process died; terminate the rest first
    (reference torchrl/_utils.py:520)."""
    terminate = False
    for p in processes:
        if not p.is_alive():
            terminate = True
            break
    if terminate:
        for p in processes:
            if p.is_alive():
                p.terminate()
        raise RuntimeError(
            "At least one simulation process failed; tearing the others down. "
            "Check the tracebacks above for the root cause."
        )


class _ErrorCatcher:
    """Wrap a worker target so exceptions propagate with traceback text."""

    def __init__(self, fn):
        self.fn = fn

    def __call__(self, *args, **kwargs):
        try:
            return self.fn(*args, **kwargs)
        except Exception:
            traceback.print_exc()
            raise


def implement_for(module_name: str, from_version: str = None, to_version: str = None):
    """Lightweight version-dispatch decorator (reference uses pyvers'
    ``implement_for``, torchrl/_utils.py:29).  Here a no-op pass-through that
    only gates on importability of ``module_name``."""

    def deco(fn):
        try:
            __import__(module_name.split(".")[0])
            fn._implement_for_available = True
        except ImportError:
            fn._implement_for_available = False
        return fn

    return deco


def warn(msg: str, category=UserWarning, stacklevel: int = 2) -> None:
    """One-stop warning emitter (reference torchrl.warn): routes through
    the package logger AND python warnings so both sinks see it."""
    import warnings as _warnings

    logger.warning(msg)
    _warnings.warn(msg, category=category, stacklevel=stacklevel)


# ---------------------------------------------------------------------------
# reference-parity root utilities (torchrl/__init__.py exports)
# ---------------------------------------------------------------------------

torchrl_logger = logger  # parity alias: the reference exports `torchrl_logger`

_AUTO_UNWRAP = True


def set_auto_unwrap_transformed_env(value: bool):
    """Set (or context-manage) whether nested TransformedEnvs are
    auto-flattened on construction (reference _utils.py)."""
    import contextlib

    global _AUTO_UNWRAP

    @contextlib.contextmanager
    def _ctx():
        global _AUTO_UNWRAP
        prev = _AUTO_UNWRAP
        _AUTO_UNWRAP = bool(value)
        try:
            yield
        finally:
            _AUTO_UNWRAP = prev

    _AUTO_UNWRAP = bool(value)
    return _ctx()


def auto_unwrap_transformed_env(allow_none: bool = False):
    """Current auto-unwrap setting (reference _utils.py)."""
    return _AUTO_UNWRAP


def compile_with_warmup(*args, warmup: int = 1, **kwargs):
    """Decorator: run ``warmup`` eager calls, then ``torch.compile`` the
    function (reference _utils.py).  On this MI355X stack the hot paths
    are hipGraph-captured hand-written kernels instead of compiled
    Python, so the compiled path only engages where torch.compile is
    functional."""
    import torch as _torch

    def _wrap(fn):
        state = {"calls": 0, "compiled": None}

        def inner(*a, **kw):
            if state["compiled"] is not None:
                return state["compiled"](*a, **kw)
            state["calls"] += 1
            out = fn(*a, **kw)
            if state["calls"] >= warmup:
                try:
                    state["compiled"] = _torch.compile(fn, **kwargs)
                except Exception:
                    state["compiled"] = fn
            return out

        return inner

    if args and callable(args[0]):
        return _wrap(args[0])
    return _wrap


def cuda_memory_stats(device=None) -> dict:
    """Allocator summary for one device (reference _utils.py): current /
    peak allocated+reserved bytes, in a plain dict."""
    import torch as _torch

    if not _torch.cuda.is_available():
        return {}
    return {
        "allocated": _torch.cuda.memory_allocated(device),
        "max_allocated": _torch.cuda.max_memory_allocated(device),
        "reserved": _torch.cuda.memory_reserved(device),
        "max_reserved": _torch.cuda.max_memory_reserved(device),
    }


def cuda_memory_profile(device=None) -> str:
    """Human-readable allocator report (reference _utils.py)."""
    import torch as _torch

    if not _torch.cuda.is_available():
        return "cuda unavailable"
    return _torch.cuda.memory_summary(device)


def reset_cuda_peak_stats(device=None) -> None:
    """Zero the peak-allocated counters (reference _utils.py)."""
    import torch as _torch

    if _torch.cuda.is_available():
        _torch.cuda.reset_peak_memory_stats(device)


def get_ray_default_runtime_env() -> dict:
    """Default ray runtime_env for rl_amd workers (reference
    _utils.py): propagate the ROCm/RCCL env vars workers need."""
    import os

    keys = ["HSA_ENABLE_IPC_MODE_LEGACY", "PYTORCH_ROCM_ARCH", "MASTER_ADDR", "MASTER_PORT"]
    return {"env_vars": {k: os.environ[k] for k in keys if k in os.environ}}


def merge_ray_runtime_env(base: dict, extra: dict) -> dict:
    """Merge two ray runtime_env dicts; ``extra`` wins per key, env_vars
    merge recursively (reference _utils.py)."""
    out = dict(base or {})
    for k, v in (extra or {}).items():
        if k == "env_vars":
            out[k] = {**out.get(k, {}), **v}
        else:
            out[k] = v
    return out


import contextlib as _contextlib


@_contextlib.contextmanager
def transport_backend(backend: str):
    """Scoped default payload transport for distributed components
    (reference _comm/backends.py:221).  On MI355X nodes the payload
    plane is RCCL over xGMI ("nccl"/"rccl"); "gloo" covers CPU-only
    runs and "rpc" the torch.distributed.rpc services."""
    global _TRANSPORT_BACKEND
    valid = {"nccl", "rccl", "gloo", "rpc", "shared_memory"}
    if str(backend) not in valid:
        raise ValueError(f"unknown transport backend {backend!r}; pick from {sorted(valid)}")
    prev = globals().get("_TRANSPORT_BACKEND", "nccl")
    _TRANSPORT_BACKEND = str(backend)
    try:
        yield
    finally:
        _TRANSPORT_BACKEND = prev


__all__ += [
    "torchrl_logger", "set_auto_unwrap_transformed_env",
    "auto_unwrap_transformed_env", "compile_with_warmup",
    "cuda_memory_stats", "cuda_memory_profile", "reset_cuda_peak_stats",
    "get_ray_default_runtime_env", "merge_ray_runtime_env",
    "transport_backend",
]
