"""Gym-backend selection and spec-conversion registration.

Reference: pytorch/rl torchrl/envs/libs/gym.py (set_gym_backend,
gym_backend, register_gym_spec_conversion) and libs/utils.py
(get_available_libraries): torchrl lets users pin which of
gym/gymnasium backs the ``GymEnv`` wrappers and register converters
from third-party space types to tensor specs.  The same contract is
kept here; in this offline image neither backend is installed, so the
context manager mostly records the preference for the gated wrappers.
"""
from __future__ import annotations

import importlib
import importlib.util
from typing import Callable, Dict, Optional

__all__ = [
    "set_gym_backend",
    "gym_backend",
    "register_gym_spec_conversion",
    "get_available_libraries",
]

_GYM_BACKEND: Optional[str] = None
_SPEC_CONVERSIONS: Dict[type, Callable] = {}


class set_gym_backend:
    """Pin the gym backend ("gym" or "gymnasium"); usable as a context
    manager or called for a global effect (reference libs/gym.py)."""

    def __init__(self, backend):
        if not isinstance(backend, str):
            backend = getattr(backend, "__name__", str(backend))
        if backend not in ("gym", "gymnasium"):
            raise ValueError(f"unknown gym backend {backend!r}")
        self.backend = backend
        self._prev: Optional[str] = None

    def _apply(self):
        global _GYM_BACKEND
        self._prev = _GYM_BACKEND
        _GYM_BACKEND = self.backend

    def __enter__(self):
        self._apply()
        return self

    def __exit__(self, *exc):
        global _GYM_BACKEND
        _GYM_BACKEND = self._prev
        return False

    def set(self):
        self._apply()


def gym_backend(submodule: Optional[str] = None):
    """Import and return the active gym backend (or a submodule of it)."""
    name = _GYM_BACKEND
    if name is None:
        for cand in ("gymnasium", "gym"):
            if importlib.util.find_spec(cand) is not None:
                name = cand
                break
    if name is None:
        raise ImportError("neither gymnasium nor gym is installed in this image")
    target = name if submodule is None else f"{name}.{submodule}"
    return importlib.import_module(target)


def register_gym_spec_conversion(space_type: type, converter: Optional[Callable] = None):
    """Register ``converter(space) -> TensorSpec`` for a third-party
    space type; usable as a decorator (reference libs/gym.py)."""

    def _register(fn):
        _SPEC_CONVERSIONS[space_type] = fn
        return fn

    if converter is not None:
        return _register(converter)
    return _register


def get_registered_spec_conversion(space_type: type) -> Callable:
    for klass in getattr(space_type, "__mro__", (space_type,)):
        if klass in _SPEC_CONVERSIONS:
            return _SPEC_CONVERSIONS[klass]
    raise KeyError(f"no spec conversion registered for {space_type}")


def get_available_libraries() -> Dict[str, bool]:
    """Which wrappable simulator/env libraries are importable here
    (reference libs/utils.py)."""
    libs = [
        "gym", "gymnasium", "dm_control", "brax", "jumanji", "vmas",
        "pettingzoo", "envpool", "isaacgym", "meltingpot", "open_spiel",
        "smacv2", "mlagents_envs", "procgen", "robohive", "habitat",
        "minari", "gym_robotics", "mo_gymnasium",
    ]
    return {name: importlib.util.find_spec(name) is not None for name in libs}
