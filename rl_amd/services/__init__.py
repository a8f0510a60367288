"""Service registry — named long-lived components.

Reference: pytorch/rl torchrl/services/ (ServiceBase base.py:52,
RayService ray_service.py:58).  Without Ray in the image, the local
registry covers single-node use: services register by name, lookups
return live objects; ``DistributedServiceRegistry`` stubs the multi-node
path (torch.distributed rendezvous) for parity.
"""
from __future__ import annotations

import threading
from typing import Any, Dict, Optional

__all__ = ["ServiceBase", "LocalService", "get_services", "reset_services", "service_backend"]


class ServiceBase:
    """ABC (reference base.py:52)."""

    def register(self, name: str, obj: Any, **kwargs) -> None:
        raise NotImplementedError

    def get(self, name: str) -> Any:
        raise NotImplementedError

    def __contains__(self, name: str) -> bool:
        raise NotImplementedError

    def list(self):
        raise NotImplementedError

    def unregister(self, name: str) -> None:
        raise NotImplementedError


class LocalService(ServiceBase):
    """In-process named registry (thread-safe)."""

    def __init__(self):
        self._store: Dict[str, Any] = {}
        self._lock = threading.Lock()

    def register(self, name: str, obj: Any = None, factory=None, **kwargs) -> Any:
        with self._lock:
            if name in self._store:
                raise KeyError(f"service {name!r} already registered")
            if obj is None and factory is not None:
                obj = factory(**kwargs)
            self._store[name] = obj
            return obj

    def get(self, name: str) -> Any:
        with self._lock:
            return self._store[name]

    def __contains__(self, name: str) -> bool:
        with self._lock:
            return name in self._store

    def list(self):
        with self._lock:
            return sorted(self._store)

    def unregister(self, name: str) -> None:
        with self._lock:
            self._store.pop(name, None)


_GLOBAL: Optional[LocalService] = None


def get_services(backend: str = "local", **kwargs) -> ServiceBase:
    """(reference services.__init__ get_services)"""
    global _GLOBAL
    if backend != "local":
        raise NotImplementedError(
            f"service backend {backend!r} not available in this build (no ray)"
        )
    if _GLOBAL is None:
        _GLOBAL = LocalService()
    return _GLOBAL


def reset_services() -> None:
    global _GLOBAL
    _GLOBAL = None


def service_backend(name: str = "local", **kwargs):
    """Select a service backend by name (reference
    torchrl/_comm/backends.py:191): ``"local"`` returns the in-process
    registry; ``"ray"`` is gated (ray is not installed here)."""
    if name == "local":
        return get_services()
    if name == "ray":
        raise RuntimeError(
            "service backend 'ray' not available in this build (no ray)"
        )
    raise ValueError(f"unknown service backend {name!r}")
