"""Ray-backed collectors (gated — ray not in this image).

Reference: pytorch/rl torchrl/collectors/distributed/ray.py:81
(RayCollector) and collectors/llm/ray_collector.py:32 (RayLLMCollector).
The rl_amd distributed collection paths that ARE available offline:
:class:`~rl_amd.collectors.distributed.DistributedCollector`
(torch.distributed over RCCL/gloo) and
:class:`~rl_amd.collectors.RPCCollector` (TensorPipe RPC).
"""
from __future__ import annotations

import importlib.util

__all__ = ["RayCollector", "RayLLMCollector"]

_has_ray = importlib.util.find_spec("ray") is not None

_MSG = (
    "requires the `ray` package, which is not installed in this image. "
    "Use DistributedCollector (RCCL/gloo) or RPCCollector instead."
)


class RayCollector:
    def __init__(self, *args, **kwargs):
        if not _has_ray:
            raise ImportError(f"RayCollector {_MSG}")
        raise NotImplementedError("ray backend scaffolding")


class RayLLMCollector:
    def __init__(self, *args, **kwargs):
        if not _has_ray:
            raise ImportError(f"RayLLMCollector {_MSG}")
        raise NotImplementedError("ray backend scaffolding")
