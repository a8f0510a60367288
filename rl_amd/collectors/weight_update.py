"""Legacy (v1) weight-updater API kept for reference parity.

Reference: pytorch/rl torchrl/collectors/weight_update.py
(WeightUpdaterBase, VanillaWeightUpdater, MultiProcessedWeightUpdater,
RemoteModuleWeightUpdater) — v1 API retained alongside the v2 schemes in
rl_amd.weight_update.
"""
from __future__ import annotations

from typing import Callable, Dict, Iterable, Optional

import torch

from ..tensordict import TensorDict, TensorDictBase

__all__ = [
    "WeightUpdaterBase",
    "VanillaWeightUpdater",
    "MultiProcessedWeightUpdater",
    "RemoteModuleWeightUpdater",
    "RayWeightUpdater",
    "DistributedWeightUpdater",
    "RPCWeightUpdater",
]


class WeightUpdaterBase:
    """Push/pull policy weights into a collector's workers."""

    collector = None

    def register_collector(self, collector) -> "WeightUpdaterBase":
        self.collector = collector
        return self

    def _get_server_weights(self) -> TensorDictBase:
        raise NotImplementedError

    def _push_weights(self, weights: TensorDictBase) -> None:
        raise NotImplementedError

    def update_weights(self, weights: Optional[TensorDictBase] = None) -> None:
        if weights is None:
            weights = self._get_server_weights()
        self._push_weights(weights)

    __call__ = update_weights


class VanillaWeightUpdater(WeightUpdaterBase):
    """Single-process: copy the train policy's weights into the collector
    policy."""

    def __init__(self, weight_getter: Optional[Callable] = None, policy_weights: Optional[TensorDictBase] = None):
        self.weight_getter = weight_getter
        self.policy_weights = policy_weights

    def _get_server_weights(self):
        if self.weight_getter is not None:
            src = self.weight_getter()
            if isinstance(src, TensorDictBase):
                return src
            return TensorDict.from_module(src).apply(lambda t: t.detach())
        return self.policy_weights

    def _push_weights(self, weights):
        with torch.no_grad():
            weights.to_module(self.collector.policy)


class MultiProcessedWeightUpdater(WeightUpdaterBase):
    """Broadcast a weight copy to every worker pipe (v1 analog of the
    pipe scheme)."""

    def __init__(self, get_server_weights: Callable, policy_weights: Optional[Dict] = None):
        self.get_server_weights = get_server_weights

    def _get_server_weights(self):
        src = self.get_server_weights()
        if isinstance(src, TensorDictBase):
            return src
        return TensorDict.from_module(src).apply(lambda t: t.detach().cpu())

    def _push_weights(self, weights):
        self.collector.update_policy_weights_(weights)


class RemoteModuleWeightUpdater(MultiProcessedWeightUpdater):
    """Alias for parity: remote-module flavored pipe updater."""


class RayWeightUpdater(WeightUpdaterBase):
    """Ray-actor weight pushes (reference weight_update.py
    RayWeightUpdater): broadcasts the learner state through the object
    store to every remote collector actor — gated on `ray`."""

    def __init__(self, policy, remote_collectors, max_interval: int = 0):
        import importlib.util

        if importlib.util.find_spec("ray") is None:
            raise ImportError(
                "RayWeightUpdater requires the `ray` package, which is not "
                "installed in this image. Use MultiProcessedWeightUpdater or "
                "the RCCL DistributedWeightSyncScheme instead."
            )
        self.policy = policy
        self.remote_collectors = list(remote_collectors)
        self.max_interval = max_interval
        self._updates = 0

    def update_weights(self, weights=None) -> None:
        import ray

        if weights is None:
            weights = {
                k: v.detach().cpu() for k, v in self.policy.state_dict().items()
            }
        ref = ray.put(weights)
        ray.get(
            [w.update_policy_weights_.remote(ref) for w in self.remote_collectors]
        )
        self._updates += 1


class DistributedWeightUpdater(WeightUpdaterBase):
    """Weight pushes to torch.distributed collector worker ranks over
    RCCL/gloo stores (reference distributed/generic.py:1209; deprecated
    there in favor of DistributedWeightSyncScheme — rl_amd keeps the
    name as a thin adapter over the collector's own push)."""

    def __init__(self, collector):
        self.collector = collector

    def push_weights(self, policy_or_weights=None, worker_ids=None):
        self.collector.update_policy_weights_(policy_or_weights)

    __call__ = push_weights


class RPCWeightUpdater(WeightUpdaterBase):
    """Weight pushes to RPC collector workers (reference
    distributed/rpc.py:951) — adapter over RPCCollector's push."""

    def __init__(self, collector):
        self.collector = collector

    def push_weights(self, policy_or_weights=None, worker_ids=None):
        self.collector.update_policy_weights_(policy_or_weights)

    __call__ = push_weights
