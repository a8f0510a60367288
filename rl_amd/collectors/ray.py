"""Ray-backed collectors (import-gated — ray does not ship in this
image, so construction raises ImportError; the code paths are complete
against Ray's public actor API and run when ray is installed).

Reference: pytorch/rl torchrl/collectors/distributed/ray.py:81
(RayCollector) and collectors/llm/ray_collector.py:32 (RayLLMCollector).
The rl_amd distributed collection paths that ARE testable offline:
:class:`~rl_amd.collectors.distributed.DistributedCollector`
(torch.distributed over RCCL/gloo) and
:class:`~rl_amd.collectors.RPCCollector` (TensorPipe RPC).
"""
from __future__ import annotations

import importlib.util
from typing import Callable, List, Optional, Sequence, Union

__all__ = ["RayCollector", "RayLLMCollector"]

_has_ray = importlib.util.find_spec("ray") is not None

_MSG = (
    "requires the `ray` package, which is not installed in this image. "
    "Use DistributedCollector (RCCL/gloo) or RPCCollector instead."
)


class _CollectorWorker:
    """Actor body: one inner Collector per Ray worker (decorated with
    ``ray.remote`` at runtime — this module must import without ray)."""

    def __init__(self, collector_cls, create_env_fn, policy, **kwargs):
        self.inner = collector_cls(create_env_fn, policy, **kwargs)
        self._iter = iter(self.inner)

    def next(self):
        try:
            return next(self._iter)
        except StopIteration:
            return None

    def update_policy_weights_(self, weights):
        self.inner.update_policy_weights_(weights)

    def set_seed(self, seed):
        return self.inner.set_seed(seed)

    def shutdown(self):
        self.inner.shutdown()


def _ray():
    if not _has_ray:
        raise ImportError(f"RayCollector {_MSG}")
    import ray

    if not ray.is_initialized():
        ray.init(ignore_reinit_error=True)
    return ray


class RayCollector:
    """Collection sharded over Ray actor workers (reference ray.py:81).

    ``sync=True``: every iteration gathers one batch from EVERY worker
    and concatenates (MultiSync semantics over the cluster).
    ``sync=False``: first-come-first-served batches (MultiAsync).
    """

    def __init__(
        self,
        create_env_fn: Union[Callable, Sequence[Callable]],
        policy=None,
        *,
        frames_per_batch: int,
        total_frames: int = -1,
        num_workers: Optional[int] = None,
        sync: bool = True,
        remote_configs: Optional[dict] = None,
        collector_kwargs: Optional[dict] = None,
    ):
        ray = _ray()
        from .collectors import Collector

        if callable(create_env_fn):
            create_env_fn = [create_env_fn] * (num_workers or 1)
        self.num_workers = len(create_env_fn)
        self.frames_per_batch = frames_per_batch
        self.total_frames = total_frames if total_frames > 0 else float("inf")
        self.sync = sync
        per_worker = frames_per_batch // self.num_workers
        Worker = ray.remote(**(remote_configs or {"num_cpus": 1}))(_CollectorWorker)
        self._workers = [
            Worker.remote(
                Collector,
                fn,
                policy,
                frames_per_batch=per_worker,
                total_frames=-1,
                **(collector_kwargs or {}),
            )
            for fn in create_env_fn
        ]
        self._frames = 0

    def __iter__(self):
        import ray

        from ..tensordict import cat as td_cat

        if self.sync:
            while self._frames < self.total_frames:
                outs = ray.get([w.next.remote() for w in self._workers])
                outs = [o for o in outs if o is not None]
                if not outs:
                    return
                batch = td_cat([o.reshape(-1) for o in outs], 0)
                self._frames += batch.batch_size[0]
                yield batch
        else:
            pending = {w.next.remote(): w for w in self._workers}
            while self._frames < self.total_frames and pending:
                done, _ = ray.wait(list(pending.keys()), num_returns=1)
                ref = done[0]
                worker = pending.pop(ref)
                out = ray.get(ref)
                if out is not None:
                    self._frames += out.reshape(-1).batch_size[0]
                    pending[worker.next.remote()] = worker
                    yield out

    def update_policy_weights_(self, weights=None) -> None:
        import ray

        ray.get([w.update_policy_weights_.remote(weights) for w in self._workers])

    def set_seed(self, seed: int) -> int:
        import ray

        outs = ray.get(
            [w.set_seed.remote(seed + i) for i, w in enumerate(self._workers)]
        )
        return outs[-1]

    def shutdown(self, timeout: Optional[float] = None) -> None:
        import ray

        ray.get([w.shutdown.remote() for w in self._workers])
        for w in self._workers:
            ray.kill(w)


class _LLMCollectorWorker:
    def __init__(self, collector_cls, env_factory, policy_factory, **kwargs):
        self.inner = collector_cls(env_factory(), policy_factory(), **kwargs)
        self._iter = iter(self.inner)

    def next(self):
        try:
            return next(self._iter)
        except StopIteration:
            return None

    def update_policy_weights_(self, weights):
        self.inner.update_policy_weights_(weights)

    def shutdown(self):
        self.inner.shutdown()


class RayLLMCollector:
    """LLM rollout collection in a Ray actor (reference
    llm/ray_collector.py:32): generation runs remotely (typically on the
    engine's placement group); batches of conversation TensorDicts come
    back through the object store."""

    def __init__(
        self,
        env_factory: Callable,
        policy_factory: Callable,
        *,
        dialog_turns_per_batch: int = 16,
        total_dialog_turns: int = -1,
        remote_configs: Optional[dict] = None,
        collector_kwargs: Optional[dict] = None,
    ):
        ray = _ray()
        from .llm import LLMCollector

        Worker = ray.remote(**(remote_configs or {"num_cpus": 1}))(_LLMCollectorWorker)
        self._worker = Worker.remote(
            LLMCollector,
            env_factory,
            policy_factory,
            dialog_turns_per_batch=dialog_turns_per_batch,
            total_dialog_turns=total_dialog_turns,
            **(collector_kwargs or {}),
        )

    def __iter__(self):
        import ray

        while True:
            out = ray.get(self._worker.next.remote())
            if out is None:
                return
            yield out

    def update_policy_weights_(self, weights=None) -> None:
        import ray

        ray.get(self._worker.update_policy_weights_.remote(weights))

    def shutdown(self, timeout: Optional[float] = None) -> None:
        import ray

        ray.get(self._worker.shutdown.remote())
        ray.kill(self._worker)
