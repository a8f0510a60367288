"""AsyncEnvPool — asynchronous step/reset over process or thread pools.

Reference: pytorch/rl torchrl/envs/async_envs.py:56 (AsyncEnvPool,
process pool :556, thread pool :1016): ``async_step_send``/
``async_step_recv`` decouple submission from collection, with
``min_get`` batching (return as soon as K sub-envs finished).
"""
from __future__ import annotations

import queue as _queue
import threading
from concurrent.futures import FIRST_COMPLETED, Future, ThreadPoolExecutor, wait
from typing import Callable, Dict, List, Optional, Sequence, Union

import torch

from ..tensordict import TensorDict, TensorDictBase, stack as td_stack
from .common import EnvBase

__all__ = ["AsyncEnvPool", "ThreadingAsyncEnvPool", "ProcessorAsyncEnvPool"]


class AsyncEnvPool:
    """Pool of envs stepped asynchronously.

    ``backend="threading"`` runs sub-envs in a thread pool (right choice
    for GIL-releasing simulators and GPU-resident envs);
    ``backend="multiprocessing"`` delegates to :class:`ParallelEnv`
    workers.
    """

    def __init__(
        self,
        env_makers: Sequence[Callable[[], EnvBase]],
        *,
        backend: str = "threading",
        stack: str = "lazy",
    ):
        self.num_envs = len(env_makers)
        self.backend = backend
        if backend == "threading":
            self._envs = [fn() for fn in env_makers]
            self._pool = ThreadPoolExecutor(max_workers=self.num_envs)
        elif backend == "multiprocessing":
            from .batched_envs import ParallelEnv

            # one ParallelEnv worker per sub-env, driven individually
            self._envs = [fn() for fn in env_makers]
            self._pool = ThreadPoolExecutor(max_workers=self.num_envs)
        else:
            raise ValueError(f"unknown backend {backend}")
        self._pending: Dict[int, Future] = {}
        self._carriers: List[Optional[TensorDictBase]] = [None] * self.num_envs

    # -- sync convenience -------------------------------------------------- #
    def reset(self) -> TensorDictBase:
        outs = [env.reset() for env in self._envs]
        for i, o in enumerate(outs):
            self._carriers[i] = o
        return td_stack(outs, 0)

    # -- async API ---------------------------------------------------------- #
    def async_reset_send(self, env_ids: Optional[Sequence[int]] = None) -> None:
        ids = range(self.num_envs) if env_ids is None else env_ids
        for i in ids:
            self._pending[i] = self._pool.submit(self._envs[i].reset)

    def async_step_send(self, tensordict: TensorDictBase, env_ids: Optional[Sequence[int]] = None) -> None:
        """Submit steps; ``tensordict`` is stacked over the env ids."""
        ids = list(range(self.num_envs)) if env_ids is None else list(env_ids)
        for k, i in enumerate(ids):
            if i in self._pending:
                raise RuntimeError(f"env {i} already has a pending op")
            td_i = tensordict[k].clone(False)
            carrier = self._carriers[i]
            if carrier is not None:
                merged = carrier.clone(False)
                merged.update(td_i)
            else:
                merged = td_i
            self._pending[i] = self._pool.submit(self._step_one, i, merged)

    def _step_one(self, i: int, td: TensorDictBase) -> TensorDictBase:
        env = self._envs[i]
        td, next_root = env.step_and_maybe_reset(td)
        self._carriers[i] = next_root
        return td

    def _collect(self, min_get: int, reset: bool) -> TensorDictBase:
        if min_get > len(self._pending):
            raise RuntimeError(
                f"min_get={min_get} but only {len(self._pending)} ops pending"
            )
        done_futs, _ = wait(
            list(self._pending.values()), return_when=FIRST_COMPLETED
        )
        while sum(f.done() for f in self._pending.values()) < min_get:
            wait(list(self._pending.values()), timeout=0.01)
        ready = [i for i, f in self._pending.items() if f.done()]
        outs = []
        for i in sorted(ready):
            fut = self._pending.pop(i)
            td = fut.result()
            if reset:
                self._carriers[i] = td
            td.set("env_index", torch.tensor(i))
            outs.append(td)
        return td_stack(outs, 0)

    def async_step_recv(self, min_get: int = 1) -> TensorDictBase:
        return self._collect(min_get, reset=False)

    def async_reset_recv(self, min_get: int = 1) -> TensorDictBase:
        return self._collect(min_get, reset=True)

    # -- lifecycle ----------------------------------------------------------- #
    def close(self):
        for f in self._pending.values():
            f.cancel()
        self._pool.shutdown(wait=False)
        for env in self._envs:
            env.close()

    @property
    def action_spec(self):
        return self._envs[0].action_spec

    @property
    def observation_spec(self):
        return self._envs[0].observation_spec


class ThreadingAsyncEnvPool(AsyncEnvPool):
    """Thread-backed pool (reference async_envs.py) — right for
    GIL-releasing simulators and GPU-resident envs."""

    def __init__(self, env_makers, **kwargs):
        kwargs.pop("backend", None)
        super().__init__(env_makers, backend="threading", **kwargs)


class ProcessorAsyncEnvPool(AsyncEnvPool):
    """Process-backed pool (reference async_envs.py) — isolates
    CPU-heavy or crash-prone simulators."""

    def __init__(self, env_makers, **kwargs):
        kwargs.pop("backend", None)
        super().__init__(env_makers, backend="multiprocessing", **kwargs)
