"""Multiprocess collectors: MultiSyncCollector / MultiAsyncCollector /
AsyncCollector.

Reference: pytorch/rl torchrl/collectors/_multi_sync.py:27 (MultiSync,
iterator:220), _multi_async.py:25, _single_async.py:18, worker loop
_runner.py:36 (``_main_async_collector``), naming per
torchrl/collectors/__init__.py:27-47.

Each worker process owns env+policy copies and runs an inner
:class:`Collector`; results travel back through an mp.Queue.  On an
MI355X node workers are pinned round-robin across the 8 GPUs
(``device="cuda:{i%8}"``) and the queue hop carries only CPU handles —
the payload stays in HBM when storing_device is a GPU.
"""
from __future__ import annotations

import multiprocessing as mp
import os
import queue as _queue
import time
from typing import Any, Callable, List, Optional, Sequence, Union

import torch

from .._utils import _check_for_faulty_process, _ProcessNoWarn, logger
from ..envs.common import EnvBase
from ..envs.utils import ExplorationType
from ..tensordict import TensorDict, TensorDictBase, cat as td_cat, stack as td_stack
from .collectors import BaseCollector, Collector
from .utils import split_trajectories

__all__ = [
    "MultiSyncCollector",
    "MultiAsyncCollector",
    "AsyncCollector",
    "MultiSyncDataCollector",
    "MultiaSyncDataCollector",
    "_Interruptor",
]

_TIMEOUT = 60.0


class _Interruptor:
    """Shared flag for preemptive straggler interruption
    (reference collectors/_constants.py)."""

    def __init__(self):
        # spawn context: the flag crosses into spawn-context workers
        self._stop = mp.get_context("spawn").Value("b", False)

    def start_collection(self):
        with self._stop.get_lock():
            self._stop.value = False

    def stop_collection(self):
        with self._stop.get_lock():
            self._stop.value = True

    def collection_stopped(self) -> bool:
        return bool(self._stop.value)


def _worker_loop(
    pipe,
    queue_out,
    create_env_fn,
    policy,
    collector_kwargs,
    idx: int,
    seed: Optional[int],
    interruptor=None,
):
    """Child-process loop (reference _runner.py:36)."""
    torch.set_num_threads(max(1, torch.get_num_threads() // 2))
    try:
        inner = Collector(
            create_env_fn,
            policy,
            interruptor=interruptor,
            **collector_kwargs,
        )
        if seed is not None:
            inner.set_seed(seed)
        it = inner.iterator()
        pipe.send(("ready", idx))
        while True:
            cmd, arg = pipe.recv()
            if cmd == "continue":
                batch = next(it)
                queue_out.put((idx, batch))
            elif cmd == "seed":
                out = inner.set_seed(arg)
                pipe.send(("seeded", out))
            elif cmd == "update":
                inner.update_policy_weights_(arg)
                pipe.send(("updated", idx))
            elif cmd == "state_dict":
                pipe.send(("state_dict", inner.state_dict()))
            elif cmd == "load_state_dict":
                inner.load_state_dict(arg)
                pipe.send(("loaded", idx))
            elif cmd == "close":
                inner.shutdown()
                pipe.send(("closed", idx))
                break
    except KeyboardInterrupt:
        pass
    except Exception as err:
        import traceback

        traceback.print_exc()
        try:
            pipe.send(("error", repr(err)))
        except Exception:
            pass
        raise


class _MultiCollectorBase(BaseCollector):
    """Shared process management (reference _multi_base.py:79)."""

    def __init__(
        self,
        create_env_fn: Sequence[Union[EnvBase, Callable[[], EnvBase]]],
        policy=None,
        *,
        frames_per_batch: int,
        total_frames: int = -1,
        device=None,
        storing_device=None,
        policy_device=None,
        env_device=None,
        reset_at_each_iter: bool = False,
        postproc=None,
        split_trajs: bool = False,
        exploration_type: ExplorationType = ExplorationType.RANDOM,
        update_at_each_batch: bool = False,
        preemptive_threshold: Optional[float] = None,
        num_threads: Optional[int] = None,
        cat_results: Union[str, int, None] = None,
        seed: Optional[int] = None,
        **kwargs,
    ):
        if callable(create_env_fn) and not isinstance(create_env_fn, (list, tuple)):
            raise TypeError("pass a LIST of env constructors (one per worker)")
        self.num_workers = len(create_env_fn)
        self.frames_per_batch = frames_per_batch
        self.total_frames = total_frames if total_frames > 0 else float("inf")
        self.postproc = postproc
        self.split_trajs = split_trajs
        self.update_at_each_batch = update_at_each_batch
        self.preemptive_threshold = preemptive_threshold
        self.cat_results = cat_results
        self._frames = 0
        self.closed = True
        self.policy = policy

        frames_per_worker = frames_per_batch // self.num_workers
        if frames_per_batch % self.num_workers != 0:
            raise ValueError(
                f"frames_per_batch ({frames_per_batch}) must divide by "
                f"num_workers ({self.num_workers})"
            )
        devices = self._resolve_devices(device, self.num_workers)
        storing_devices = self._resolve_devices(storing_device, self.num_workers)

        ctx = mp.get_context("spawn")
        self._queue_out = ctx.Queue()
        self.pipes = []
        self.procs = []
        self.interruptor = (
            _Interruptor() if preemptive_threshold is not None else None
        )
        collector_kwargs = dict(
            frames_per_batch=frames_per_worker,
            total_frames=-1,
            reset_at_each_iter=reset_at_each_iter,
            exploration_type=exploration_type,
            **kwargs,
        )
        for i, env_fn in enumerate(create_env_fn):
            parent_pipe, child_pipe = ctx.Pipe()
            ckw = dict(collector_kwargs)
            if devices[i] is not None:
                ckw["device"] = devices[i]
            if storing_devices[i] is not None:
                ckw["storing_device"] = storing_devices[i]
            proc = _ProcessNoWarn(
                target=_worker_loop,
                args=(
                    child_pipe,
                    self._queue_out,
                    env_fn,
                    policy,
                    ckw,
                    i,
                    None if seed is None else seed + i,
                    self.interruptor,
                ),
            )
            proc.daemon = True
            proc.start()
            child_pipe.close()
            self.pipes.append(parent_pipe)
            self.procs.append(proc)
        for pipe in self.pipes:
            msg, _ = pipe.recv()
            if msg == "error":
                self.shutdown()
                raise RuntimeError("collector worker failed during startup")
            assert msg == "ready"
        self.closed = False

    @staticmethod
    def _resolve_devices(device, n):
        if device is None:
            return [None] * n
        if isinstance(device, (list, tuple)):
            return list(device)
        return [device] * n

    def _check_procs(self):
        _check_for_faulty_process(self.procs)

    def update_policy_weights_(self, policy_or_weights=None, **kwargs) -> None:
        if policy_or_weights is None and self.policy is not None:
            policy_or_weights = {
                k: v.cpu() for k, v in self.policy.state_dict().items()
            }
        elif hasattr(policy_or_weights, "state_dict"):
            policy_or_weights = {
                k: v.cpu() for k, v in policy_or_weights.state_dict().items()
            }
        if policy_or_weights is None:
            return
        for pipe in self.pipes:
            pipe.send(("update", policy_or_weights))
        for pipe in self.pipes:
            msg, _ = pipe.recv()
            assert msg == "updated"

    def set_seed(self, seed: int, static_seed: bool = False) -> int:
        for i, pipe in enumerate(self.pipes):
            pipe.send(("seed", seed + i if not static_seed else seed))
        out = seed
        for pipe in self.pipes:
            msg, out = pipe.recv()
            assert msg == "seeded"
        return out

    def state_dict(self) -> dict:
        out = {}
        for i, pipe in enumerate(self.pipes):
            pipe.send(("state_dict", None))
            msg, sd = pipe.recv()
            out[f"worker{i}"] = sd
        return out

    def load_state_dict(self, sd: dict) -> None:
        for i, pipe in enumerate(self.pipes):
            key = f"worker{i}"
            if key in sd:
                pipe.send(("load_state_dict", sd[key]))
                msg, _ = pipe.recv()

    def shutdown(self, timeout: Optional[float] = None) -> None:
        if self.closed:
            return
        for pipe in self.pipes:
            try:
                pipe.send(("close", None))
            except (BrokenPipeError, OSError):
                pass
        deadline = time.time() + (timeout or 10.0)
        for pipe in self.pipes:
            try:
                if pipe.poll(max(0.1, deadline - time.time())):
                    pipe.recv()
            except (BrokenPipeError, OSError, EOFError):
                pass
        for proc in self.procs:
            proc.join(timeout=max(0.1, deadline - time.time()))
            if proc.is_alive():
                proc.terminate()
        self.closed = True

    def __del__(self):
        try:
            self.shutdown()
        except Exception:
            pass


class MultiSyncCollector(_MultiCollectorBase):
    """Gather one batch from EVERY worker each iteration
    (reference _multi_sync.py:27)."""

    def iterator(self):
        while self._frames < self.total_frames:
            if self.update_at_each_batch:
                self.update_policy_weights_()
            if self.interruptor is not None:
                self.interruptor.start_collection()
            for pipe in self.pipes:
                pipe.send(("continue", None))
            results: dict = {}
            n_needed = self.num_workers
            if self.interruptor is not None:
                threshold = max(1, int(self.preemptive_threshold * n_needed))
            while len(results) < n_needed:
                self._check_procs()
                try:
                    idx, batch = self._queue_out.get(timeout=_TIMEOUT)
                except _queue.Empty:
                    self._check_procs()
                    continue
                results[idx] = batch
                if (
                    self.interruptor is not None
                    and len(results) >= threshold
                ):
                    self.interruptor.stop_collection()
            ordered = [results[i] for i in sorted(results)]
            if self.interruptor is not None:
                # preempted stragglers return fewer time steps; right-pad
                # them (zeros) so the worker batches stack
                t_max = max(b.batch_size[-1] for b in ordered)
                from ..tensordict import pad as td_pad

                ordered = [
                    b
                    if b.batch_size[-1] == t_max
                    else td_pad(b, [0, 0] * (len(b.batch_size) - 1) + [0, t_max - b.batch_size[-1]])
                    for b in ordered
                ]
            cat_results = self.cat_results
            if cat_results in (None, "stack"):
                out = td_stack(ordered, 0)
            else:
                dim = 0 if cat_results == 0 else int(cat_results)
                out = td_cat(ordered, dim)
            self._frames += self.frames_per_batch
            if self.postproc is not None:
                out = self.postproc(out)
            if self.split_trajs:
                out = split_trajectories(out)
            yield out


class MultiAsyncCollector(_MultiCollectorBase):
    """First-come-first-served batches (reference _multi_async.py:25)."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._primed = False
        self.frames_per_worker = self.frames_per_batch // self.num_workers

    def iterator(self):
        if not self._primed:
            for pipe in self.pipes:
                pipe.send(("continue", None))
            self._primed = True
        while self._frames < self.total_frames:
            self._check_procs()
            try:
                idx, batch = self._queue_out.get(timeout=_TIMEOUT)
            except _queue.Empty:
                self._check_procs()
                continue
            # keep that worker rolling
            self.pipes[idx].send(("continue", None))
            self._frames += self.frames_per_worker
            if self.postproc is not None:
                batch = self.postproc(batch)
            if self.split_trajs:
                batch = split_trajectories(batch)
            yield batch

    def shutdown(self, timeout: Optional[float] = None) -> None:
        # drain queue before closing
        if not self.closed:
            try:
                while True:
                    self._queue_out.get_nowait()
            except _queue.Empty:
                pass
        super().shutdown(timeout)


class AsyncCollector(MultiAsyncCollector):
    """Single-worker async collector (reference _single_async.py:18)."""

    def __init__(self, create_env_fn, policy=None, **kwargs):
        if not isinstance(create_env_fn, (list, tuple)):
            create_env_fn = [create_env_fn]
        super().__init__(create_env_fn, policy, **kwargs)


# Reference compat aliases (old names)
MultiSyncDataCollector = MultiSyncCollector
MultiaSyncDataCollector = MultiAsyncCollector
