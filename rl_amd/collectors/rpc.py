"""RPCCollector — collection over torch.distributed.rpc (TensorPipe).

Reference: pytorch/rl torchrl/collectors/distributed/rpc.py:107
(RPCDataCollector): worker ranks host remote Collector objects; the
coordinator pulls batches with rpc_sync and pushes weights with
remote calls.
"""
from __future__ import annotations

import os
import socket
from typing import Callable, List, Optional, Sequence

import torch
import torch.distributed.rpc as rpc

from .._utils import _ProcessNoWarn
from ..tensordict import TensorDict, TensorDictBase, stack as td_stack
from .collectors import BaseCollector, Collector

__all__ = ["RPCCollector", "rpc_worker_main"]

_REMOTE_COLLECTORS = {}


def _make_remote_collector(key, env_fn, policy, collector_kwargs):
    col = Collector(env_fn, policy, **collector_kwargs)
    _REMOTE_COLLECTORS[key] = iter(col.iterator())
    _REMOTE_COLLECTORS[f"{key}_col"] = col
    return True


def _next_batch(key):
    batch = next(_REMOTE_COLLECTORS[key])
    return batch


def _update_weights(key, state_dict):
    _REMOTE_COLLECTORS[f"{key}_col"].update_policy_weights_(state_dict)
    return True


def _shutdown_remote(key):
    _REMOTE_COLLECTORS[f"{key}_col"].shutdown()
    return True


def rpc_worker_main(rank: int, world_size: int, master_addr: str, master_port: int):
    """Worker entry: join the RPC group and serve until shutdown."""
    os.environ["MASTER_ADDR"] = master_addr
    os.environ["MASTER_PORT"] = str(master_port)
    rpc.init_rpc(f"worker{rank}", rank=rank, world_size=world_size)
    rpc.shutdown()  # blocks serving requests until the group shuts down


def _find_free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class RPCCollector(BaseCollector):
    """Coordinator (rank 0) over N RPC worker ranks."""

    def __init__(
        self,
        create_env_fn: Sequence[Callable],
        policy=None,
        *,
        frames_per_batch: int,
        total_frames: int = -1,
        master_addr: str = "127.0.0.1",
        master_port: Optional[int] = None,
        collector_kwargs: Optional[dict] = None,
        launcher: str = "mp",
    ):
        self.num_workers = len(create_env_fn)
        self.frames_per_batch = frames_per_batch
        self.total_frames = total_frames if total_frames > 0 else float("inf")
        self.policy = policy
        self._frames = 0
        frames_per_worker = frames_per_batch // self.num_workers
        master_port = master_port or _find_free_port()
        world = self.num_workers + 1
        self.procs: List = []
        if launcher == "mp":
            for i in range(self.num_workers):
                p = _ProcessNoWarn(
                    target=rpc_worker_main,
                    args=(i + 1, world, master_addr, master_port),
                )
                p.daemon = True
                p.start()
                self.procs.append(p)
        os.environ["MASTER_ADDR"] = master_addr
        os.environ["MASTER_PORT"] = str(master_port)
        rpc.init_rpc("coordinator", rank=0, world_size=world)
        ckw = dict(frames_per_batch=frames_per_worker, total_frames=-1, **(collector_kwargs or {}))
        futs = [
            rpc.rpc_async(
                f"worker{i + 1}",
                _make_remote_collector,
                args=(f"c{i}", create_env_fn[i], policy, ckw),
            )
            for i in range(self.num_workers)
        ]
        for f in futs:
            f.wait()
        self.closed = False

    def iterator(self):
        while self._frames < self.total_frames:
            futs = [
                rpc.rpc_async(f"worker{i + 1}", _next_batch, args=(f"c{i}",))
                for i in range(self.num_workers)
            ]
            outs = [f.wait() for f in futs]
            self._frames += self.frames_per_batch
            yield td_stack(outs, 0)

    def update_policy_weights_(self, policy_or_weights=None, **kwargs) -> None:
        src = policy_or_weights if policy_or_weights is not None else self.policy
        if src is None:
            return
        if hasattr(src, "state_dict"):
            sd = {k: v.detach().cpu() for k, v in src.state_dict().items()}
        else:
            sd = src
        futs = [
            rpc.rpc_async(f"worker{i + 1}", _update_weights, args=(f"c{i}", sd))
            for i in range(self.num_workers)
        ]
        for f in futs:
            f.wait()

    def shutdown(self, timeout: Optional[float] = None) -> None:
        if self.closed:
            return
        try:
            futs = [
                rpc.rpc_async(f"worker{i + 1}", _shutdown_remote, args=(f"c{i}",))
                for i in range(self.num_workers)
            ]
            for f in futs:
                f.wait()
            rpc.shutdown()
        except Exception:
            pass
        for p in self.procs:
            p.join(timeout=10)
            if p.is_alive():
                p.terminate()
        self.closed = True
