"""Distributed collectors over torch.distributed (RCCL on GPU, gloo on CPU).

Reference: pytorch/rl torchrl/collectors/distributed/generic.py:351
(DistributedCollector), sync.py:136 (DistributedSyncCollector), worker
loop generic.py:101-157; data path = leaf-wise isend/irecv of TensorDict
batches (§2.3 of the survey), control via small broadcast commands.

MI355X topology note: trajectory gather is per-peer point-to-point — each
worker GPU sends its batch to the learner over its own xGMI link (send/recv,
not ring all-gather), which is exactly how 7-link point-to-point xGMI wants
to be fed.

Launchers: "mp" spawns worker processes locally (gloo/RCCL over
127.0.0.1) — also the CPU test path; with an external launcher
(torchrun) pass ``launcher="env"`` and let every rank call
:func:`distributed_worker_main` itself.
"""
from __future__ import annotations

import io
import os
import pickle
import socket
import time
from typing import Callable, List, Optional, Sequence, Union

import torch
import torch.distributed as dist

from .._utils import _check_for_faulty_process, _ProcessNoWarn, logger
from ..parallel.comm import (
    broadcast_tensordict,
    irecv_tensordict,
    recv_tensordict,
    send_tensordict,
)
from ..tensordict import TensorDict, TensorDictBase, stack as td_stack
from .collectors import BaseCollector, Collector

__all__ = ["DistributedCollector", "DistributedSyncCollector", "distributed_worker_main"]

_CMD_COLLECT = 0
_CMD_UPDATE = 1
_CMD_STOP = 2
_CMD_SEED = 3


def _find_free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _td_structure(td: TensorDictBase) -> TensorDictBase:
    """Zero-filled CPU clone used as a recv buffer blueprint."""
    return td._fast_apply(lambda t: torch.zeros_like(t, device="cpu")).cpu()


def distributed_worker_main(
    rank: int,
    world_size: int,
    master_addr: str,
    master_port: int,
    backend: str,
    create_env_fn,
    policy,
    collector_kwargs: dict,
    device=None,
):
    """Worker entry (reference _DistributedDataCollectorWorker loop,
    generic.py:101-157)."""
    os.environ["MASTER_ADDR"] = master_addr
    os.environ["MASTER_PORT"] = str(master_port)
    dist.init_process_group(backend=backend, rank=rank, world_size=world_size)
    if device is not None and torch.cuda.is_available():
        torch.cuda.set_device(device)
    inner = Collector(create_env_fn, policy, **collector_kwargs)
    it = inner.iterator()
    # publish the batch structure once (pickle via object collectives)
    example = next(it)
    struct = _td_structure(example)
    dist.gather_object(struct, None, dst=0)
    pending = example
    cmd = torch.zeros(1, dtype=torch.long)
    while True:
        dist.broadcast(cmd, src=0)
        c = int(cmd.item())
        if c == _CMD_COLLECT:
            batch = pending if pending is not None else next(it)
            pending = None
            send_tensordict(batch.cpu(), dst=0)
        elif c == _CMD_UPDATE:
            weights = TensorDict.from_module(inner.policy).apply(
                lambda t: t.detach().cpu().clone()
            )
            broadcast_tensordict(weights, src=0)
            with torch.no_grad():
                weights.to_module(inner.policy)
        elif c == _CMD_SEED:
            seed_t = torch.zeros(1, dtype=torch.long)
            dist.broadcast(seed_t, src=0)
            inner.set_seed(int(seed_t.item()) + rank)
        elif c == _CMD_STOP:
            break
    inner.shutdown()
    dist.destroy_process_group()


class DistributedCollector(BaseCollector):
    """Rank-0 coordinator over N worker ranks (reference generic.py:351).

    ``launcher="mp"``: spawns the workers itself and joins the process
    group as rank 0.
    """

    def __init__(
        self,
        create_env_fn: Sequence[Callable],
        policy=None,
        *,
        frames_per_batch: int,
        total_frames: int = -1,
        backend: Optional[str] = None,
        launcher: str = "mp",
        master_addr: str = "127.0.0.1",
        master_port: Optional[int] = None,
        update_after_each_batch: bool = False,
        collector_kwargs: Optional[dict] = None,
        sync: bool = True,
        **kwargs,
    ):
        self.num_workers = len(create_env_fn)
        self.frames_per_batch = frames_per_batch
        self.total_frames = total_frames if total_frames > 0 else float("inf")
        self.update_after_each_batch = update_after_each_batch
        self.policy = policy
        self.sync = sync
        self._frames = 0
        frames_per_worker = frames_per_batch // self.num_workers
        if backend is None:
            backend = "gloo"
        self.backend = backend
        world_size = self.num_workers + 1
        master_port = master_port or _find_free_port()
        ckw = dict(
            frames_per_batch=frames_per_worker, total_frames=-1, **(collector_kwargs or {})
        )
        self.procs: List = []
        if launcher == "mp":
            for i, env_fn in enumerate(create_env_fn):
                proc = _ProcessNoWarn(
                    target=distributed_worker_main,
                    args=(
                        i + 1,
                        world_size,
                        master_addr,
                        master_port,
                        backend,
                        env_fn,
                        policy,
                        ckw,
                    ),
                )
                proc.daemon = True
                proc.start()
                self.procs.append(proc)
            os.environ["MASTER_ADDR"] = master_addr
            os.environ["MASTER_PORT"] = str(master_port)
            dist.init_process_group(backend=backend, rank=0, world_size=world_size)
        elif launcher == "env":
            # externally launched (torchrun): PG must already exist
            if not dist.is_initialized():
                raise RuntimeError("launcher='env' expects an initialized PG")
        else:
            raise ValueError(f"unknown launcher {launcher}")
        # gather batch structures from all workers
        structures: List = [None] * world_size
        dist.gather_object(None, structures, dst=0)
        self._buffers = [structures[i + 1] for i in range(self.num_workers)]
        self.closed = False

    def _broadcast_cmd(self, c: int):
        cmd = torch.tensor([c], dtype=torch.long)
        dist.broadcast(cmd, src=0)

    def iterator(self):
        while self._frames < self.total_frames:
            if self.update_after_each_batch:
                self.update_policy_weights_()
            self._broadcast_cmd(_CMD_COLLECT)
            outs = []
            for i in range(self.num_workers):
                recv_tensordict(self._buffers[i], src=i + 1)
                outs.append(self._buffers[i].clone())
            batch = td_stack(outs, 0)
            self._frames += self.frames_per_batch
            yield batch

    def update_policy_weights_(self, policy_or_weights=None, **kwargs) -> None:
        if self.policy is None and policy_or_weights is None:
            return
        self._broadcast_cmd(_CMD_UPDATE)
        src = policy_or_weights if policy_or_weights is not None else self.policy
        if isinstance(src, TensorDictBase):
            weights = src.cpu()
        else:
            weights = TensorDict.from_module(src).apply(
                lambda t: t.detach().cpu().clone()
            )
        broadcast_tensordict(weights, src=0)

    def set_seed(self, seed: int, static_seed: bool = False) -> int:
        self._broadcast_cmd(_CMD_SEED)
        dist.broadcast(torch.tensor([seed], dtype=torch.long), src=0)
        return seed

    def shutdown(self, timeout: Optional[float] = None) -> None:
        if self.closed:
            return
        try:
            self._broadcast_cmd(_CMD_STOP)
        except Exception:
            pass
        for p in self.procs:
            p.join(timeout=10.0)
            if p.is_alive():
                p.terminate()
        if dist.is_initialized():
            dist.destroy_process_group()
        self.closed = True


class DistributedSyncCollector(DistributedCollector):
    """Alias emphasizing the synchronous gather (reference sync.py:136)."""

    def __init__(self, *args, **kwargs):
        kwargs["sync"] = True
        super().__init__(*args, **kwargs)
