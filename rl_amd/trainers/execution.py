"""Learner execution groups — multi-GPU DDP-style training orchestration.

Reference: pytorch/rl torchrl/trainers/_distributed.py:62
(_DDPProcessGroup, DDP wrap :138) and _execution.py.

rl_amd form: `LearnerGroup` wraps a loss module's parameters with the
bucketed overlapped :class:`~rl_amd.parallel.comm.GradAllReducer` (xGMI
point-to-point topology; buckets sized for per-link ring bandwidth), and
`launch_learners` spawns one process per GPU over RCCL for standalone
use.  torch's DistributedDataParallel is deliberately NOT used: its
forward wrapping breaks TensorDict-module losses (multiple forwards per
step, dict outputs); explicit bucketed all-reduce after backward gives
the same overlap without the wrapper.
"""
from __future__ import annotations

import os
from typing import Callable, List, Optional, Sequence

import torch
import torch.distributed as dist

from .._utils import _ProcessNoWarn
from ..parallel.comm import GradAllReducer, init_distributed

__all__ = ["LearnerGroup", "launch_learners"]


class LearnerGroup:
    """Wrap (loss_module, optimizer) for data-parallel training.

    Usage in each rank's training loop::

        group = LearnerGroup(loss_module)
        ...
        loss.backward()
        group.finalize_grads()   # waits overlapped all-reduces
        optimizer.step()
    """

    def __init__(
        self,
        module: torch.nn.Module,
        *,
        world_size: Optional[int] = None,
        bucket_bytes: int = 32 * 1024 * 1024,
        broadcast_init: bool = True,
    ):
        if not dist.is_initialized():
            raise RuntimeError("init_distributed() first")
        self.module = module
        self.world_size = world_size or dist.get_world_size()
        if broadcast_init:
            with torch.no_grad():
                for p in module.parameters():
                    dist.broadcast(p.data, src=0)
        self.reducer = GradAllReducer(
            list(module.parameters()), world_size=self.world_size, bucket_bytes=bucket_bytes
        )

    def finalize_grads(self):
        self.reducer.finalize()

    def remove(self):
        self.reducer.remove()


def launch_learners(
    fn: Callable[[int, int], None],
    world_size: int,
    *,
    master_addr: str = "127.0.0.1",
    master_port: int = 29512,
    backend: Optional[str] = None,
) -> List:
    """Spawn ``world_size`` learner processes; each runs
    ``fn(rank, world_size)`` after process-group init (one per GPU when
    CUDA is available)."""

    def _entry(rank):
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["LOCAL_RANK"] = str(rank)
        init_distributed(
            rank=rank,
            world_size=world_size,
            backend=backend,
            master_addr=master_addr,
            master_port=master_port,
        )
        if torch.cuda.is_available():
            torch.cuda.set_device(rank % torch.cuda.device_count())
        try:
            fn(rank, world_size)
        finally:
            if dist.is_initialized():
                dist.destroy_process_group()

    procs = []
    for r in range(world_size):
        p = _ProcessNoWarn(target=_entry, args=(r,))
        p.start()
        procs.append(p)
    return procs
