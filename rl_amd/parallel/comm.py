"""Comm substrate: process-group bootstrap, TensorDict send/recv,
bucketed gradient all-reduce with stream overlap.

Reference: pytorch/rl torchrl/_comm/ (TorchDistributedTransport
distributed.py:512, rendezvous.py:30-79) and trainers/_distributed.py:138
(DDP wrap).  MI355X design: backend "nccl" IS RCCL on ROCm; xGMI is
point-to-point (7 links/GPU), so gradient buckets are sized for per-link
ring bandwidth and the all-reduce runs on a SECOND HIP stream so it
overlaps backward compute.
"""
from __future__ import annotations

import datetime
import os
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist

from ..tensordict import TensorDict, TensorDictBase

__all__ = [
    "init_distributed",
    "rendezvous_store",
    "send_tensordict",
    "recv_tensordict",
    "isend_tensordict",
    "irecv_tensordict",
    "broadcast_tensordict",
    "all_reduce_grads",
    "GradAllReducer",
]

DEFAULT_BUCKET_BYTES = 32 * 1024 * 1024  # sized for per-link xGMI ring chunks


def init_distributed(
    rank: Optional[int] = None,
    world_size: Optional[int] = None,
    backend: Optional[str] = None,
    master_addr: str = "127.0.0.1",
    master_port: int = 29500,
    timeout_s: float = 300.0,
) -> None:
    """init_process_group with env-var fallbacks; RCCL when GPUs exist."""
    if dist.is_initialized():
        return
    rank = int(os.environ.get("RANK", rank if rank is not None else 0))
    world_size = int(
        os.environ.get("WORLD_SIZE", world_size if world_size is not None else 1)
    )
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", master_addr)
    os.environ.setdefault("MASTER_PORT", str(master_port))
    dist.init_process_group(
        backend=backend,
        rank=rank,
        world_size=world_size,
        timeout=datetime.timedelta(seconds=timeout_s),
    )


def rendezvous_store(
    is_server: bool, addr: str = "127.0.0.1", port: int = 29510, world_size: int = 2
):
    """TCPStore rendezvous (reference _comm/rendezvous.py:51)."""
    return dist.TCPStore(
        addr, port, world_size, is_server, timeout=datetime.timedelta(seconds=120)
    )


def _ordered_leaves(td: TensorDictBase):
    keys = sorted(
        (k for k, v in td.items(True, True) if isinstance(v, torch.Tensor)),
        key=lambda k: k if isinstance(k, str) else ".".join(k),
    )
    return [(k, td.get(k)) for k in keys]


def send_tensordict(td: TensorDictBase, dst: int, group=None) -> None:
    """Leaf-wise ordered send; receiver must hold a same-spec td
    (reference collectors/distributed/generic.py:277)."""
    for _k, v in _ordered_leaves(td):
        dist.send(v.contiguous(), dst=dst, group=group)


def recv_tensordict(td: TensorDictBase, src: int, group=None) -> TensorDictBase:
    for _k, v in _ordered_leaves(td):
        buf = v.contiguous()
        dist.recv(buf, src=src, group=group)
        v.copy_(buf)
    return td


def isend_tensordict(td: TensorDictBase, dst: int, group=None) -> List:
    return [
        dist.isend(v.contiguous(), dst=dst, group=group)
        for _k, v in _ordered_leaves(td)
    ]


def irecv_tensordict(td: TensorDictBase, src: int, group=None) -> List:
    works = []
    for _k, v in _ordered_leaves(td):
        if not v.is_contiguous():
            raise RuntimeError("irecv needs contiguous leaves")
        works.append(dist.irecv(v, src=src, group=group))
    return works


def broadcast_tensordict(td: TensorDictBase, src: int = 0, group=None) -> TensorDictBase:
    """Packed single-buffer broadcast — one RCCL call for all leaves
    (reference weight_update/llm/vllm_nccl.py:356 packed broadcast)."""
    leaves = _ordered_leaves(td)
    if not leaves:
        return td
    device = leaves[0][1].device
    total = sum(v.numel() * v.element_size() for _k, v in leaves)
    buf = torch.empty(total, dtype=torch.uint8, device=device)
    rank = dist.get_rank(group)
    if rank == src:
        off = 0
        for _k, v in leaves:
            n = v.numel() * v.element_size()
            buf[off : off + n] = v.detach().contiguous().view(-1).view(torch.uint8)
            off += n
    dist.broadcast(buf, src=src, group=group)
    if rank != src:
        off = 0
        with torch.no_grad():
            for _k, v in leaves:
                n = v.numel() * v.element_size()
                v.copy_(buf[off : off + n].view(v.dtype).view(v.shape))
                off += n
    return td


def all_reduce_grads(params: Sequence[torch.nn.Parameter], world_size: int, group=None) -> None:
    """One-shot flat all-reduce of all grads (small models: a single RCCL
    call beats per-tensor calls on xGMI)."""
    grads = [p.grad for p in params if p.grad is not None]
    if not grads:
        return
    flat = torch.cat([g.reshape(-1) for g in grads])
    dist.all_reduce(flat, group=group)
    flat /= world_size
    off = 0
    for g in grads:
        n = g.numel()
        g.copy_(flat[off : off + n].view_as(g))
        off += n


class GradAllReducer:
    """Bucketed gradient all-reduce overlapped with backward.

    Registers post-accumulate-grad hooks; when a bucket fills, its
    all-reduce is launched on a dedicated HIP stream so communication
    overlaps the rest of backward (the pattern the reference gets from
    torch DDP, trainers/_distributed.py:138, rebuilt explicitly for
    TensorDict-module training where DDP's forward-wrapping doesn't fit).
    """

    def __init__(
        self,
        params: Sequence[torch.nn.Parameter],
        world_size: Optional[int] = None,
        bucket_bytes: int = DEFAULT_BUCKET_BYTES,
        group=None,
        register_hooks: bool = True,
    ):
        """``register_hooks=True`` overlaps bucket all-reduces with an
        EAGER backward (hooks fire as grads accumulate).  With
        ``register_hooks=False`` the backward may be a hipGraph replay
        (hooks cannot fire inside a replay): call :meth:`reduce` after
        the replay — buckets still go out as async RCCL calls on the
        comm stream, then :meth:`finalize` applies them."""
        self.params = [p for p in params if p.requires_grad]
        self.world_size = world_size or dist.get_world_size(group)
        self.group = group
        self.bucket_bytes = bucket_bytes
        self._use_stream = torch.cuda.is_available()
        self._comm_stream = torch.cuda.Stream() if self._use_stream else None
        # buckets in REVERSE parameter order (grads arrive back-to-front)
        self.buckets: List[List[torch.nn.Parameter]] = []
        cur: List[torch.nn.Parameter] = []
        size = 0
        for p in reversed(self.params):
            cur.append(p)
            size += p.numel() * p.element_size()
            if size >= bucket_bytes:
                self.buckets.append(cur)
                cur, size = [], 0
        if cur:
            self.buckets.append(cur)
        self._param_bucket = {}
        for bi, bucket in enumerate(self.buckets):
            for p in bucket:
                self._param_bucket[p] = bi
        self._pending = [0] * len(self.buckets)
        self._works: List = []
        self._flat: List[Optional[torch.Tensor]] = [None] * len(self.buckets)
        self.hooks_enabled = True
        self._hooks = []
        if register_hooks:
            for p in self.params:
                self._hooks.append(
                    p.register_post_accumulate_grad_hook(self._on_grad)
                )
        self._reset_counts()

    def reduce(self):
        """Explicit launch of every bucket's all-reduce (for use after a
        graph-replayed backward, where hooks cannot fire).  Follow with
        :meth:`finalize` before the optimizer step."""
        for bi in range(len(self.buckets)):
            self._launch(bi)

    def _reset_counts(self):
        self._pending = [len(b) for b in self.buckets]
        self._works = []

    def _on_grad(self, p: torch.nn.Parameter):
        if not self.hooks_enabled:
            return
        bi = self._param_bucket[p]
        self._pending[bi] -= 1
        if self._pending[bi] == 0:
            self._launch(bi)

    def _launch(self, bi: int):
        bucket = self.buckets[bi]
        grads = [p.grad for p in bucket if p.grad is not None]
        if not grads:
            return
        if self._use_stream:
            self._comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._comm_stream):
                flat = torch.cat([g.reshape(-1) for g in grads])
                work = dist.all_reduce(flat, group=self.group, async_op=True)
                self._flat[bi] = flat
                self._works.append((bi, work))
        else:
            flat = torch.cat([g.reshape(-1) for g in grads])
            work = dist.all_reduce(flat, group=self.group, async_op=True)
            self._flat[bi] = flat
            self._works.append((bi, work))

    def finalize(self):
        """Call after backward(), before optimizer.step()."""
        for bi, work in self._works:
            work.wait()
        if self._use_stream:
            torch.cuda.current_stream().wait_stream(self._comm_stream)
        for bi, _w in self._works:
            flat = self._flat[bi]
            flat /= self.world_size
            off = 0
            for p in self.buckets[bi]:
                if p.grad is not None:
                    n = p.grad.numel()
                    p.grad.copy_(flat[off : off + n].view_as(p.grad))
                    off += n
            self._flat[bi] = None
        self._reset_counts()

    def remove(self):
        for h in self._hooks:
            h.remove()
