"""InferenceServer — batched cross-client policy inference.

Reference: pytorch/rl torchrl/modules/inference_server/_server.py:261
(InferenceServer), _client.py:114 (PolicyClientModule), transports
_mp.py / _shared_memory.py.

Clients submit TensorDicts; the server batches pending requests, runs ONE
policy forward (amortizing the GPU launch + MFMA efficiency over the
batch), and scatters results back.  In-process threads use queues; the
mp transport uses pipes.
"""
from __future__ import annotations

import queue
import threading
import time
from typing import Callable, List, Optional, Tuple

import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase, cat as td_cat

__all__ = ["InferenceServer", "PolicyClient"]


class InferenceServer:
    """Thread-backed batched inference service.

    ``max_batch_size`` requests (or whatever arrived within
    ``max_latency_ms``) are batched per forward.
    """

    def __init__(
        self,
        policy: Callable[[TensorDictBase], TensorDictBase],
        *,
        max_batch_size: int = 64,
        max_latency_ms: float = 5.0,
        device=None,
    ):
        self.policy = policy
        self.max_batch_size = max_batch_size
        self.max_latency_ms = max_latency_ms
        self.device = device
        self._requests: "queue.Queue[Tuple[TensorDictBase, queue.Queue]]" = queue.Queue()
        self._running = False
        self._thread: Optional[threading.Thread] = None
        self.stats = {"batches": 0, "requests": 0, "mean_batch": 0.0}

    # -- lifecycle ------------------------------------------------------- #
    def start(self) -> "InferenceServer":
        self._running = True
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self) -> None:
        self._running = False
        if self._thread is not None:
            self._thread.join(timeout=10)

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()

    # -- server loop ------------------------------------------------------ #
    def _loop(self):
        while self._running:
            batch: List[Tuple[TensorDictBase, queue.Queue]] = []
            try:
                item = self._requests.get(timeout=0.05)
                batch.append(item)
            except queue.Empty:
                continue
            deadline = time.perf_counter() + self.max_latency_ms / 1000.0
            while len(batch) < self.max_batch_size:
                remaining = deadline - time.perf_counter()
                if remaining <= 0:
                    break
                try:
                    batch.append(self._requests.get(timeout=remaining))
                except queue.Empty:
                    break
            tds = [b[0] for b in batch]
            sizes = [td.batch_size[0] if td.batch_size else 1 for td in tds]
            joined = td_cat(
                [td if td.batch_size else td.unsqueeze(0) for td in tds], 0
            )
            if self.device is not None:
                joined = joined.to(self.device)
            with torch.no_grad():
                out = self.policy(joined)
            off = 0
            for (td, reply_q), n in zip(batch, sizes):
                piece = out[off : off + n]
                if not td.batch_size:
                    piece = piece[0]
                reply_q.put(piece.cpu() if self.device is not None else piece)
                off += n
            self.stats["batches"] += 1
            self.stats["requests"] += len(batch)
            self.stats["mean_batch"] = self.stats["requests"] / self.stats["batches"]

    # -- client API -------------------------------------------------------- #
    def submit(self, td: TensorDictBase, timeout: float = 30.0) -> TensorDictBase:
        reply: queue.Queue = queue.Queue()
        self._requests.put((td, reply))
        return reply.get(timeout=timeout)

    def make_client(self) -> "PolicyClient":
        return PolicyClient(self)


class PolicyClient(TensorDictModuleBase):
    """Callable handle usable as a policy inside collectors
    (reference _client.py:114)."""

    def __init__(self, server: InferenceServer):
        super().__init__()
        self.server = server
        self.in_keys = getattr(server.policy, "in_keys", [])
        self.out_keys = getattr(server.policy, "out_keys", [])

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        out = self.server.submit(td)
        td.update(out)
        return td


# --------------------------------------------------------------------- #
# Cross-process serving: shared-memory slots + a server process
# (reference _server.py:961 ProcessInferenceServer, _slot.py slots,
# _shared_memory.py transport)
# --------------------------------------------------------------------- #
_SLOT_FREE = 0
_SLOT_REQUEST = 1
_SLOT_RESPONSE = 2


class _SlotBlock:
    """Pre-allocated shared-memory request/response slots.

    Each slot is one row of a shared request TensorDict and one row of a
    shared response TensorDict plus an atomic state flag — a client
    writes its request IN PLACE and flips the flag; the server batches
    every ready slot with one gather, runs ONE policy forward, scatters
    responses and flips back.  No pickling anywhere on the hot path."""

    def __init__(self, request_example: TensorDictBase, response_example: TensorDictBase, n_slots: int):
        import multiprocessing as mp

        ctx = mp.get_context("spawn")

        def expand(example):
            out = TensorDict({}, batch_size=[n_slots, *example.batch_size])
            for k in example.keys(True, True):
                v = example.get(k)
                out.set(k, torch.zeros(n_slots, *v.shape, dtype=v.dtype))
            return out.share_memory_()

        self.requests = expand(request_example)
        self.responses = expand(response_example)
        self.flags = torch.zeros(n_slots, dtype=torch.int64).share_memory_()
        self.n_slots = n_slots
        self._next_slot = ctx.Value("l", 0)
        self.batches = ctx.Value("l", 0)
        self.served = ctx.Value("l", 0)

    def acquire_slot(self) -> int:
        with self._next_slot.get_lock():
            idx = self._next_slot.value
            if idx >= self.n_slots:
                raise RuntimeError("no free inference slots")
            self._next_slot.value += 1
        return idx


class SlotPolicyClient(TensorDictModuleBase):
    """Client side of the slot transport: usable as a collector policy
    in a worker process (reference _client.py:114 over shared memory)."""

    def __init__(self, block: _SlotBlock, slot: Optional[int] = None, timeout: float = 30.0):
        super().__init__()
        self.block = block
        self.slot = block.acquire_slot() if slot is None else slot
        self.timeout = timeout
        self.in_keys = []
        self.out_keys = []

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        block = self.block
        i = self.slot
        req = block.requests[i]
        for k in req.keys(True, True):
            v = td.get(k, None)
            if v is not None:
                req.get(k).copy_(v.reshape(req.get(k).shape))
        block.flags[i] = _SLOT_REQUEST
        deadline = time.monotonic() + self.timeout
        spins = 0
        while int(block.flags[i]) != _SLOT_RESPONSE:
            spins += 1
            if spins > 2000:
                time.sleep(5e-5)
            if spins % 1024 == 0 and time.monotonic() > deadline:
                raise TimeoutError("inference server did not respond")
        resp = block.responses[i]
        for k in resp.keys(True, True):
            td.set(k, resp.get(k).clone())
        block.flags[i] = _SLOT_FREE
        return td


def _inference_server_proc(policy_factory, block: _SlotBlock, stop_flag, max_batch: int):
    torch.set_num_threads(1)
    policy = policy_factory()
    flags = block.flags
    while not stop_flag.value:
        ready = (flags == _SLOT_REQUEST).nonzero().reshape(-1)
        if ready.numel() == 0:
            time.sleep(5e-5)
            continue
        ready = ready[:max_batch]
        batch = block.requests[ready].clone()
        with torch.no_grad():
            out = policy(batch)
        for j, i in enumerate(ready.tolist()):
            resp = block.responses[i]
            for k in resp.keys(True, True):
                src = out.get(k, None)
                if src is not None:
                    resp.get(k).copy_(src[j].reshape(resp.get(k).shape))
            flags[i] = _SLOT_RESPONSE
        with block.batches.get_lock():
            block.batches.value += 1
        with block.served.get_lock():
            block.served.value += int(ready.numel())


class ProcessInferenceServer:
    """Inference server running in its OWN process over shared-memory
    slots (reference _server.py:961).

    Args:
        policy_factory: picklable zero-arg callable building the policy
            inside the server process.
        request_example / response_example: per-request TensorDicts
            fixing the slot layout (shapes + dtypes).
        n_slots: max concurrent clients.
        max_batch_size: slots batched per forward.
    """

    def __init__(
        self,
        policy_factory: Callable[[], Callable],
        request_example: TensorDictBase,
        response_example: TensorDictBase,
        *,
        n_slots: int = 16,
        max_batch_size: int = 64,
    ):
        import multiprocessing as mp

        self.block = _SlotBlock(request_example, response_example, n_slots)
        self._stop = mp.get_context("spawn").Value("b", False)
        self._proc = None
        self._factory = policy_factory
        self.max_batch_size = max_batch_size

    def start(self) -> "ProcessInferenceServer":
        import multiprocessing as mp

        ctx = mp.get_context("spawn")
        self._proc = ctx.Process(
            target=_inference_server_proc,
            args=(self._factory, self.block, self._stop, self.max_batch_size),
            daemon=True,
        )
        self._proc.start()
        return self

    def stop(self) -> None:
        self._stop.value = True
        if self._proc is not None:
            self._proc.join(timeout=10)
            if self._proc.is_alive():
                self._proc.terminate()

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()

    def make_client(self, timeout: float = 30.0) -> SlotPolicyClient:
        return SlotPolicyClient(self.block, timeout=timeout)

    @property
    def stats(self) -> dict:
        b = int(self.block.batches.value)
        s = int(self.block.served.value)
        return {"batches": b, "requests": s, "mean_batch": s / b if b else 0.0}


__all__ += ["ProcessInferenceServer", "SlotPolicyClient"]
