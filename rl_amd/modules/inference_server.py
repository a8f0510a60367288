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
