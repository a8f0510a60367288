"""ReplayBufferService — a buffer served to other processes.

Reference: pytorch/rl torchrl/_comm/replay_service.py:344 (RB as a
service): producers extend from collector processes, the learner samples
— all over mp pipes with a server thread owning the buffer.  Payloads
ride torch's shared-memory pickling, so tensors cross without copies.
"""
from __future__ import annotations

import multiprocessing as mp
import threading
from typing import Any, Optional

from ..tensordict import TensorDictBase

__all__ = ["ReplayBufferService", "ReplayBufferClient"]


class ReplayBufferService:
    """Owns the buffer; serves add/extend/sample/update_priority/len."""

    def __init__(self, buffer):
        self.buffer = buffer
        self._pipes = []
        self._threads = []
        self._stop = threading.Event()

    def make_client_conn(self):
        """Create a pipe endpoint for a (possibly remote-process) client."""
        ctx = mp.get_context("spawn")
        server_end, client_end = ctx.Pipe()
        t = threading.Thread(target=self._serve, args=(server_end,), daemon=True)
        t.start()
        self._pipes.append(server_end)
        self._threads.append(t)
        return client_end

    def _serve(self, conn):
        while not self._stop.is_set():
            try:
                if not conn.poll(0.2):
                    continue
                msg = conn.recv()
            except (EOFError, OSError):
                return
            op = msg["op"]
            try:
                if op == "extend":
                    idx = self.buffer.extend(msg["data"])
                    conn.send({"ok": True, "index": idx})
                elif op == "add":
                    idx = self.buffer.add(msg["data"])
                    conn.send({"ok": True, "index": idx})
                elif op == "sample":
                    data = self.buffer.sample(msg.get("batch_size"))
                    conn.send({"ok": True, "data": data})
                elif op == "update_priority":
                    self.buffer.update_priority(msg["index"], msg["priority"])
                    conn.send({"ok": True})
                elif op == "len":
                    conn.send({"ok": True, "len": len(self.buffer)})
                elif op == "close":
                    conn.send({"ok": True})
                    return
                else:
                    conn.send({"ok": False, "error": f"unknown op {op}"})
            except Exception as e:  # propagate errors to the client
                conn.send({"ok": False, "error": repr(e)})

    def shutdown(self):
        self._stop.set()
        for t in self._threads:
            t.join(timeout=2)


class ReplayBufferClient:
    """Buffer-like handle over a service connection — drop-in for the
    collector's ``replay_buffer`` argument in another process."""

    def __init__(self, conn):
        self.conn = conn
        self._lock = threading.Lock()

    def _call(self, op: str, **kwargs) -> dict:
        with self._lock:
            self.conn.send({"op": op, **kwargs})
            out = self.conn.recv()
        if not out.get("ok"):
            raise RuntimeError(f"replay service error: {out.get('error')}")
        return out

    def extend(self, data: TensorDictBase):
        return self._call("extend", data=data).get("index")

    def add(self, data: TensorDictBase):
        return self._call("add", data=data).get("index")

    def sample(self, batch_size: Optional[int] = None) -> TensorDictBase:
        return self._call("sample", batch_size=batch_size)["data"]

    def update_priority(self, index, priority):
        self._call("update_priority", index=index, priority=priority)

    def __len__(self):
        return self._call("len")["len"]

    def close(self):
        try:
            self._call("close")
        except Exception:
            pass
