"""Mailbox / CommandChannel / request-reply — control-plane IPC.

Reference: pytorch/rl torchrl/_comm/ (Mailbox mailbox.py:185 with
liveness watch :26, CommandChannel command.py, request_reply.py).
"""
from __future__ import annotations

import pickle
import threading
import time
import uuid
from multiprocessing.connection import Connection
from typing import Any, Callable, Dict, Optional

__all__ = ["Mailbox", "CommandChannel", "RequestReply"]


class Mailbox:
    """Bidirectional message channel over an mp.Pipe with a liveness
    watcher (reference mailbox.py:185)."""

    def __init__(self, conn: Connection, peer_process=None, watch_interval: float = 5.0):
        self.conn = conn
        self.peer_process = peer_process
        self.watch_interval = watch_interval
        self._dead = threading.Event()
        self._watcher: Optional[threading.Thread] = None
        if peer_process is not None:
            self._watcher = threading.Thread(target=self._watch, daemon=True)
            self._watcher.start()

    def _watch(self):
        while not self._dead.is_set():
            if self.peer_process is not None and not self.peer_process.is_alive():
                self._dead.set()
                return
            time.sleep(self.watch_interval)

    @property
    def peer_alive(self) -> bool:
        return not self._dead.is_set()

    def send(self, msg: Any) -> None:
        if not self.peer_alive:
            raise BrokenPipeError("mailbox peer died")
        self.conn.send(msg)

    def recv(self, timeout: Optional[float] = None) -> Any:
        deadline = None if timeout is None else time.monotonic() + timeout
        while True:
            wait = 0.5 if deadline is None else min(0.5, deadline - time.monotonic())
            if wait <= 0:
                raise TimeoutError("mailbox recv timed out")
            if self.conn.poll(wait):
                return self.conn.recv()
            if not self.peer_alive:
                raise BrokenPipeError("mailbox peer died while waiting")

    def poll(self, timeout: float = 0.0) -> bool:
        return self.conn.poll(timeout)

    def close(self):
        self._dead.set()
        try:
            self.conn.close()
        except OSError:
            pass


class CommandChannel:
    """Named-command dispatch over a Mailbox (reference command.py)."""

    def __init__(self, mailbox: Mailbox):
        self.mailbox = mailbox
        self._handlers: Dict[str, Callable] = {}

    def register(self, name: str, handler: Callable) -> None:
        self._handlers[name] = handler

    def send_command(self, name: str, payload: Any = None) -> None:
        self.mailbox.send({"cmd": name, "payload": payload})

    def serve_once(self, timeout: Optional[float] = None) -> bool:
        try:
            msg = self.mailbox.recv(timeout)
        except TimeoutError:
            return False
        handler = self._handlers.get(msg["cmd"])
        if handler is None:
            raise KeyError(f"no handler for command {msg['cmd']!r}")
        result = handler(msg.get("payload"))
        if result is not None:
            self.mailbox.send({"cmd": f"{msg['cmd']}_result", "payload": result})
        return True

    def serve_forever(self, stop_event: Optional[threading.Event] = None):
        while stop_event is None or not stop_event.is_set():
            try:
                self.serve_once(timeout=1.0)
            except BrokenPipeError:
                return


class RequestReply:
    """Correlated request/reply over a Mailbox (reference
    request_reply.py): requests carry ids, replies match them."""

    def __init__(self, mailbox: Mailbox):
        self.mailbox = mailbox
        self._pending: Dict[str, Any] = {}

    def request(self, payload: Any, timeout: float = 30.0) -> Any:
        rid = uuid.uuid4().hex
        self.mailbox.send({"rid": rid, "payload": payload})
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if rid in self._pending:
                return self._pending.pop(rid)
            if self.mailbox.poll(0.1):
                msg = self.mailbox.recv()
                if msg.get("rid") == rid:
                    return msg["payload"]
                self._pending[msg["rid"]] = msg["payload"]
        raise TimeoutError("request timed out")

    def serve(self, handler: Callable[[Any], Any], timeout: Optional[float] = None) -> bool:
        try:
            msg = self.mailbox.recv(timeout)
        except TimeoutError:
            return False
        reply = handler(msg["payload"])
        self.mailbox.send({"rid": msg["rid"], "payload": reply})
        return True
