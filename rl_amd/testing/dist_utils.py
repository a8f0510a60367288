"""Distributed-test helpers (reference torchrl/testing/dist_utils.py):
process-leak detection for multiprocess suites."""
from __future__ import annotations

import os
from typing import List, Set

__all__ = ["snapshot_python_processes", "assert_no_new_python_processes"]


def snapshot_python_processes() -> Set[int]:
    """PIDs of live python children of this process."""
    try:
        import psutil

        me = psutil.Process(os.getpid())
        return {c.pid for c in me.children(recursive=True) if c.is_running()}
    except ImportError:
        return set()


def assert_no_new_python_processes(before: Set[int]) -> None:
    """Raise if python child processes leaked versus the snapshot."""
    after = snapshot_python_processes()
    leaked = after - before
    assert not leaked, f"leaked child processes: {sorted(leaked)}"
