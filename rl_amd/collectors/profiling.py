"""Collector profiling: torch.profiler integration + rocprof marker hooks.

Reference: pytorch/rl torchrl/collectors/_base.py:33 (ProfileConfig),
:144 (_ProfilerHook), :468 (enable_profile), :96 (get_save_path),
:132 (should_profile_worker).  On ROCm the chrome traces come out of
torch.profiler's kineto/roctracer backend; for kernel-level detail run
the workload under rocprofv3 instead (profiles/README.md has the recipe).
"""
from __future__ import annotations

import dataclasses
import os
from typing import List, Optional, Sequence

import torch

__all__ = ["ProfileConfig", "ProfilerHook", "enable_profile"]


@dataclasses.dataclass
class ProfileConfig:
    """What/when/where to profile (reference _base.py:33)."""

    save_dir: str = "profiler_traces"
    wait: int = 2
    warmup: int = 2
    active: int = 4
    repeat: int = 1
    with_stack: bool = False
    profile_memory: bool = False
    record_shapes: bool = False
    workers: Optional[Sequence[int]] = None  # None → main process only

    def should_profile_worker(self, worker_idx: Optional[int]) -> bool:
        """(reference :132)"""
        if worker_idx is None:
            return self.workers is None
        return self.workers is not None and worker_idx in self.workers

    def get_save_path(self, worker_idx: Optional[int] = None) -> str:
        """(reference :96)"""
        os.makedirs(self.save_dir, exist_ok=True)
        suffix = "main" if worker_idx is None else f"worker{worker_idx}"
        return os.path.join(self.save_dir, f"collector_trace_{suffix}")


class ProfilerHook:
    """Owns a torch.profiler.profile and steps it once per collector
    iteration (reference _ProfilerHook:144)."""

    def __init__(self, config: ProfileConfig, worker_idx: Optional[int] = None):
        self.config = config
        self.worker_idx = worker_idx
        self._prof: Optional[torch.profiler.profile] = None

    def start(self) -> None:
        activities = [torch.profiler.ProfilerActivity.CPU]
        if torch.cuda.is_available():
            activities.append(torch.profiler.ProfilerActivity.CUDA)
        path = self.config.get_save_path(self.worker_idx)

        def on_ready(prof):
            prof.export_chrome_trace(f"{path}_{prof.step_num}.json")

        self._prof = torch.profiler.profile(
            activities=activities,
            schedule=torch.profiler.schedule(
                wait=self.config.wait,
                warmup=self.config.warmup,
                active=self.config.active,
                repeat=self.config.repeat,
            ),
            on_trace_ready=on_ready,
            with_stack=self.config.with_stack,
            profile_memory=self.config.profile_memory,
            record_shapes=self.config.record_shapes,
        )
        self._prof.__enter__()

    def step(self) -> None:
        if self._prof is not None:
            self._prof.step()

    def stop(self) -> None:
        if self._prof is not None:
            self._prof.__exit__(None, None, None)
            self._prof = None


def enable_profile(collector, config: Optional[ProfileConfig] = None) -> ProfilerHook:
    """Attach a profiler to a collector's iteration loop
    (reference enable_profile:468): wraps ``iterator`` so every yielded
    batch advances the profiler schedule."""
    config = config or ProfileConfig()
    hook = ProfilerHook(config)
    orig_iterator = collector.iterator

    def iterator():
        hook.start()
        try:
            for batch in orig_iterator():
                yield batch
                hook.step()
        finally:
            hook.stop()

    collector.iterator = iterator
    collector._profiler_hook = hook
    return hook
