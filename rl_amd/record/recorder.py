"""Recorders: VideoRecorder / TensorDictRecorder transforms, LoggerMonitor.

Reference: pytorch/rl torchrl/record/recorder.py (VideoRecorder:43,
TensorDictRecorder:433, PixelRenderTransform:501) and
record/loggers/monitoring.py:128 (LoggerMonitor, Every:31).
"""
from __future__ import annotations

import time
from typing import Callable, List, Optional, Sequence

import torch

from ..envs.transforms import Transform
from ..tensordict import TensorDictBase

__all__ = ["VideoRecorder", "TensorDictRecorder", "PixelRenderTransform", "LoggerMonitor", "Every"]


class VideoRecorder(Transform):
    """Accumulate pixel frames during rollouts and flush them to a logger
    (reference recorder.py:43)."""

    def __init__(
        self,
        logger,
        tag: str = "rollout_video",
        in_keys: Sequence = ("pixels",),
        skip: int = 2,
        center_crop: Optional[int] = None,
        fps: int = 30,
    ):
        super().__init__(in_keys=list(in_keys), out_keys=list(in_keys))
        self.logger = logger
        self.tag = tag
        self.skip = skip
        self.center_crop = center_crop
        self.fps = fps
        self._frames: List[torch.Tensor] = []
        self._count = 0
        self._iter = 0

    def _apply_transform(self, obs: torch.Tensor) -> torch.Tensor:
        self._count += 1
        if self._count % self.skip == 0:
            frame = obs.detach().cpu()
            if frame.dtype.is_floating_point:
                frame = (frame.clamp(0, 1) * 255).to(torch.uint8)
            if self.center_crop:
                c = self.center_crop
                H, W = frame.shape[-2:]
                frame = frame[..., (H - c) // 2 : (H + c) // 2, (W - c) // 2 : (W + c) // 2]
            self._frames.append(frame)
        return obs

    def dump(self, suffix: Optional[str] = None) -> None:
        if not self._frames:
            return
        video = torch.stack(self._frames, 0)
        if video.dim() == 4:  # [T, C, H, W] → [1, T, C, H, W]
            video = video.unsqueeze(0)
        tag = f"{self.tag}_{suffix}" if suffix else self.tag
        if self.logger is not None:
            self.logger.log_video(tag, video, step=self._iter, fps=self.fps)
        self._frames = []
        self._iter += 1

    def _reset(self, td, td_reset):
        return td_reset


class TensorDictRecorder(Transform):
    """Accumulate whole TensorDicts during rollout (reference :433)."""

    def __init__(self, out_file_base: str, skip: int = 2, in_keys: Sequence = ()):
        super().__init__(in_keys=list(in_keys), out_keys=list(in_keys))
        self.out_file_base = out_file_base
        self.skip = skip
        self._tds: List = []
        self._count = 0
        self._iter = 0

    def _call(self, td: TensorDictBase) -> TensorDictBase:
        self._count += 1
        if self._count % self.skip == 0:
            keep = td.select(*self.in_keys, strict=False) if self.in_keys else td
            self._tds.append(keep.clone().cpu())
        return td

    def dump(self) -> None:
        if not self._tds:
            return
        from ..tensordict import stack as td_stack

        torch.save(
            td_stack(self._tds, 0), f"{self.out_file_base}_{self._iter}.pt"
        )
        self._tds = []
        self._iter += 1


class PixelRenderTransform(Transform):
    """Call the env's render() each step and store frames under ``pixels``
    (reference :501)."""

    def __init__(self, out_keys: Sequence = ("pixels",), render_fn: Optional[Callable] = None):
        super().__init__(in_keys=[], out_keys=list(out_keys))
        self.render_fn = render_fn

    def _step(self, td, next_td):
        parent = self.parent
        fn = self.render_fn
        if fn is None and parent is not None and hasattr(parent.base_env, "render"):
            fn = parent.base_env.render
        if fn is not None:
            frame = fn()
            if frame is not None:
                next_td.set(self.out_keys[0], torch.as_tensor(frame))
        return next_td


class Every:
    """Rate limiter (reference monitoring.py:31)."""

    def __init__(self, interval: float):
        self.interval = interval
        self._last = 0.0

    def __call__(self) -> bool:
        now = time.monotonic()
        if now - self._last >= self.interval:
            self._last = now
            return True
        return False


class LoggerMonitor:
    """Rate-limited system-stat logging (reference monitoring.py:128):
    GPU memory/utilization + host RSS to the experiment logger."""

    def __init__(self, logger, interval: float = 30.0, prefix: str = "sys"):
        self.logger = logger
        self.every = Every(interval)
        self.prefix = prefix

    def step(self, global_step: Optional[int] = None) -> None:
        if not self.every():
            return
        stats = {}
        if torch.cuda.is_available():
            for i in range(torch.cuda.device_count()):
                stats[f"{self.prefix}/gpu{i}_mem_alloc_gb"] = (
                    torch.cuda.memory_allocated(i) / 1e9
                )
                stats[f"{self.prefix}/gpu{i}_mem_reserved_gb"] = (
                    torch.cuda.memory_reserved(i) / 1e9
                )
        try:
            import psutil

            p = psutil.Process()
            stats[f"{self.prefix}/host_rss_gb"] = p.memory_info().rss / 1e9
            stats[f"{self.prefix}/cpu_percent"] = p.cpu_percent()
        except ImportError:
            pass
        for k, v in stats.items():
            self.logger.log_scalar(k, v, step=global_step)
