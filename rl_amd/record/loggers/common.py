"""Experiment loggers.

Reference: pytorch/rl torchrl/record/loggers/ (Logger ABC common.py:186,
csv.py:131, tensorboard.py:20, wandb.py:54, mlflow.py:28,
utils.py get_logger/generate_exp_name).  CSV is the always-available
backend; TensorBoard/W&B/MLflow are import-gated.
"""
from __future__ import annotations

import csv
import datetime
import os
import uuid
from typing import Any, Dict, Optional, Sequence, Union

import torch

__all__ = [
    "Logger",
    "CSVLogger",
    "TensorboardLogger",
    "WandbLogger",
    "MLFlowLogger",
    "get_logger",
    "generate_exp_name",
]


class Logger:
    """ABC: log_scalar / log_video / log_hparams / log_histogram."""

    def __init__(self, exp_name: str, log_dir: Optional[str] = None):
        self.exp_name = exp_name
        self.log_dir = log_dir
        self.experiment = self._create_experiment()

    def _create_experiment(self):
        return None

    def log_scalar(self, name: str, value: float, step: Optional[int] = None) -> None:
        raise NotImplementedError

    def log_video(self, name: str, video, step: Optional[int] = None, **kwargs) -> None:
        raise NotImplementedError

    def log_hparams(self, cfg: Dict[str, Any]) -> None:
        raise NotImplementedError

    def log_histogram(self, name: str, data, **kwargs) -> None:
        raise NotImplementedError

    def __repr__(self):
        return f"{type(self).__name__}(exp_name={self.exp_name})"


class CSVLogger(Logger):
    """File-system logger: scalars to csv, videos to .pt/.mp4, hparams to
    txt (reference csv.py:131)."""

    def __init__(self, exp_name: str, log_dir: Optional[str] = None, video_format: str = "pt", video_fps: int = 30):
        if log_dir is None:
            log_dir = "csv_logs"
        self.video_format = video_format
        self.video_fps = video_fps
        super().__init__(exp_name, log_dir)
        self._files: Dict[str, Any] = {}

    def _create_experiment(self):
        self.experiment_dir = os.path.join(self.log_dir or "csv_logs", self.exp_name)
        os.makedirs(os.path.join(self.experiment_dir, "scalars"), exist_ok=True)
        os.makedirs(os.path.join(self.experiment_dir, "videos"), exist_ok=True)
        os.makedirs(os.path.join(self.experiment_dir, "texts"), exist_ok=True)
        return self.experiment_dir

    def log_scalar(self, name: str, value, step: Optional[int] = None) -> None:
        safe = name.replace("/", ".")
        path = os.path.join(self.experiment_dir, "scalars", f"{safe}.csv")
        if isinstance(value, torch.Tensor):
            value = value.item()
        with open(path, "a", newline="") as f:
            csv.writer(f).writerow([step if step is not None else "", value])

    def log_video(self, name: str, video, step: Optional[int] = None, video_format: str = "gif", **kwargs) -> None:
        safe = name.replace("/", ".")
        suffix = f"_{step}" if step is not None else ""
        base = os.path.join(self.experiment_dir, "videos", f"{safe}{suffix}")
        if video_format == "gif":
            from ...render.video import write_gif

            v = video
            if hasattr(v, "dim") and v.dim() == 5:  # [B, T, C, H, W]
                v = v[0]
            write_gif(v, base + ".gif", fps=int(kwargs.get("fps", 30)))
        else:
            torch.save(video, base + ".pt")

    def log_hparams(self, cfg: Dict[str, Any]) -> None:
        path = os.path.join(self.experiment_dir, "texts", "hparams.txt")
        with open(path, "a") as f:
            for k, v in sorted(dict(cfg).items()):
                f.write(f"{k}: {v}\n")

    def log_histogram(self, name: str, data, **kwargs) -> None:
        safe = name.replace("/", ".")
        path = os.path.join(self.experiment_dir, "scalars", f"{safe}_hist.pt")
        torch.save(torch.as_tensor(data), path)

    def print_log_dir(self):
        return self.experiment_dir


class TensorboardLogger(Logger):
    """torch.utils.tensorboard wrapper (reference tensorboard.py:20);
    requires the tensorboard package."""

    def __init__(self, exp_name: str, log_dir: str = "tb_logs"):
        super().__init__(exp_name, log_dir)

    def _create_experiment(self):
        from torch.utils.tensorboard import SummaryWriter  # raises if absent

        return SummaryWriter(log_dir=os.path.join(self.log_dir, self.exp_name))

    def log_scalar(self, name, value, step=None):
        self.experiment.add_scalar(name, value, global_step=step)

    def log_video(self, name, video, step=None, **kwargs):
        self.experiment.add_video(name, video, global_step=step, fps=kwargs.get("fps", 30))

    def log_hparams(self, cfg):
        self.experiment.add_hparams(dict(cfg), {})

    def log_histogram(self, name, data, **kwargs):
        self.experiment.add_histogram(name, data, global_step=kwargs.get("step"))


class WandbLogger(Logger):
    """Weights & Biases (reference wandb.py:54); import-gated."""

    def __init__(self, exp_name: str, offline: bool = True, project: str = "rl_amd", **kwargs):
        self.offline = offline
        self.project = project
        self._wandb_kwargs = kwargs
        super().__init__(exp_name, kwargs.get("save_dir"))

    def _create_experiment(self):
        import wandb  # raises if absent

        mode = "offline" if self.offline else "online"
        return wandb.init(
            project=self.project, name=self.exp_name, mode=mode, **self._wandb_kwargs
        )

    def log_scalar(self, name, value, step=None):
        self.experiment.log({name: value}, step=step)

    def log_video(self, name, video, step=None, **kwargs):
        import wandb

        self.experiment.log({name: wandb.Video(video, fps=kwargs.get("fps", 30))}, step=step)

    def log_hparams(self, cfg):
        self.experiment.config.update(dict(cfg), allow_val_change=True)

    def log_histogram(self, name, data, **kwargs):
        import wandb

        self.experiment.log({name: wandb.Histogram(data)})


class MLFlowLogger(Logger):
    """MLflow (reference mlflow.py:28); import-gated."""

    def __init__(self, exp_name: str, tracking_uri: Optional[str] = None, **kwargs):
        self.tracking_uri = tracking_uri
        super().__init__(exp_name, tracking_uri)

    def _create_experiment(self):
        import mlflow  # raises if absent

        if self.tracking_uri:
            mlflow.set_tracking_uri(self.tracking_uri)
        mlflow.set_experiment(self.exp_name)
        return mlflow

    def log_scalar(self, name, value, step=None):
        self.experiment.log_metric(name.replace("/", "_"), value, step=step)

    def log_hparams(self, cfg):
        self.experiment.log_params(dict(cfg))

    def log_video(self, name, video, step=None, **kwargs):
        raise NotImplementedError("mlflow has no native video logging")

    def log_histogram(self, name, data, **kwargs):
        raise NotImplementedError


def generate_exp_name(model_name: str, experiment_name: str) -> str:
    """Unique run name (reference utils.py)."""
    ts = datetime.datetime.now().strftime("%Y_%m_%d-%H_%M_%S")
    return f"{model_name}_{experiment_name}_{ts}_{str(uuid.uuid4())[:8]}"


def get_logger(
    logger_type: Optional[str],
    logger_name: str,
    experiment_name: str,
    **kwargs,
) -> Optional[Logger]:
    if logger_type in (None, "", "none"):
        return None
    if logger_type == "csv":
        return CSVLogger(experiment_name, log_dir=logger_name, **kwargs)
    if logger_type == "tensorboard":
        return TensorboardLogger(experiment_name, log_dir=logger_name)
    if logger_type == "wandb":
        return WandbLogger(experiment_name, **kwargs)
    if logger_type == "mlflow":
        return MLFlowLogger(experiment_name, **kwargs)
    raise NotImplementedError(f"unknown logger type {logger_type}")


class TrackioLogger(Logger):
    """trackio experiment logger (reference record/loggers/trackio.py:21)
    — gated on the `trackio` package."""

    def __init__(self, exp_name: str, project: str = "rl_amd", **kwargs):
        import importlib.util

        if importlib.util.find_spec("trackio") is None:
            raise ImportError(
                "TrackioLogger requires the `trackio` package, which is not "
                "installed in this image. Use CSVLogger instead."
            )
        import trackio

        super().__init__(exp_name)
        self.run = trackio.init(project=project, name=exp_name, **kwargs)

    def log_scalar(self, name, value, step=None):
        import trackio

        trackio.log({name: value}, step=step)

    def log_hparams(self, cfg):
        self.run.config.update(dict(cfg))

    def log_video(self, name, video, step=None, **kw):
        import trackio

        trackio.log({name: video}, step=step)


class ProcessLogger(Logger):
    """Logger service living in a dedicated process (reference
    record/loggers/process.py:132): log_* calls are enqueued through a
    multiprocessing queue and drained by a child that owns the concrete
    logger — workers never block on IO.  Only the owner can flush/stop."""

    def __init__(self, logger_factory, *, exp_name: str = "rl_amd", log_dir: str = None):
        import multiprocessing as mp

        super().__init__(exp_name=exp_name, log_dir=log_dir or "./logs")
        ctx = mp.get_context("spawn")
        self._queue = ctx.Queue()
        self._proc = ctx.Process(
            target=self._serve, args=(self._queue, logger_factory), daemon=True
        )
        self._proc.start()

    @staticmethod
    def _serve(queue, logger_factory):
        inner = logger_factory()
        while True:
            item = queue.get()
            if item is None:
                break
            method, args, kwargs = item
            getattr(inner, method)(*args, **kwargs)

    def log_scalar(self, name, value, step=None):
        self._queue.put(("log_scalar", (name, value), {"step": step}))

    def log_video(self, name, video, step=None, **kwargs):
        self._queue.put(("log_video", (name, video), {"step": step, **kwargs}))

    def log_hparams(self, cfg):
        self._queue.put(("log_hparams", (cfg,), {}))

    def log_histogram(self, name, data, **kwargs):
        self._queue.put(("log_histogram", (name, data), kwargs))

    def __repr__(self):
        return f"ProcessLogger(exp_name={self.exp_name})"

    def stop(self):
        self._queue.put(None)
        self._proc.join(timeout=10)


class _RayLoggerActor:
    """Actor body hosting the real logger (decorated at runtime)."""

    def __init__(self, logger_factory):
        self.logger = logger_factory()

    def log_scalar(self, name, value, step=None):
        self.logger.log_scalar(name, value, step=step)

    def log_hparams(self, cfg):
        self.logger.log_hparams(cfg)

    def log_video(self, name, video, step=None, **kw):
        self.logger.log_video(name, video, step=step, **kw)


class RayLogger(Logger):
    """Ray-actor-hosted logger (reference record/loggers/ray.py): every
    worker logs through one actor so multi-node runs share a single
    experiment sink — gated on `ray` (ProcessLogger covers the
    out-of-process pattern locally)."""

    def __init__(self, logger_factory, *, remote_configs=None):
        import importlib.util

        if importlib.util.find_spec("ray") is None:
            raise ImportError(
                "RayLogger requires the `ray` package, which is not installed "
                "in this image. Use ProcessLogger instead."
            )
        import ray

        if not ray.is_initialized():
            ray.init(ignore_reinit_error=True)
        Actor = ray.remote(**(remote_configs or {"num_cpus": 0.1}))(_RayLoggerActor)
        self._actor = Actor.remote(logger_factory)
        self.exp_name = "ray"

    def log_scalar(self, name, value, step=None):
        self._actor.log_scalar.remote(name, value, step)

    def log_hparams(self, cfg):
        self._actor.log_hparams.remote(cfg)

    def log_video(self, name, video, step=None, **kw):
        self._actor.log_video.remote(name, video, step, **kw)


__all__ += ["ProcessLogger", "RayLogger", "TrackioLogger"]
