"""Config system — dataclass configs with ``_target_`` instantiation.

Reference: pytorch/rl torchrl/trainers/algorithms/configs/ (15 files of
Hydra dataclasses; ConfigBase common.py:18).  Hydra isn't in the image,
so rl_amd ships the same pattern self-contained: dataclass configs, a
YAML loader, and :func:`instantiate` resolving ``_target_`` dotted paths
recursively — ``instantiate(load_config("ppo.yaml"))`` builds a full
training job.
"""
from __future__ import annotations

import dataclasses
import importlib
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence, Union

import yaml

__all__ = [
    "ConfigBase",
    "instantiate",
    "load_config",
    "save_config",
    "EnvConfig",
    "MLPConfig",
    "ActorConfig",
    "CriticConfig",
    "CollectorConfig",
    "ReplayBufferConfig",
    "LossConfig",
    "OptimizerConfig",
    "LoggerConfig",
    "TrainerConfig",
    "PPOTrainerConfig",
    "SACTrainerConfig",
    "DQNTrainerConfig",
]


@dataclass
class ConfigBase:
    """Base config (reference configs/common.py:18)."""

    def to_dict(self) -> dict:
        return dataclasses.asdict(self)


def _locate(path: str):
    mod, _, name = path.rpartition(".")
    if not mod:
        raise ValueError(f"_target_ {path!r} must be a dotted path")
    return getattr(importlib.import_module(mod), name)


def instantiate(cfg: Any, **overrides):
    """Recursively build objects from configs.

    dict/dataclass with ``_target_`` → call the target with instantiated
    kwargs; lists instantiate element-wise; everything else passes
    through.  ``_partial_: true`` returns a functools.partial.
    """
    import functools

    if dataclasses.is_dataclass(cfg) and not isinstance(cfg, type):
        cfg = dataclasses.asdict(cfg)
    if isinstance(cfg, (list, tuple)):
        return type(cfg)(instantiate(c) for c in cfg)
    if not isinstance(cfg, dict):
        return cfg
    cfg = dict(cfg)
    cfg.update(overrides)
    target = cfg.pop("_target_", None)
    partial = cfg.pop("_partial_", False)
    kwargs = {k: instantiate(v) for k, v in cfg.items()}
    if target is None:
        return kwargs
    fn = _locate(target)
    if partial:
        return functools.partial(fn, **kwargs)
    return fn(**kwargs)


def load_config(path: str) -> dict:
    with open(path) as f:
        return yaml.safe_load(f)


def save_config(cfg: Any, path: str) -> None:
    if dataclasses.is_dataclass(cfg) and not isinstance(cfg, type):
        cfg = dataclasses.asdict(cfg)
    with open(path, "w") as f:
        yaml.safe_dump(cfg, f)


# --------------------------------------------------------------------------- #
# Canonical component configs (reference configs/envs.py, modules.py, ...)
# --------------------------------------------------------------------------- #
@dataclass
class EnvConfig(ConfigBase):
    _target_: str = "rl_amd.envs.PendulumEnv"
    batch_size: Optional[List[int]] = None
    device: Optional[str] = None


@dataclass
class MLPConfig(ConfigBase):
    _target_: str = "rl_amd.modules.MLP"
    in_features: Optional[int] = None
    out_features: int = 1
    num_cells: List[int] = field(default_factory=lambda: [64, 64])
    activation_class: str = "torch.nn.Tanh"


@dataclass
class ActorConfig(ConfigBase):
    obs_dim: int = 3
    act_dim: int = 1
    hidden: List[int] = field(default_factory=lambda: [64, 64])


@dataclass
class CriticConfig(ConfigBase):
    obs_dim: int = 3
    hidden: List[int] = field(default_factory=lambda: [64, 64])


@dataclass
class CollectorConfig(ConfigBase):
    _target_: str = "rl_amd.collectors.Collector"
    frames_per_batch: int = 1000
    total_frames: int = 1_000_000
    init_random_frames: int = 0


@dataclass
class ReplayBufferConfig(ConfigBase):
    size: int = 1_000_000
    batch_size: int = 256
    prioritized: bool = False
    alpha: float = 0.7
    beta: float = 0.5
    device: Optional[str] = None


@dataclass
class LossConfig(ConfigBase):
    gamma: float = 0.99


@dataclass
class OptimizerConfig(ConfigBase):
    _target_: str = "torch.optim.Adam"
    _partial_: bool = True
    lr: float = 3e-4


@dataclass
class LoggerConfig(ConfigBase):
    backend: str = "csv"
    exp_name: str = "rl_amd_run"
    log_dir: str = "logs"


@dataclass
class TrainerConfig(ConfigBase):
    total_frames: int = 1_000_000
    seed: Optional[int] = None
    clip_grad_norm: bool = True
    clip_norm: float = 1.0
    progress_bar: bool = True


@dataclass
class PPOTrainerConfig(TrainerConfig):
    gamma: float = 0.99
    lmbda: float = 0.95
    clip_epsilon: float = 0.2
    entropy_coeff: float = 0.01
    critic_coeff: float = 1.0
    lr: float = 3e-4
    minibatch_size: int = 256
    num_epochs: int = 4


@dataclass
class SACTrainerConfig(TrainerConfig):
    gamma: float = 0.99
    tau: float = 0.005
    lr: float = 3e-4
    batch_size: int = 256
    buffer_size: int = 1_000_000
    prioritized: bool = False


@dataclass
class DQNTrainerConfig(TrainerConfig):
    gamma: float = 0.99
    tau: float = 0.02
    lr: float = 1e-3
    batch_size: int = 256
    buffer_size: int = 100_000
    eps_init: float = 1.0
    eps_end: float = 0.05
