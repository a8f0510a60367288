"""Trainer — hook-driven training orchestration.

Reference: pytorch/rl torchrl/trainers/trainers.py (Trainer:320, hook
registry register_op:1058, train:1400, optim_steps:1653; hooks:
ReplayBufferTrainer:1852, OptimizerHook:1969, LogScalar:2165,
RewardNormalizer:2271, BatchSubSampler:2400, UpdateWeights:2690,
CountFramesLog:2812, TargetNetUpdaterHook:2882, ValueEstimatorHook:2911,
LRSchedulerHook:2961, EarlyStopping:3092, SelectKeys:1807,
ClearCudaCache:2059, LogTiming:2088).
"""
from __future__ import annotations

import os
import pathlib
import time
import warnings
from collections import OrderedDict, defaultdict
from typing import Any, Callable, Dict, List, Optional, Sequence, Tuple, Union

import numpy as np
import torch

from .._utils import logger as rl_logger, timeit
from ..checkpoint._checkpoint import Checkpoint, GlobalRNGState
from ..collectors.collectors import BaseCollector
from ..objectives.common import LossModule
from ..objectives.utils import TargetNetUpdater
from ..tensordict import TensorDict, TensorDictBase

__all__ = [
    "Trainer",
    "TrainerHookBase",
    "OptimizationStepper",
    "DefaultOptimizationStepper",
    "TD3OptimizationStepper",
    "ReplayBufferTrainer",
    "OptimizerHook",
    "ClearCudaCache",
    "LogScalar",
    "LogTiming",
    "RewardNormalizer",
    "BatchSubSampler",
    "UpdateWeights",
    "CountFramesLog",
    "TargetNetUpdaterHook",
    "ValueEstimatorHook",
    "LRSchedulerHook",
    "EarlyStopping",
    "SelectKeys",
    "UTDRHook",
    "LogValidationReward",
]

REPLAY_BUFFER_CLASS_ERROR = "hook expects a replay buffer instance"
_HOOK_STAGES = (
    "batch_process",
    "pre_steps_log",
    "pre_optim_steps",
    "process_optim_batch",
    "post_loss",
    "optimizer",
    "post_optim",
    "post_steps",
    "post_steps_log",
)


class TrainerHookBase:
    """Checkpointable hook (reference trainers.py hook protocol)."""

    def state_dict(self) -> dict:
        return {}

    def load_state_dict(self, sd: dict) -> None:
        pass

    def register(self, trainer: "Trainer", name: str):
        raise NotImplementedError


class Trainer:
    """Generic training loop: ``collector → hooks → optim_steps → log``.

    Hooks attach via :meth:`register_op` at named stages.
    """

    def __init__(
        self,
        *,
        collector: BaseCollector,
        total_frames: int,
        loss_module: Union[LossModule, Callable[[TensorDictBase], TensorDictBase]],
        optimizer: Optional[torch.optim.Optimizer] = None,
        logger=None,
        optim_steps_per_batch: int = 1,
        clip_grad_norm: bool = True,
        clip_norm: Optional[float] = None,
        progress_bar: bool = True,
        seed: Optional[int] = None,
        save_trainer_interval: int = 10000,
        log_interval: int = 10000,
        save_trainer_file: Optional[str] = None,
        num_epochs: int = 1,
        async_collection: bool = False,
    ):
        self.collector = collector
        self.total_frames = total_frames
        self.loss_module = loss_module
        self.optimizer = optimizer
        self.logger = logger
        self.optim_steps_per_batch = optim_steps_per_batch
        self.num_epochs = num_epochs
        self.clip_grad_norm = clip_grad_norm
        self.clip_norm = clip_norm if clip_norm is not None else 1.0
        self.progress_bar = progress_bar
        self.save_trainer_interval = save_trainer_interval
        self.log_interval = log_interval
        self.save_trainer_file = save_trainer_file
        self.async_collection = async_collection
        if seed is not None:
            self.set_seed(seed)

        self._hooks: Dict[str, List[Tuple[Callable, str]]] = {
            s: [] for s in _HOOK_STAGES
        }
        self.collected_frames = 0
        self._optim_count = 0
        self._last_log: Dict[str, float] = {}
        self._log_cache: Dict[str, float] = {}
        self._ckpt = Checkpoint()
        self._ckpt.register(GlobalRNGState(), "rng")
        if hasattr(loss_module, "state_dict"):
            self._ckpt.register(loss_module, "loss_module")
        if optimizer is not None:
            self._ckpt.register(optimizer, "optimizer")
        if hasattr(collector, "state_dict"):
            self._ckpt.register(collector, "collector")

    # ------------------------------------------------------------------ #
    # Hooks
    # ------------------------------------------------------------------ #
    def register_op(self, dest: str, op: Callable, name: Optional[str] = None) -> None:
        if dest not in self._hooks:
            raise ValueError(f"unknown hook stage {dest!r}; stages: {_HOOK_STAGES}")
        self._hooks[dest].append((op, name or getattr(op, "__name__", repr(op))))

    def _run_hooks(self, stage: str, arg=None):
        out = arg
        for op, _name in self._hooks[stage]:
            if out is not None or stage == "process_optim_batch":
                # batch-processing hooks accept an (optionally None)
                # batch — async collection samples with no batch at all
                res = op(out)
            else:
                res = op()
            if res is not None:
                out = res
        return out

    def _log_hooks(self, stage: str, batch):
        for op, _name in self._hooks[stage]:
            res = op(batch)
            if isinstance(res, dict):
                for k, v in res.items():
                    self._log(k, v)

    def _log(self, key: str, value) -> None:
        if isinstance(value, torch.Tensor):
            if value.numel() != 1:
                return
            value = value.item()
        self._log_cache[key] = value
        if self.logger is not None:
            self.logger.log_scalar(key, value, step=self.collected_frames)

    # ------------------------------------------------------------------ #
    def set_seed(self, seed: int) -> None:
        torch.manual_seed(seed)
        np.random.seed(seed % 2**32)
        if hasattr(self.collector, "set_seed"):
            self.collector.set_seed(seed)

    def shutdown(self):
        self.collector.shutdown()

    # ------------------------------------------------------------------ #
    def optim_steps(self, batch: TensorDictBase) -> None:
        """(reference trainers.py:1653)"""
        for _ in range(self.optim_steps_per_batch):
            self._run_hooks("pre_optim_steps")
            sub = self._run_hooks("process_optim_batch", batch)
            losses_td = self.loss_module(sub)
            losses_td = self._run_hooks("post_loss", losses_td) or losses_td
            if self.optimizer is not None:
                loss = sum(
                    v
                    for k, v in losses_td.items()
                    if isinstance(k, str) and k.startswith("loss")
                )
                self.optimizer.zero_grad(set_to_none=True)
                loss.backward()
                if self.clip_grad_norm:
                    gn = torch.nn.utils.clip_grad_norm_(
                        [p for g in self.optimizer.param_groups for p in g["params"]],
                        self.clip_norm,
                    )
                    self._log("grad_norm", gn)
                self.optimizer.step()
            else:
                self._run_hooks("optimizer", losses_td)
            self._run_hooks("post_optim")
            self._optim_count += 1
            for k, v in losses_td.items():
                if isinstance(k, str) and k.startswith("loss"):
                    self._log(k, v)

    def train(self) -> None:
        """(reference trainers.py:1400)"""
        if self.async_collection:
            return self._train_async()
        pbar = None
        if self.progress_bar:
            try:
                import tqdm

                pbar = tqdm.tqdm(total=self.total_frames, unit="frames")
            except ImportError:
                pbar = None
        for batch in self.collector:
            if batch is None:
                # async replay-buffer mode: synthesize a counter batch
                n_frames = self.collector.frames_per_batch
            else:
                n_frames = batch.numel()
                batch = self._run_hooks("batch_process", batch) or batch
            self.collected_frames += n_frames
            if pbar is not None:
                pbar.update(n_frames)
            self._log_hooks("pre_steps_log", batch)
            for _ in range(self.num_epochs):
                self.optim_steps(batch)
            self._run_hooks("post_steps")
            self._log_hooks("post_steps_log", batch)
            if (
                self.save_trainer_file is not None
                and self.collected_frames % self.save_trainer_interval < n_frames
            ):
                self.save_trainer()
            if hasattr(self.collector, "update_policy_weights_"):
                pass  # UpdateWeights hook drives this
            if self.collected_frames >= self.total_frames:
                break
        if pbar is not None:
            pbar.close()
        self.collector.shutdown()

    def _train_async(self) -> None:
        """Collect/train overlap (reference trainers.py:1409-1415): the
        collector streams into its replay buffer from a background
        thread while the learner samples and steps concurrently."""
        import time as _time

        rb = getattr(self.collector, "replay_buffer", None)
        if rb is None:
            # adopt the trainer's own buffer (OffPolicyTrainer builds one)
            rb = getattr(self, "replay_buffer", None)
            if rb is not None:
                self.collector.replay_buffer = rb
                self.collector.extend_buffer = True
        if rb is None:
            raise RuntimeError(
                "async_collection requires a replay buffer (on the collector "
                "or the trainer)"
            )
        self.collector.start()
        min_fill = getattr(rb, "_batch_size", None) or 1
        deadline = _time.monotonic() + 300
        while len(rb) < min_fill:
            if _time.monotonic() > deadline:
                raise TimeoutError("async collection never filled the buffer")
            _time.sleep(0.01)
        while self.collector.frames < self.total_frames:
            self.optim_steps(None)
            self._run_hooks("post_steps")
            self.collected_frames = self.collector.frames
        self.collected_frames = self.collector.frames
        self.collector.async_shutdown()
        self.collector.shutdown()

    # ------------------------------------------------------------------ #
    # Checkpointing (reference trainers.py:919,928)
    # ------------------------------------------------------------------ #
    def state_dict(self) -> dict:
        return {
            "collected_frames": self.collected_frames,
            "optim_count": self._optim_count,
        }

    def load_state_dict(self, sd: dict) -> None:
        self.collected_frames = sd.get("collected_frames", 0)
        self._optim_count = sd.get("optim_count", 0)

    def save_trainer(self, force_save: bool = False) -> None:
        if self.save_trainer_file is None:
            if force_save:
                raise RuntimeError("no save_trainer_file configured")
            return
        self._ckpt.register(self, "trainer_state")
        self._ckpt.save(str(self.save_trainer_file))

    def load_from_file(self, file: str) -> "Trainer":
        self._ckpt.register(self, "trainer_state")
        self._ckpt.load(str(file))
        return self


# --------------------------------------------------------------------------- #
# Hooks
# --------------------------------------------------------------------------- #
class SelectKeys(TrainerHookBase):
    """Keep only selected keys in the batch (reference :1807)."""

    def __init__(self, keys: Sequence):
        self.keys = list(keys)

    def __call__(self, batch: TensorDictBase) -> TensorDictBase:
        return batch.select(*self.keys, strict=False)

    def register(self, trainer: Trainer, name: str = "select_keys"):
        trainer.register_op("batch_process", self, name)


class ReplayBufferTrainer(TrainerHookBase):
    """extend on batch_process, sample on process_optim_batch, priority
    update on post_loss (reference :1852)."""

    def __init__(
        self,
        replay_buffer,
        batch_size: Optional[int] = None,
        memmap: bool = False,
        device=None,
        flatten_tensordicts: bool = True,
        max_dims: Optional[Sequence[int]] = None,
        iterate: bool = False,
    ):
        self.replay_buffer = replay_buffer
        self.batch_size = batch_size
        self.device = device
        self.flatten_tensordicts = flatten_tensordicts

    def extend(self, batch: TensorDictBase) -> TensorDictBase:
        if batch is not None:
            data = batch.reshape(-1) if self.flatten_tensordicts else batch
            self.replay_buffer.extend(data)
        return batch

    def sample(self, batch: Optional[TensorDictBase]) -> TensorDictBase:
        sample = self.replay_buffer.sample(self.batch_size)
        return sample.to(self.device) if self.device is not None else sample

    def update_priority(self, losses_td: TensorDictBase) -> None:
        # priorities were stamped onto the sampled td by the loss forward
        pass

    def register(self, trainer: Trainer, name: str = "replay_buffer"):
        trainer.register_op("batch_process", self.extend, name + "_extend")
        trainer.register_op("process_optim_batch", self.sample, name + "_sample")

    def state_dict(self):
        return {"rb": self.replay_buffer.state_dict()}

    def load_state_dict(self, sd):
        self.replay_buffer.load_state_dict(sd["rb"])


class OptimizerHook(TrainerHookBase):
    """Explicit optimizer stage for multi-optimizer setups
    (reference :1969)."""

    def __init__(self, optimizer: torch.optim.Optimizer, loss_components: Optional[Sequence[str]] = None):
        self.optimizer = optimizer
        self.loss_components = list(loss_components) if loss_components else None

    def __call__(self, losses_td: TensorDictBase) -> TensorDictBase:
        if self.loss_components is None:
            loss = sum(
                v
                for k, v in losses_td.items()
                if isinstance(k, str) and k.startswith("loss")
            )
        else:
            loss = sum(losses_td.get(k) for k in self.loss_components)
        self.optimizer.zero_grad(set_to_none=True)
        loss.backward(retain_graph=self.loss_components is not None)
        self.optimizer.step()
        return losses_td

    def register(self, trainer: Trainer, name: str = "optimizer"):
        trainer.register_op("optimizer", self, name)

    def state_dict(self):
        return {"optimizer": self.optimizer.state_dict()}

    def load_state_dict(self, sd):
        self.optimizer.load_state_dict(sd["optimizer"])


class ClearCudaCache(TrainerHookBase):
    """(reference :2059)"""

    def __init__(self, interval: int = 100):
        self.interval = interval
        self._count = 0

    def __call__(self, *args):
        self._count += 1
        if self._count % self.interval == 0 and torch.cuda.is_available():
            torch.cuda.empty_cache()

    def register(self, trainer: Trainer, name: str = "clear_cuda_cache"):
        trainer.register_op("pre_optim_steps", self, name)


class LogScalar(TrainerHookBase):
    """Log a batch statistic, default episode reward (reference :2165)."""

    def __init__(self, key=("next", "reward"), logname: str = "r_training", log_pbar: bool = False, reduction: str = "mean"):
        self.key = key
        self.logname = logname
        self.reduction = reduction

    def __call__(self, batch: Optional[TensorDictBase]):
        if batch is None:
            return None
        try:
            val = batch.get(self.key)
        except KeyError:
            return None
        if "mask" in batch:
            mask = batch.get("mask")
            while mask.dim() < val.dim():
                mask = mask.unsqueeze(-1)
            val = val[mask.expand_as(val)]
        red = getattr(val.float(), self.reduction)()
        return {self.logname: red.item()}

    def register(self, trainer: Trainer, name: str = "log_reward"):
        trainer.register_op("pre_steps_log", self, name)


class LogTiming(TrainerHookBase):
    """Emit timeit accumulators as scalars (reference :2088)."""

    def __call__(self, *args):
        return {f"timing/{k}": v for k, v in timeit.todict().items()}

    def register(self, trainer: Trainer, name: str = "log_timing"):
        trainer.register_op("post_steps_log", self, name)


class RewardNormalizer(TrainerHookBase):
    """Running-std reward normalization (reference :2271)."""

    def __init__(self, decay: float = 0.999, scale: float = 1.0, eps: float = 1e-4, reward_key=("next", "reward")):
        self.decay = decay
        self.scale = scale
        self.eps = eps
        self.reward_key = reward_key
        self._ssq = 0.0
        self._count = 1e-8

    def update_reward_stats(self, batch: TensorDictBase) -> TensorDictBase:
        r = batch.get(self.reward_key)
        self._ssq = self.decay * self._ssq + float((r * r).sum())
        self._count = self.decay * self._count + r.numel()
        return batch

    def normalize_reward(self, batch: TensorDictBase) -> TensorDictBase:
        r = batch.get(self.reward_key)
        std = max((self._ssq / self._count) ** 0.5, self.eps)
        batch.set(self.reward_key, r / std * self.scale)
        return batch

    def __call__(self, batch):
        self.update_reward_stats(batch)
        return self.normalize_reward(batch)

    def register(self, trainer: Trainer, name: str = "reward_normalizer"):
        trainer.register_op("batch_process", self, name)

    def state_dict(self):
        return {"ssq": self._ssq, "count": self._count}

    def load_state_dict(self, sd):
        self._ssq = sd["ssq"]
        self._count = sd["count"]


class BatchSubSampler(TrainerHookBase):
    """Random sub-batch per optim step (on-policy minibatching)
    (reference :2400)."""

    def __init__(self, batch_size: int, sub_traj_len: int = 0):
        self.batch_size = batch_size
        self.sub_traj_len = sub_traj_len

    def __call__(self, batch: TensorDictBase) -> TensorDictBase:
        flat = batch.reshape(-1)
        n = flat.batch_size[0]
        idx = torch.randint(0, n, (min(self.batch_size, n),), device=flat.device)
        return flat[idx]

    def register(self, trainer: Trainer, name: str = "batch_subsampler"):
        trainer.register_op("process_optim_batch", self, name)


class UpdateWeights(TrainerHookBase):
    """Push learner weights into the collector every N optim steps
    (reference :2690)."""

    def __init__(self, collector, update_weights_interval: int = 1, policy_weights_getter=None):
        self.collector = collector
        self.interval = update_weights_interval
        self._count = 0
        self.policy_weights_getter = policy_weights_getter

    def __call__(self, *args):
        self._count += 1
        if self._count % self.interval == 0:
            w = (
                self.policy_weights_getter()
                if self.policy_weights_getter is not None
                else None
            )
            self.collector.update_policy_weights_(w)

    def register(self, trainer: Trainer, name: str = "update_weights"):
        trainer.register_op("post_steps", self, name)


class CountFramesLog(TrainerHookBase):
    """(reference :2812)"""

    def __init__(self, frame_skip: int = 1):
        self.frame_skip = frame_skip
        self.frame_count = 0

    def __call__(self, batch: Optional[TensorDictBase]):
        if batch is None:
            return None
        if "mask" in batch:
            n = int(batch.get("mask").sum())
        else:
            n = batch.numel()
        self.frame_count += n * self.frame_skip
        return {"n_frames": self.frame_count}

    def register(self, trainer: Trainer, name: str = "count_frames"):
        trainer.register_op("pre_steps_log", self, name)

    def state_dict(self):
        return {"frame_count": self.frame_count}

    def load_state_dict(self, sd):
        self.frame_count = sd["frame_count"]


class TargetNetUpdaterHook(TrainerHookBase):
    """Step a SoftUpdate/HardUpdate after each optim step
    (reference :2882)."""

    def __init__(self, target_updater: TargetNetUpdater):
        self.updater = target_updater

    def __call__(self, *args):
        self.updater.step()

    def register(self, trainer: Trainer, name: str = "target_net_updater"):
        trainer.register_op("post_optim", self, name)


class ValueEstimatorHook(TrainerHookBase):
    """Run the loss's value estimator (GAE etc.) on the fresh batch
    (reference :2911)."""

    def __init__(self, value_estimator):
        self.value_estimator = value_estimator

    def __call__(self, batch: TensorDictBase) -> TensorDictBase:
        with torch.no_grad():
            self.value_estimator(batch)
        return batch

    def register(self, trainer: Trainer, name: str = "value_estimator"):
        trainer.register_op("batch_process", self, name)


class LRSchedulerHook(TrainerHookBase):
    """(reference :2961)"""

    def __init__(self, scheduler):
        self.scheduler = scheduler

    def __call__(self, *args):
        self.scheduler.step()

    def register(self, trainer: Trainer, name: str = "lr_scheduler"):
        trainer.register_op("post_steps", self, name)

    def state_dict(self):
        return {"scheduler": self.scheduler.state_dict()}

    def load_state_dict(self, sd):
        self.scheduler.load_state_dict(sd["scheduler"])


class EarlyStopping(TrainerHookBase):
    """Stop when a logged metric stalls (reference :3092)."""

    def __init__(self, trainer: Trainer, metric: str = "r_training", patience: int = 10, min_delta: float = 0.0):
        self.trainer = trainer
        self.metric = metric
        self.patience = patience
        self.min_delta = min_delta
        self._best = -float("inf")
        self._bad = 0

    def __call__(self, *args):
        val = self.trainer._log_cache.get(self.metric)
        if val is None:
            return
        if val > self._best + self.min_delta:
            self._best = val
            self._bad = 0
        else:
            self._bad += 1
        if self._bad >= self.patience:
            self.trainer.total_frames = 0  # forces loop exit

    def register(self, trainer: Trainer, name: str = "early_stopping"):
        trainer.register_op("post_steps", self, name)


class UTDRHook(TrainerHookBase):
    """Log the update-to-data ratio (reference :3024)."""

    def __init__(self, trainer: Trainer):
        self.trainer = trainer

    def __call__(self, batch=None):
        frames = max(1, self.trainer.collected_frames)
        return {"utd_ratio": self.trainer._optim_count / frames}

    def register(self, trainer: Trainer, name: str = "utdr"):
        trainer.register_op("post_steps_log", self, name)


class LogValidationReward(TrainerHookBase):
    """Periodic eval-env rollouts logged as validation reward
    (reference :2530)."""

    def __init__(
        self,
        *,
        record_interval: int,
        record_frames: int,
        environment,
        policy_exploration=None,
        log_keyname: str = "r_evaluation",
    ):
        self.record_interval = record_interval
        self.record_frames = record_frames
        self.environment = environment
        self.policy_exploration = policy_exploration
        self.log_keyname = log_keyname
        self._count = 0

    def __call__(self, batch=None):
        self._count += 1
        if self._count % self.record_interval:
            return None
        from ..envs.utils import ExplorationType, set_exploration_type

        with set_exploration_type(ExplorationType.DETERMINISTIC), torch.no_grad():
            rollout = self.environment.rollout(
                self.record_frames,
                policy=self.policy_exploration,
                break_when_any_done=True,
            )
        return {self.log_keyname: rollout.get(("next", "reward")).sum().item()}

    def register(self, trainer: Trainer, name: str = "log_validation_reward"):
        trainer.register_op("post_steps_log", self, name)


class OptimizationStepper(TrainerHookBase):
    """Encapsulates one optimization step per sub-batch (reference
    trainers.py:200) — override ``step`` for algorithms needing multiple
    optimizers or delayed updates."""

    def __init__(self):
        self.trainer = None

    def register(self, trainer, name: str = "optimization_stepper"):
        self.trainer = trainer
        trainer.register_op("optimizer", self, name)

    def __call__(self, losses_td):
        return self.step(losses_td)

    def step(self, losses_td):
        raise NotImplementedError

    def state_dict(self):
        return {}

    def load_state_dict(self, sd):
        pass


class DefaultOptimizationStepper(OptimizationStepper):
    """Single-optimizer step over the sum of ``loss_*`` entries
    (reference trainers.py:249)."""

    def __init__(self, optimizer, clip_grad_norm: Optional[float] = None):
        super().__init__()
        self.optimizer = optimizer
        self.clip_grad_norm = clip_grad_norm

    def step(self, losses_td):
        loss = sum(
            v for k, v in losses_td.items()
            if isinstance(k, str) and k.startswith("loss")
        )
        self.optimizer.zero_grad(set_to_none=True)
        loss.backward()
        if self.clip_grad_norm is not None:
            torch.nn.utils.clip_grad_norm_(
                [p for g in self.optimizer.param_groups for p in g["params"]],
                self.clip_grad_norm,
            )
        self.optimizer.step()
        return losses_td


class TD3OptimizationStepper(OptimizationStepper):
    """Delayed-actor stepper (reference trainers.py TD3 variant): the
    critic optimizer steps every call; the actor (and target-net
    updater) only every ``policy_delay`` calls."""

    def __init__(self, critic_optimizer, actor_optimizer, *, policy_delay: int = 2,
                 target_updater=None):
        super().__init__()
        self.critic_optimizer = critic_optimizer
        self.actor_optimizer = actor_optimizer
        self.policy_delay = policy_delay
        self.target_updater = target_updater
        self._calls = 0

    def step(self, losses_td):
        critic_loss = losses_td.get("loss_qvalue", None)
        if critic_loss is None:
            critic_loss = losses_td.get("loss_value")
        self.critic_optimizer.zero_grad(set_to_none=True)
        critic_loss.backward(retain_graph=True)
        self.critic_optimizer.step()
        self._calls += 1
        if self._calls % self.policy_delay == 0:
            actor_loss = losses_td.get("loss_actor")
            self.actor_optimizer.zero_grad(set_to_none=True)
            actor_loss.backward()
            self.actor_optimizer.step()
            if self.target_updater is not None:
                self.target_updater.step()
        return losses_td


def mask_batch(batch):
    """Batch-process hook: drop padded events using the collector mask
    (reference trainers.py:2379) — if ``("collector", "mask")`` exists,
    index the batch down to valid rows."""
    if ("collector", "mask") in batch.keys(True):
        return batch[batch.get(("collector", "mask"))]
    return batch


__all__.append("mask_batch")
