"""Algorithm trainers — thin assemblies over Trainer.

Reference: pytorch/rl torchrl/trainers/algorithms/ (PPOTrainer ppo.py:11,
SACTrainer, DQNTrainer, TD3Trainer, DDPGTrainer, A2CTrainer,
OnPolicyTrainer on_policy.py).  Each subclass wires collector + loss +
optimizer + the canonical hook set for its algorithm family.
"""
from __future__ import annotations

from typing import Optional

import torch

from ..collectors.collectors import Collector
from ..data import LazyTensorStorage, TensorDictPrioritizedReplayBuffer, TensorDictReplayBuffer
from ..objectives import (
    A2CLoss,
    ClipPPOLoss,
    CQLLoss,
    DDPGLoss,
    DQNLoss,
    IQLLoss,
    ReinforceLoss,
    SACLoss,
    SoftUpdate,
    TD3Loss,
)
from ..objectives.value.advantages import GAE
from .trainers import (
    BatchSubSampler,
    CountFramesLog,
    LogScalar,
    ReplayBufferTrainer,
    TargetNetUpdaterHook,
    Trainer,
    UpdateWeights,
    ValueEstimatorHook,
)

__all__ = [
    "OnPolicyTrainer",
    "OffPolicyTrainer",
    "PPOTrainer",
    "A2CTrainer",
    "SACTrainer",
    "TD3Trainer",
    "DDPGTrainer",
    "DQNTrainer",
]


class OnPolicyTrainer(Trainer):
    """Collect → advantage → minibatch epochs (reference on_policy.py)."""

    def __init__(
        self,
        *,
        collector,
        loss_module,
        optimizer,
        total_frames: int,
        value_estimator=None,
        minibatch_size: Optional[int] = None,
        num_epochs: int = 4,
        logger=None,
        **kwargs,
    ):
        super().__init__(
            collector=collector,
            total_frames=total_frames,
            loss_module=loss_module,
            optimizer=optimizer,
            logger=logger,
            num_epochs=num_epochs,
            **kwargs,
        )
        if value_estimator is not None:
            ValueEstimatorHook(value_estimator).register(self, "value_estimator")
        if minibatch_size is not None:
            BatchSubSampler(minibatch_size).register(self)
        LogScalar().register(self)
        CountFramesLog().register(self)
        UpdateWeights(collector).register(self)


class PPOTrainer(OnPolicyTrainer):
    """(reference algorithms/ppo.py:11)

    ``graphed=True`` swaps the hook-driven train loop for
    :class:`~rl_amd.trainers.GraphedPPO` — the whole PPO iteration
    (collector fast-path rollout + GAE + minibatch updates + optimizer)
    captured as one hipGraph on GPU (the exact configuration the
    flagship bench measures at 26M frames/s)."""

    def __init__(self, *, actor, critic, collector, total_frames, lr: float = 3e-4,
                 gamma: float = 0.99, lmbda: float = 0.95, clip_epsilon: float = 0.2,
                 entropy_coeff: float = 0.01, critic_coeff: float = 1.0,
                 minibatch_size: int = 256, num_epochs: int = 4, logger=None,
                 graphed: bool = False, **kwargs):
        loss = ClipPPOLoss(
            actor,
            critic,
            clip_epsilon=clip_epsilon,
            entropy_coeff=entropy_coeff,
            critic_coeff=critic_coeff,
            normalize_advantage=True,
        )
        gae = GAE(gamma=gamma, lmbda=lmbda, value_network=critic)
        optim = torch.optim.Adam(loss.parameters(), lr=lr)
        self.graphed_runner = None
        if graphed:
            from .graphed import GraphedPPO

            n = max(1, collector.frames_per_batch // minibatch_size)
            self.graphed_runner = GraphedPPO(
                collector,
                gae,
                loss,
                optim,
                minibatches=n,
                epochs=num_epochs,
            )
        super().__init__(
            collector=collector,
            loss_module=loss,
            optimizer=optim,
            total_frames=total_frames,
            value_estimator=gae,
            minibatch_size=minibatch_size,
            num_epochs=num_epochs,
            logger=logger,
            **kwargs,
        )

    def train(self):
        if self.graphed_runner is None:
            return super().train()
        frames = 0
        self.graphed_runner.initialize()
        while frames < self.total_frames:
            self.graphed_runner.step()
            frames += self.collector.frames_per_batch
        self.collected_frames = frames
        return self


class A2CTrainer(OnPolicyTrainer):
    def __init__(self, *, actor, critic, collector, total_frames, lr: float = 7e-4,
                 gamma: float = 0.99, lmbda: float = 0.95, logger=None, **kwargs):
        loss = A2CLoss(actor, critic)
        gae = GAE(gamma=gamma, lmbda=lmbda, value_network=critic)
        optim = torch.optim.Adam(loss.parameters(), lr=lr)
        super().__init__(
            collector=collector,
            loss_module=loss,
            optimizer=optim,
            total_frames=total_frames,
            value_estimator=gae,
            num_epochs=1,
            logger=logger,
            **kwargs,
        )


class OffPolicyTrainer(Trainer):
    """Collect → replay buffer → sampled optim steps."""

    def __init__(
        self,
        *,
        collector,
        loss_module,
        optimizer,
        total_frames: int,
        replay_buffer=None,
        buffer_size: int = 1_000_000,
        batch_size: int = 256,
        prioritized: bool = False,
        device=None,
        optim_steps_per_batch: int = 8,
        target_updater=None,
        logger=None,
        **kwargs,
    ):
        super().__init__(
            collector=collector,
            total_frames=total_frames,
            loss_module=loss_module,
            optimizer=optimizer,
            logger=logger,
            optim_steps_per_batch=optim_steps_per_batch,
            **kwargs,
        )
        if replay_buffer is None:
            storage = LazyTensorStorage(buffer_size, device=device)
            if prioritized:
                replay_buffer = TensorDictPrioritizedReplayBuffer(
                    storage=storage, batch_size=batch_size
                )
            else:
                replay_buffer = TensorDictReplayBuffer(
                    storage=storage, batch_size=batch_size
                )
        self.replay_buffer = replay_buffer
        ReplayBufferTrainer(replay_buffer, batch_size=batch_size).register(self)
        if target_updater is not None:
            TargetNetUpdaterHook(target_updater).register(self)
        LogScalar().register(self)
        CountFramesLog().register(self)
        UpdateWeights(collector).register(self)


class SACTrainer(OffPolicyTrainer):
    def __init__(self, *, actor, qvalue, collector, total_frames, lr: float = 3e-4,
                 gamma: float = 0.99, tau: float = 0.005, prioritized: bool = False,
                 logger=None, **kwargs):
        loss = SACLoss(actor, qvalue, gamma=gamma)
        loss.make_value_estimator()
        optim = torch.optim.Adam(loss.parameters(), lr=lr)
        updater = SoftUpdate(loss, tau=tau)
        super().__init__(
            collector=collector,
            loss_module=loss,
            optimizer=optim,
            total_frames=total_frames,
            prioritized=prioritized,
            target_updater=updater,
            logger=logger,
            **kwargs,
        )


class TD3Trainer(OffPolicyTrainer):
    def __init__(self, *, actor, qvalue, collector, total_frames, lr: float = 3e-4,
                 gamma: float = 0.99, tau: float = 0.005, logger=None, **kwargs):
        loss = TD3Loss(actor, qvalue, gamma=gamma)
        loss.make_value_estimator()
        optim = torch.optim.Adam(loss.parameters(), lr=lr)
        updater = SoftUpdate(loss, tau=tau)
        super().__init__(
            collector=collector,
            loss_module=loss,
            optimizer=optim,
            total_frames=total_frames,
            target_updater=updater,
            logger=logger,
            **kwargs,
        )


class DDPGTrainer(OffPolicyTrainer):
    def __init__(self, *, actor, value, collector, total_frames, lr: float = 1e-3,
                 gamma: float = 0.99, tau: float = 0.005, logger=None, **kwargs):
        loss = DDPGLoss(actor, value, gamma=gamma)
        loss.make_value_estimator()
        optim = torch.optim.Adam(loss.parameters(), lr=lr)
        updater = SoftUpdate(loss, tau=tau)
        super().__init__(
            collector=collector,
            loss_module=loss,
            optimizer=optim,
            total_frames=total_frames,
            target_updater=updater,
            logger=logger,
            **kwargs,
        )


class DQNTrainer(OffPolicyTrainer):
    """(reference algorithms/dqn.py) — milestone M1 config: CartPole DQN."""

    def __init__(self, *, value_network, collector, total_frames, lr: float = 1e-3,
                 gamma: float = 0.99, tau: float = 0.02, prioritized: bool = False,
                 logger=None, **kwargs):
        loss = DQNLoss(value_network, gamma=gamma)
        loss.make_value_estimator()
        optim = torch.optim.Adam(loss.parameters(), lr=lr)
        updater = SoftUpdate(loss, tau=tau)
        super().__init__(
            collector=collector,
            loss_module=loss,
            optimizer=optim,
            total_frames=total_frames,
            prioritized=prioritized,
            target_updater=updater,
            logger=logger,
            **kwargs,
        )


class ReinforceTrainer(OnPolicyTrainer):
    """(reference algorithms/reinforce.py) — vanilla policy gradient with
    a value baseline."""

    def __init__(self, *, actor, critic, collector, total_frames, lr: float = 1e-3,
                 gamma: float = 0.99, lmbda: float = 0.95, logger=None, **kwargs):
        loss = ReinforceLoss(actor, critic)
        gae = GAE(gamma=gamma, lmbda=lmbda, value_network=critic)
        optim = torch.optim.Adam(loss.parameters(), lr=lr)
        super().__init__(
            collector=collector,
            loss_module=loss,
            optimizer=optim,
            total_frames=total_frames,
            value_estimator=gae,
            num_epochs=1,
            logger=logger,
            **kwargs,
        )


class IQLTrainer(OffPolicyTrainer):
    """(reference algorithms/iql.py) — implicit Q-learning."""

    def __init__(self, *, actor, qvalue, value, collector, total_frames,
                 lr: float = 3e-4, tau: float = 0.005, expectile: float = 0.7,
                 temperature: float = 3.0, logger=None, **kwargs):
        loss = IQLLoss(actor, qvalue, value, expectile=expectile, temperature=temperature)
        loss.make_value_estimator()
        optim = torch.optim.Adam(loss.parameters(), lr=lr)
        updater = SoftUpdate(loss, tau=tau)
        super().__init__(
            collector=collector,
            loss_module=loss,
            optimizer=optim,
            total_frames=total_frames,
            target_updater=updater,
            logger=logger,
            **kwargs,
        )


class CQLTrainer(OffPolicyTrainer):
    """(reference algorithms/cql.py) — conservative Q-learning."""

    def __init__(self, *, actor, qvalue, collector, total_frames, lr: float = 3e-4,
                 tau: float = 0.005, logger=None, **kwargs):
        loss = CQLLoss(actor, qvalue)
        loss.make_value_estimator()
        optim = torch.optim.Adam(loss.parameters(), lr=lr)
        updater = SoftUpdate(loss, tau=tau)
        super().__init__(
            collector=collector,
            loss_module=loss,
            optimizer=optim,
            total_frames=total_frames,
            target_updater=updater,
            logger=logger,
            **kwargs,
        )


class OfflineToOnlineTrainer(OffPolicyTrainer):
    """(reference algorithms/offline_to_online.py) — pretrain on an
    offline dataset, then continue online with the mixed
    :class:`~rl_amd.data.offline_to_online.OfflineOnlineReplayBuffer`."""

    def __init__(self, *, loss_module, collector, total_frames, offline_buffer,
                 lr: float = 3e-4, offline_steps: int = 1000,
                 offline_fraction: float = 0.5, batch_size: int = 256,
                 buffer_size: int = 1_000_000,
                 tau: float = 0.005, device=None, logger=None, **kwargs):
        from ..data.offline_to_online import OfflineOnlineReplayBuffer

        optim = torch.optim.Adam(loss_module.parameters(), lr=lr)
        updater = SoftUpdate(loss_module, tau=tau)
        online = TensorDictReplayBuffer(
            storage=LazyTensorStorage(buffer_size, device=device),
            batch_size=batch_size,
        )
        buf = OfflineOnlineReplayBuffer(
            offline_buffer,
            online,
            offline_fraction=offline_fraction,
            batch_size=batch_size,
        )
        super().__init__(
            collector=collector,
            loss_module=loss_module,
            optimizer=optim,
            total_frames=total_frames,
            replay_buffer=buf,
            batch_size=batch_size,
            target_updater=updater,
            logger=logger,
            **kwargs,
        )
        self.offline_steps = offline_steps

    def pretrain(self):
        """Run ``offline_steps`` optimisation passes on offline data only
        before any online collection."""
        for _ in range(self.offline_steps):
            batch = self.replay_buffer.offline.sample()
            self.optim_steps(batch)
