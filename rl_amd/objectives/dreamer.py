"""Dreamer losses: world-model, actor (λ-returns through imagination),
value.

Reference: pytorch/rl torchrl/objectives/dreamer.py:28-373
(DreamerModelLoss, DreamerActorLoss, DreamerValueLoss) and
world_model_loss.py:19.
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import distance_loss

__all__ = ["DreamerModelLoss", "DreamerActorLoss", "DreamerValueLoss", "WorldModelLoss"]


def _normal_kl(mean_q, std_q, mean_p, std_p):
    """KL(q ‖ p) for diagonal Gaussians."""
    var_q = std_q.pow(2)
    var_p = std_p.pow(2)
    return 0.5 * (
        (var_q + (mean_q - mean_p).pow(2)) / var_p - 1 + 2 * (std_p.log() - std_q.log())
    )


class DreamerModelLoss(LossModule):
    """Reconstruction + reward prediction + KL(posterior ‖ prior) with
    free nats (reference dreamer.py:28)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        prior_mean: str = "prior_mean"
        prior_std: str = "prior_std"
        post_mean: str = "post_mean"
        post_std: str = "post_std"
        pixels: str = "pixels"
        reco_pixels: str = "reco_pixels"
        reward: tuple = ("next", "reward")
        pred_reward: str = "pred_reward"

    def __init__(
        self,
        world_model: TensorDictModuleBase,
        *,
        lambda_kl: float = 1.0,
        lambda_reco: float = 1.0,
        lambda_reward: float = 1.0,
        free_nats: float = 3.0,
        reco_loss: str = "l2",
        reward_loss: str = "l2",
    ):
        super().__init__()
        self.world_model = world_model
        self.lambda_kl = lambda_kl
        self.lambda_reco = lambda_reco
        self.lambda_reward = lambda_reward
        self.free_nats = free_nats
        self.reco_loss = reco_loss
        self.reward_loss = reward_loss

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = self.world_model(tensordict.clone(False))
        kl = _normal_kl(
            td.get(keys.post_mean),
            td.get(keys.post_std),
            td.get(keys.prior_mean),
            td.get(keys.prior_std),
        ).sum(-1)
        kl = kl.clamp_min(self.free_nats).mean()
        out = TensorDict({"loss_model_kl": self.lambda_kl * kl}, batch_size=[])
        reco = td.get(keys.reco_pixels, None)
        if reco is not None:
            target = tensordict.get(keys.pixels)
            out.set(
                "loss_model_reco",
                self.lambda_reco
                * distance_loss(reco, target, self.reco_loss).mean(),
            )
        pred_r = td.get(keys.pred_reward, None)
        if pred_r is not None:
            out.set(
                "loss_model_reward",
                self.lambda_reward
                * distance_loss(pred_r, tensordict.get(keys.reward), self.reward_loss).mean(),
            )
        # stash model outputs for downstream actor/value losses
        tensordict.update(
            td.select(
                keys.post_mean, keys.post_std, "stoch", "deter", strict=False
            )
        )
        return out


class DreamerActorLoss(LossModule):
    """Maximize λ-returns through imagined rollouts
    (reference dreamer.py): imagination happens in a ModelBasedEnv; this
    loss consumes the imagined trajectory."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        belief: str = "deter"
        reward: tuple = ("next", "reward")
        value: str = "state_value"
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    def __init__(
        self,
        actor_model: TensorDictModuleBase,
        value_model: TensorDictModuleBase,
        model_based_env,
        *,
        imagination_horizon: int = 15,
        gamma: float = 0.99,
        lmbda: float = 0.95,
    ):
        super().__init__()
        self.actor_model = actor_model
        self.value_model = value_model
        self.model_based_env = model_based_env
        self.imagination_horizon = imagination_horizon
        self.gamma = gamma
        self.lmbda = lmbda

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        # imagine forward from the posterior latents
        start = tensordict.clone(False)
        rollout = self.model_based_env.rollout(
            self.imagination_horizon,
            policy=self.actor_model,
            auto_reset=False,
            tensordict=start,
            break_when_any_done=False,
        )
        rollout = self.value_model(rollout)
        from .value.functional import vec_td_lambda_return_estimate

        keys = self.tensor_keys
        reward = rollout.get(keys.reward)
        value = rollout.get(keys.value)
        done = rollout.get(keys.done, torch.zeros_like(reward, dtype=torch.bool))
        returns = vec_td_lambda_return_estimate(
            self.gamma, self.lmbda, value, reward, done, done
        )
        loss_actor = -returns.mean()
        tensordict.set("lambda_returns", returns.detach())
        tensordict.set("imagined_rollout", rollout.detach())
        return TensorDict({"loss_actor": loss_actor}, batch_size=[])


class DreamerValueLoss(LossModule):
    """Regress value toward the λ-returns of the imagined rollout
    (reference dreamer.py)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        value: str = "state_value"

    def __init__(self, value_model: TensorDictModuleBase, *, value_loss: str = "l2", discount_loss: bool = False):
        super().__init__()
        self.value_model = value_model
        self.value_loss = value_loss

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        rollout = tensordict.get("imagined_rollout")
        target = tensordict.get("lambda_returns")
        td = self.value_model(rollout.clone(False))
        value = td.get(self.tensor_keys.value)
        loss = distance_loss(value, target, self.value_loss).mean()
        return TensorDict({"loss_value": loss}, batch_size=[])


class WorldModelLoss(DreamerModelLoss):
    """Generic world-model loss alias (reference world_model_loss.py:19)."""
