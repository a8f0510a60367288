"""DreamerV3 losses: world-model (categorical KL balancing + symlog
reconstruction + two-hot reward), actor (imagination REINFORCE /
reparameterized with EMA-percentile return normalization), value
(two-hot CE or symlog MSE).

Reference: pytorch/rl torchrl/objectives/dreamer_v3.py:197
(DreamerV3ModelLoss), :509 (DreamerV3ActorLoss), :972
(DreamerV3ValueLoss) and the paper "Mastering Diverse Domains in World
Models" (Hafner et al., 2023).
"""
from __future__ import annotations

import dataclasses
from typing import Optional, Tuple

import torch

from ..modules.functional import (
    default_bins,
    symexp,
    symlog,
    two_hot_cross_entropy,
    unimix_probs,
)
from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule

__all__ = [
    "categorical_kl_terms",
    "DreamerV3ModelLoss",
    "DreamerV3ActorLoss",
    "DreamerV3ValueLoss",
]


def categorical_kl_terms(
    posterior_logits: torch.Tensor,
    prior_logits: torch.Tensor,
    free_nats: float = 1.0,
    unimix: float = 0.01,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """DreamerV3 dynamics and representation KL losses over stacks of
    categoricals ``[..., num_categoricals, num_classes]``.

    dynamics = KL(sg(post) || prior), representation = KL(post || sg(prior));
    each summed over the categorical stack, then free-bits clamped and
    averaged over batch/time.
    """

    def _kl(p_logits, q_logits):
        p = unimix_probs(p_logits, unimix)
        q = unimix_probs(q_logits, unimix)
        return (p * (p.clamp_min(1e-8).log() - q.clamp_min(1e-8).log())).sum(-1)

    dyn = _kl(posterior_logits.detach(), prior_logits).sum(-1)
    rep = _kl(posterior_logits, prior_logits.detach()).sum(-1)
    return dyn.clamp_min(free_nats).mean(), rep.clamp_min(free_nats).mean()


class DreamerV3ModelLoss(LossModule):
    """World-model loss (reference dreamer_v3.py:197): balanced
    categorical KL + symlog-space reconstruction + two-hot reward CE
    (+ optional continue BCE)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        prior_logits: str = "prior_logits"
        posterior_logits: str = "posterior_logits"
        pixels: str = "pixels"
        reco_pixels: str = "reco_pixels"
        reward: tuple = ("next", "reward")
        reward_logits: str = "reward_logits"
        pred_reward: str = "pred_reward"
        continue_logits: str = "continue_logits"
        terminated: tuple = ("next", "terminated")

    def __init__(
        self,
        world_model: TensorDictModuleBase,
        *,
        lambda_kl: float = 1.0,
        lambda_reco: float = 1.0,
        lambda_reward: float = 1.0,
        lambda_continue: float = 0.0,
        kl_alpha: float = 0.8,
        free_bits: float = 1.0,
        unimix: float = 0.01,
        reco_loss: str = "l2",
        reward_two_hot: bool = True,
        num_reward_bins: int = 255,
    ):
        super().__init__()
        self.world_model = world_model
        self.lambda_kl = lambda_kl
        self.lambda_reco = lambda_reco
        self.lambda_reward = lambda_reward
        self.lambda_continue = lambda_continue
        self.kl_alpha = kl_alpha
        self.free_bits = free_bits
        self.unimix = unimix
        self.reco_loss = reco_loss
        self.reward_two_hot = reward_two_hot
        self.register_buffer("reward_bins", default_bins(num_reward_bins))

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = self.world_model(tensordict.clone(False))
        dyn, rep = categorical_kl_terms(
            td.get(keys.posterior_logits),
            td.get(keys.prior_logits),
            free_nats=self.free_bits,
            unimix=self.unimix,
        )
        # KL balancing: alpha trains the prior (dynamics), 1-alpha the posterior
        kl = self.kl_alpha * dyn + (1 - self.kl_alpha) * rep
        out = TensorDict(
            {
                "loss_model_kl": self.lambda_kl * kl,
                "kl_dynamics": dyn.detach(),
                "kl_representation": rep.detach(),
            },
            batch_size=[],
        )
        reco = td.get(keys.reco_pixels, None)
        if reco is not None:
            target = symlog(tensordict.get(keys.pixels).float())
            err = reco - target
            reco_l = err.pow(2) if self.reco_loss == "l2" else err.abs()
            out.set("loss_model_reco", self.lambda_reco * reco_l.mean())
        if self.reward_two_hot:
            logits = td.get(keys.reward_logits, None)
            if logits is not None:
                r = tensordict.get(keys.reward).squeeze(-1)
                ce = two_hot_cross_entropy(logits, r, self.reward_bins.to(logits.device))
                out.set("loss_model_reward", self.lambda_reward * ce.mean())
        else:
            pred = td.get(keys.pred_reward, None)
            if pred is not None:
                r = symlog(tensordict.get(keys.reward))
                out.set(
                    "loss_model_reward",
                    self.lambda_reward * (pred - r).pow(2).mean(),
                )
        if self.lambda_continue:
            cl = td.get(keys.continue_logits, None)
            if cl is not None:
                cont = (~tensordict.get(keys.terminated)).float().reshape(cl.shape)
                bce = torch.nn.functional.binary_cross_entropy_with_logits(cl, cont)
                out.set("loss_model_continue", self.lambda_continue * bce)
        # stash latents for downstream actor/value losses
        tensordict.update(
            td.select("stoch", "deter", keys.posterior_logits, strict=False)
        )
        return out


class DreamerV3ActorLoss(LossModule):
    """Actor loss over imagined latent rollouts (reference
    dreamer_v3.py:509): lambda-returns, REINFORCE (sg advantage) or
    straight reparameterized gradient, EMA return-percentile
    normalization, entropy bonus."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        reward: tuple = ("next", "reward")
        value: str = "state_value"
        done: tuple = ("next", "done")
        sample_log_prob: str = "sample_log_prob"

    def __init__(
        self,
        actor_model: TensorDictModuleBase,
        value_model: TensorDictModuleBase,
        model_based_env,
        *,
        imagination_horizon: int = 15,
        gamma: float = 0.997,
        lmbda: float = 0.95,
        entropy_bonus: float = 3e-4,
        use_reinforce: bool = False,
        return_normalization: bool = True,
        return_normalization_rate: float = 0.01,
        return_normalization_quantiles: Tuple[float, float] = (0.05, 0.95),
        return_normalization_min_scale: float = 1.0,
    ):
        super().__init__()
        self.actor_model = actor_model
        self.value_model = value_model
        self.model_based_env = model_based_env
        self.imagination_horizon = imagination_horizon
        self.gamma = gamma
        self.lmbda = lmbda
        self.entropy_bonus = entropy_bonus
        self.use_reinforce = use_reinforce
        self.return_normalization = return_normalization
        self.return_normalization_rate = return_normalization_rate
        self.return_normalization_quantiles = return_normalization_quantiles
        self.return_normalization_min_scale = return_normalization_min_scale
        self.register_buffer("_ret_lo", torch.tensor(0.0))
        self.register_buffer("_ret_hi", torch.tensor(0.0))
        self.register_buffer("_ret_init", torch.tensor(False))

    def _normalize(self, returns: torch.Tensor) -> torch.Tensor:
        qlo, qhi = self.return_normalization_quantiles
        lo = torch.quantile(returns.detach().float().reshape(-1), qlo)
        hi = torch.quantile(returns.detach().float().reshape(-1), qhi)
        r = self.return_normalization_rate
        if not bool(self._ret_init):
            self._ret_lo.copy_(lo)
            self._ret_hi.copy_(hi)
            self._ret_init.copy_(torch.tensor(True))
        else:
            self._ret_lo.mul_(1 - r).add_(r * lo)
            self._ret_hi.mul_(1 - r).add_(r * hi)
        span = (self._ret_hi - self._ret_lo).clamp_min(
            self.return_normalization_min_scale
        )
        return returns / span

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
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

        reward = rollout.get(keys.reward)
        value = rollout.get(keys.value)
        done = rollout.get(keys.done, torch.zeros_like(reward, dtype=torch.bool))
        returns = vec_td_lambda_return_estimate(
            self.gamma, self.lmbda, value, reward, done, done
        )
        if self.use_reinforce:
            adv = (returns - value).detach()
            if self.return_normalization:
                adv = self._normalize(adv)
            log_prob = rollout.get(keys.sample_log_prob)
            loss_actor = -(log_prob * adv.squeeze(-1).reshape(log_prob.shape)).mean()
        else:
            ret = self._normalize(returns) if self.return_normalization else returns
            loss_actor = -ret.mean()
        out = TensorDict({"loss_actor": loss_actor}, batch_size=[])
        if self.entropy_bonus:
            lp = rollout.get(keys.sample_log_prob, None)
            if lp is not None:
                out.set("loss_entropy", self.entropy_bonus * lp.mean())
        tensordict.set("lambda_returns", returns.detach())
        tensordict.set("imagined_rollout", rollout.detach())
        return out


class DreamerV3ValueLoss(LossModule):
    """Value regression toward the imagined lambda-returns (reference
    dreamer_v3.py:972): two-hot CE over symlog bins, or symlog MSE."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        value: str = "state_value"
        value_logits: str = "value_logits"

    def __init__(
        self,
        value_model: TensorDictModuleBase,
        *,
        two_hot: bool = True,
        num_bins: int = 255,
    ):
        super().__init__()
        self.value_model = value_model
        self.two_hot = two_hot
        self.register_buffer("bins", default_bins(num_bins))

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        rollout = tensordict.get("imagined_rollout")
        target = tensordict.get("lambda_returns")
        td = self.value_model(rollout.clone(False))
        if self.two_hot:
            logits = td.get(self.tensor_keys.value_logits)
            ce = two_hot_cross_entropy(
                logits, target.squeeze(-1), self.bins.to(logits.device)
            )
            loss = ce.mean()
        else:
            value = td.get(self.tensor_keys.value)
            loss = (symlog(target) - value).pow(2).mean()
        return TensorDict({"loss_value": loss}, batch_size=[])


def categorical_kl_balanced(
    posterior_logits: torch.Tensor,
    prior_logits: torch.Tensor,
    alpha: float = 0.8,
    free_bits: float = 1.0,
) -> torch.Tensor:
    """Balanced categorical KL (reference dreamer_v3.py:114):
    ``alpha·KL(sg(post)||prior) + (1−alpha)·KL(post||sg(prior))`` with
    free-bits clamping — the scalar-weighted form of
    :func:`categorical_kl_terms`."""
    dyn, rep = categorical_kl_terms(
        posterior_logits, prior_logits, free_nats=free_bits
    )
    return alpha * dyn + (1.0 - alpha) * rep


__all__.append("categorical_kl_balanced")
