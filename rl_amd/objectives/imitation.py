"""Imitation & auxiliary losses: BC, GAIL, RND, Decision Transformer.

Reference: pytorch/rl torchrl/objectives/bc.py:22, gail.py:19, rnd.py:19,
decision_transformer.py (OnlineDTLoss:22, DTLoss:285).
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import numpy as np
import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import distance_loss

__all__ = ["BCLoss", "GAILLoss", "RNDLoss", "DTLoss", "OnlineDTLoss"]


class BCLoss(LossModule):
    """Behavior cloning: −logπ(a|s) (probabilistic) or MSE (deterministic)
    (reference bc.py:22)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"

    def __init__(self, actor_network: TensorDictModuleBase, *, loss_function: str = "l2", reduction: str = "mean"):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        self.loss_function = loss_function
        self.reduction = reduction

    def _reduce(self, x):
        return x.mean() if self.reduction == "mean" else (x.sum() if self.reduction == "sum" else x)

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        action = td.get(self.tensor_keys.action)
        if hasattr(self.actor_network, "get_dist"):
            dist = self.actor_network.get_dist(td)
            loss = -dist.log_prob(action)
        else:
            out = self.actor_network(td.clone(False))
            pred = out.get(self.tensor_keys.action)
            loss = distance_loss(pred, action, self.loss_function).sum(-1)
        return TensorDict({"loss_bc": self._reduce(loss)}, batch_size=[])


class GAILLoss(LossModule):
    """Discriminator loss for adversarial imitation (reference gail.py:19).

    ``discriminator`` maps (obs, action) → logit; expert batches get label
    1, collected batches 0; optional gradient penalty."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        expert_action: str = "action"
        expert_observation: str = "observation"
        collector_action: str = "collector_action"
        collector_observation: str = "collector_observation"
        discriminator_pred: str = "d_logits"

    def __init__(
        self,
        discriminator_network: TensorDictModuleBase,
        *,
        use_grad_penalty: bool = False,
        gp_lambda: float = 10.0,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(discriminator_network, "discriminator_network")
        self.use_grad_penalty = use_grad_penalty
        self.gp_lambda = gp_lambda
        self.reduction = reduction

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        expert = TensorDict(
            {
                "observation": tensordict.get(keys.expert_observation),
                "action": tensordict.get(keys.expert_action),
            },
            batch_size=tensordict.batch_size,
        )
        collected = TensorDict(
            {
                "observation": tensordict.get(keys.collector_observation),
                "action": tensordict.get(keys.collector_action),
            },
            batch_size=tensordict.batch_size,
        )
        d_expert = self.discriminator_network(expert).get(keys.discriminator_pred)
        d_col = self.discriminator_network(collected).get(keys.discriminator_pred)
        bce = torch.nn.functional.binary_cross_entropy_with_logits
        loss = bce(d_expert, torch.ones_like(d_expert)) + bce(
            d_col, torch.zeros_like(d_col)
        )
        out = TensorDict({"loss": loss}, batch_size=[])
        if self.use_grad_penalty:
            eps = torch.rand(
                tensordict.batch_size[0], 1, device=d_expert.device
            )
            mix_obs = (
                eps * expert.get("observation") + (1 - eps) * collected.get("observation")
            ).requires_grad_(True)
            mix_act = (
                eps * expert.get("action").float()
                + (1 - eps) * collected.get("action").float()
            ).requires_grad_(True)
            mix = TensorDict(
                {"observation": mix_obs, "action": mix_act},
                batch_size=tensordict.batch_size,
            )
            d_mix = self.discriminator_network(mix).get(keys.discriminator_pred)
            grads = torch.autograd.grad(
                d_mix.sum(), (mix_obs, mix_act), create_graph=True
            )
            gnorm = torch.cat([g.reshape(g.shape[0], -1) for g in grads], -1).norm(
                2, dim=-1
            )
            gp = self.gp_lambda * (gnorm - 1).pow(2).mean()
            out.set("gp", gp)
            out.set("loss", loss + gp)
        return out


class RNDLoss(LossModule):
    """Random network distillation intrinsic-reward predictor loss
    (reference rnd.py:19; also envs/transforms/rnd.py:80 for the reward
    side)."""

    def __init__(self, predictor: torch.nn.Module, target: torch.nn.Module, *, observation_key: str = "observation", reduction: str = "mean"):
        super().__init__()
        self.predictor = predictor
        self.target = target
        for p in self.target.parameters():
            p.requires_grad_(False)
        self.observation_key = observation_key
        self.reduction = reduction

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        obs = tensordict.get(self.observation_key)
        with torch.no_grad():
            t = self.target(obs)
        p = self.predictor(obs)
        loss = (p - t).pow(2).mean(-1)
        if self.reduction == "mean":
            loss = loss.mean()
        elif self.reduction == "sum":
            loss = loss.sum()
        return TensorDict({"loss_rnd": loss}, batch_size=[])

    def intrinsic_reward(self, tensordict: TensorDictBase) -> torch.Tensor:
        with torch.no_grad():
            obs = tensordict.get(self.observation_key)
            return (self.predictor(obs) - self.target(obs)).pow(2).mean(-1, keepdim=True)


class DTLoss(LossModule):
    """Decision-transformer supervised action loss (reference
    decision_transformer.py:285)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        action_target: str = "action"
        action_pred: str = "action"

    def __init__(self, actor_network: TensorDictModuleBase, *, loss_function: str = "l2", reduction: str = "mean"):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        self.loss_function = loss_function
        self.reduction = reduction

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        target = td.get(self.tensor_keys.action_target)
        out = self.actor_network(td)
        pred = out.get(self.tensor_keys.action_pred)
        loss = distance_loss(pred, target, self.loss_function)
        if self.reduction == "mean":
            loss = loss.mean()
        elif self.reduction == "sum":
            loss = loss.sum()
        return TensorDict({"loss": loss}, batch_size=[])


class OnlineDTLoss(LossModule):
    """Stochastic-policy DT: −logπ + entropy temperature
    (reference decision_transformer.py:22)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        action_target: str = "action"

    def __init__(self, actor_network: TensorDictModuleBase, *, alpha_init: float = 0.1, target_entropy="auto", samples_mc_entropy: int = 1, reduction: str = "mean"):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        self.register_buffer("log_alpha", torch.tensor(float(np.log(alpha_init))))
        self.log_alpha = torch.nn.Parameter(self.log_alpha)
        self._target_entropy = -1.0 if target_entropy == "auto" else float(target_entropy)
        self.samples_mc_entropy = samples_mc_entropy
        self.reduction = reduction

    @property
    def alpha(self):
        return self.log_alpha.detach().exp()

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        target = td.get(self.tensor_keys.action_target)
        dist = self.actor_network.get_dist(td)
        log_likelihood = dist.log_prob(target)
        try:
            entropy = dist.entropy()
        except NotImplementedError:
            x = dist.rsample((self.samples_mc_entropy,))
            entropy = -dist.log_prob(x).mean(0)
        loss_log_likelihood = -log_likelihood
        loss_entropy = -self.alpha * entropy
        loss_alpha = self.log_alpha.exp() * (entropy.detach() - self._target_entropy)
        red = (lambda x: x.mean()) if self.reduction == "mean" else (lambda x: x.sum())
        return TensorDict(
            {
                "loss_log_likelihood": red(loss_log_likelihood),
                "loss_entropy": red(loss_entropy),
                "loss_alpha": red(loss_alpha),
                "entropy": entropy.detach().mean(),
                "alpha": self.alpha,
            },
            batch_size=[],
        )
