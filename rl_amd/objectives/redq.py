"""REDQ and CrossQ losses (high-UTD ensemble Q-learning).

Reference: pytorch/rl torchrl/objectives/redq.py:33, crossq.py:44.
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import numpy as np
import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import ValueEstimators, distance_loss

__all__ = ["REDQLoss", "CrossQLoss"]


class REDQLoss(LossModule):
    """Randomized-ensemble double Q (reference redq.py:33; Chen et al.
    2021): N Q-nets, targets use the min over a random SUBSET of M."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        state_action_value: str = "state_action_value"
        priority: str = "td_error"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.TD0

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        qvalue_network: TensorDictModuleBase,
        *,
        num_qvalue_nets: int = 10,
        sub_sample_len: int = 2,
        loss_function: str = "smooth_l1",
        alpha_init: float = 1.0,
        target_entropy="auto",
        gamma: Optional[float] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        self.convert_to_functional(
            qvalue_network, "qvalue_network", expand_dim=num_qvalue_nets, create_target_params=True
        )
        self.num_qvalue_nets = num_qvalue_nets
        self.sub_sample_len = sub_sample_len
        self.loss_function = loss_function
        self.reduction = reduction
        self._gamma_init = gamma
        self.register_buffer("log_alpha", torch.tensor(float(np.log(alpha_init))))
        self.log_alpha = torch.nn.Parameter(self.log_alpha)
        self._target_entropy = -1.0 if target_entropy == "auto" else float(target_entropy)

    @property
    def alpha(self):
        return self.log_alpha.detach().exp()

    def make_value_estimator(self, value_type=None, **hyperparams):
        if self._gamma_init is not None:
            hyperparams.setdefault("gamma", self._gamma_init)
        out = super().make_value_estimator(value_type, **hyperparams)
        out.value_network = None
        return out

    def _reduce(self, x):
        return x.mean() if self.reduction == "mean" else (x.sum() if self.reduction == "sum" else x)

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            dist = self.actor_network.get_dist(nxt)
            next_action = dist.rsample()
            next_lp = dist.log_prob(next_action)
            nxt.set(keys.action, next_action)
            next_q_all = self.qvalue_network_target(nxt).get(keys.state_action_value)
            sel = torch.randperm(self.num_qvalue_nets)[: self.sub_sample_len]
            next_q = next_q_all[sel].min(0).values
            next_value = next_q - self.alpha * next_lp.unsqueeze(-1)
            if self.value_estimator is None:
                self.make_value_estimator()
            target = self.value_estimator.value_estimate(td, next_value=next_value)
        q_pred = self.qvalue_network(td.clone(False)).get(keys.state_action_value)
        td_error = (q_pred - target.unsqueeze(0)).abs().detach().max(0).values
        loss_q = distance_loss(
            q_pred, target.unsqueeze(0).expand_as(q_pred), self.loss_function
        ).sum(0)
        d = td.clone(False)
        dist = self.actor_network.get_dist(d)
        pi_action = dist.rsample()
        lp = dist.log_prob(pi_action)
        d.set(keys.action, pi_action)
        q_pi = self.qvalue_network(d).get(keys.state_action_value).mean(0)
        loss_actor = self.alpha * lp - q_pi.squeeze(-1)
        loss_alpha = -self.log_alpha.exp() * (lp.detach() + self._target_entropy)
        tensordict.set(keys.priority, td_error)
        return TensorDict(
            {
                "loss_actor": self._reduce(loss_actor),
                "loss_qvalue": self._reduce(loss_q),
                "loss_alpha": self._reduce(loss_alpha),
                "alpha": self.alpha,
                "entropy": -lp.detach().mean(),
            },
            batch_size=[],
        )


class CrossQLoss(LossModule):
    """SAC without target networks: joint batch-norm forward over (s,a) and
    (s',a') (reference crossq.py:44; Bhatt et al. 2023)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        state_action_value: str = "state_action_value"
        priority: str = "td_error"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.TD0

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        qvalue_network: TensorDictModuleBase,
        *,
        num_qvalue_nets: int = 2,
        loss_function: str = "smooth_l1",
        alpha_init: float = 1.0,
        target_entropy="auto",
        gamma: Optional[float] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        self.convert_to_functional(
            qvalue_network, "qvalue_network", expand_dim=num_qvalue_nets
        )
        self.loss_function = loss_function
        self.reduction = reduction
        self._gamma_init = gamma
        self.register_buffer("log_alpha", torch.tensor(float(np.log(alpha_init))))
        self.log_alpha = torch.nn.Parameter(self.log_alpha)
        self._target_entropy = -1.0 if target_entropy == "auto" else float(target_entropy)

    @property
    def alpha(self):
        return self.log_alpha.detach().exp()

    def make_value_estimator(self, value_type=None, **hyperparams):
        if self._gamma_init is not None:
            hyperparams.setdefault("gamma", self._gamma_init)
        out = super().make_value_estimator(value_type, **hyperparams)
        out.value_network = None
        return out

    def _reduce(self, x):
        return x.mean() if self.reduction == "mean" else (x.sum() if self.reduction == "sum" else x)

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        B = td.batch_size[0]
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            dist = self.actor_network.get_dist(nxt)
            next_action = dist.rsample()
            next_lp = dist.log_prob(next_action)
        # JOINT forward of (s, a) and (s', a') through the same nets so
        # batch-norm stats see both distributions (the CrossQ trick)
        joint = TensorDict({}, batch_size=[2 * B], device=td.device)
        for k in self.qvalue_network.in_keys:
            if k == keys.action:
                cur = td.get(keys.action)
                joint.set(k, torch.cat([cur, next_action], 0))
            else:
                joint.set(k, torch.cat([td.get(k), td.get("next").get(k)], 0))
        q_joint = self.qvalue_network(joint).get(keys.state_action_value)
        q_pred, next_q = q_joint[:, :B], q_joint[:, B:]
        with torch.no_grad():
            next_min = next_q.detach().min(0).values
            next_value = next_min - self.alpha * next_lp.unsqueeze(-1)
            if self.value_estimator is None:
                self.make_value_estimator()
            target = self.value_estimator.value_estimate(td, next_value=next_value)
        td_error = (q_pred - target.unsqueeze(0)).abs().detach().max(0).values
        loss_q = distance_loss(
            q_pred, target.unsqueeze(0).expand_as(q_pred), self.loss_function
        ).sum(0)
        d = td.clone(False)
        dist = self.actor_network.get_dist(d)
        pi_action = dist.rsample()
        lp = dist.log_prob(pi_action)
        d.set(keys.action, pi_action)
        q_pi = self.qvalue_network(d).get(keys.state_action_value).min(0).values
        loss_actor = self.alpha * lp - q_pi.squeeze(-1)
        loss_alpha = -self.log_alpha.exp() * (lp.detach() + self._target_entropy)
        tensordict.set(keys.priority, td_error)
        return TensorDict(
            {
                "loss_actor": self._reduce(loss_actor),
                "loss_qvalue": self._reduce(loss_q),
                "loss_alpha": self._reduce(loss_alpha),
                "alpha": self.alpha,
            },
            batch_size=[],
        )
