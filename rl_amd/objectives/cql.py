"""CQL losses (conservative Q-learning, offline RL).

Reference: pytorch/rl torchrl/objectives/cql.py (CQLLoss:38,
DiscreteCQLLoss:993).
"""
from __future__ import annotations

import dataclasses
import math
from typing import Optional

import numpy as np
import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import ValueEstimators, distance_loss

__all__ = ["CQLLoss", "DiscreteCQLLoss"]


class CQLLoss(LossModule):
    """SAC-style losses + the CQL conservative regularizer
    (reference cql.py:38; Kumar et al. 2020):
    logsumexp over sampled actions minus Q at data actions.
    """

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        state_action_value: str = "state_action_value"
        log_prob: str = "sample_log_prob"
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
        temperature: float = 1.0,
        min_q_weight: float = 1.0,
        max_q_backup: bool = False,
        deterministic_backup: bool = False,
        num_random: int = 10,
        with_lagrange: bool = False,
        lagrange_thresh: float = 0.0,
        gamma: Optional[float] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        self.convert_to_functional(
            qvalue_network, "qvalue_network", expand_dim=num_qvalue_nets, create_target_params=True
        )
        self.loss_function = loss_function
        self.temperature = temperature
        self.min_q_weight = min_q_weight
        self.num_random = num_random
        self.with_lagrange = with_lagrange
        self.lagrange_thresh = lagrange_thresh
        self.reduction = reduction
        self._gamma_init = gamma
        self.register_buffer("log_alpha", torch.tensor(float(np.log(alpha_init))))
        self.log_alpha = torch.nn.Parameter(self.log_alpha)
        if with_lagrange:
            self.log_alpha_prime = torch.nn.Parameter(torch.zeros(()))
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

    def _q(self, net, td):
        return net(td).get(self.tensor_keys.state_action_value)

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        B = td.batch_size[0]
        action = td.get(keys.action)
        act_dim = action.shape[-1]
        # --- SAC-style qvalue target
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            next_dist = self.actor_network.get_dist(nxt)
            next_action = next_dist.rsample()
            next_lp = next_dist.log_prob(next_action)
            nxt.set(keys.action, next_action)
            next_q = self._q(self.qvalue_network_target, nxt).min(0).values
            next_value = next_q - self.alpha * next_lp.unsqueeze(-1)
            if self.value_estimator is None:
                self.make_value_estimator()
            target = self.value_estimator.value_estimate(td, next_value=next_value)
        q_pred = self._q(self.qvalue_network, td.clone(False))
        td_error = (q_pred - target.unsqueeze(0)).abs().detach().max(0).values
        loss_q_td = distance_loss(
            q_pred, target.unsqueeze(0).expand_as(q_pred), self.loss_function
        ).sum(0)
        # --- conservative term: logsumexp over {random, current, next} actions
        n = self.num_random
        rand_actions = (
            torch.rand(n, B, act_dim, device=action.device) * 2 - 1
        )
        cur_dist = self.actor_network.get_dist(td.clone(False))
        cur_actions = cur_dist.sample((n,))
        cur_lp = cur_dist.log_prob(cur_actions)

        def q_of(actions):
            # actions: [n, B, A] → evaluate all at once
            rep = td.clone(False)
            exp_td = TensorDict({}, batch_size=[n * B], device=td.device)
            for k in self.qvalue_network.in_keys:
                if k == keys.action:
                    continue
                v = rep.get(k, None)
                if v is not None:
                    exp_td.set(k, v.unsqueeze(0).expand(n, *v.shape).reshape(n * B, *v.shape[1:]))
            exp_td.set(keys.action, actions.reshape(n * B, act_dim))
            q = self._q(self.qvalue_network, exp_td)
            return q.reshape(q.shape[0], n, B, 1)

        q_rand = q_of(rand_actions)
        q_cur = q_of(cur_actions)
        rand_density = math.log(0.5**act_dim)
        cat = torch.cat(
            [
                q_rand / self.temperature - rand_density,
                q_cur / self.temperature - cur_lp.detach().reshape(1, n, B, 1),
            ],
            dim=1,
        )
        logsumexp = torch.logsumexp(cat, dim=1) * self.temperature
        q_data = q_pred
        cql_term = (logsumexp - q_data).sum(0) * self.min_q_weight
        loss_q = loss_q_td + cql_term.squeeze(-1) if cql_term.dim() > loss_q_td.dim() else loss_q_td + cql_term
        # --- actor (SAC)
        d = td.clone(False)
        dist = self.actor_network.get_dist(d)
        pi_action = dist.rsample()
        lp = dist.log_prob(pi_action)
        d.set(keys.action, pi_action)
        q_pi = self._q(self.qvalue_network, d).min(0).values
        loss_actor = self.alpha * lp - q_pi.squeeze(-1)
        loss_alpha = -self.log_alpha.exp() * (lp.detach() + self._target_entropy)
        tensordict.set(keys.priority, td_error)
        return TensorDict(
            {
                "loss_actor": self._reduce(loss_actor),
                "loss_qvalue": self._reduce(loss_q),
                "loss_cql": self._reduce(cql_term),
                "loss_alpha": self._reduce(loss_alpha),
                "alpha": self.alpha,
                "entropy": -lp.detach().mean(),
            },
            batch_size=[],
        )


class DiscreteCQLLoss(LossModule):
    """DQN + conservative regularizer logsumexp(Q) - Q(a_data)
    (reference cql.py:993)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        action_value: str = "action_value"
        priority: str = "td_error"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.TD0

    def __init__(self, value_network, *, loss_function: str = "l2", gamma: Optional[float] = None, action_space=None, reduction: str = "mean"):
        super().__init__()
        self.convert_to_functional(value_network, "value_network", create_target_params=True)
        self.loss_function = loss_function
        self.reduction = reduction
        self._gamma_init = gamma

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
        q_all = self.value_network(td.clone(False)).get(keys.action_value)
        action = td.get(keys.action)
        idx = action.argmax(-1, keepdim=True) if action.dtype != torch.int64 else action.unsqueeze(-1)
        chosen = q_all.gather(-1, idx)
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            next_q = self.value_network_target(nxt).get(keys.action_value)
            next_value = next_q.max(-1, keepdim=True).values
            if self.value_estimator is None:
                self.make_value_estimator()
            target = self.value_estimator.value_estimate(td, next_value=next_value)
        td_error = (chosen - target).abs().detach()
        loss_td = distance_loss(chosen, target, self.loss_function)
        cql = torch.logsumexp(q_all, dim=-1, keepdim=True) - chosen
        tensordict.set(keys.priority, td_error)
        return TensorDict(
            {
                "loss_qvalue": self._reduce(loss_td),
                "loss_cql": self._reduce(cql),
                "loss": self._reduce(loss_td + cql),
            },
            batch_size=[],
        )
