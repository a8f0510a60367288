"""SAC losses (v2 formulation, α auto-tuning, Q-ensembles).

Reference: pytorch/rl torchrl/objectives/sac.py (SACLoss:59,
DiscreteSACLoss:979).
"""
from __future__ import annotations

import dataclasses
import math
from typing import Optional, Union

import numpy as np
import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import ValueEstimators, distance_loss

__all__ = ["SACLoss", "DiscreteSACLoss"]


class SACLoss(LossModule):
    """Soft actor-critic (v2: no explicit value net)
    (reference sac.py:59).

    * actor loss: E[α·logπ(a|s) − min_i Q_i(s, a)] with reparametrized a
    * qvalue loss: TD0 target with entropy term through the target Q nets
    * alpha loss: −α·(logπ + target_entropy)
    """

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        value: str = "state_value"
        state_action_value: str = "state_action_value"
        log_prob: str = "sample_log_prob"
        priority: str = "td_error"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.TD0
    out_keys = ["loss_actor", "loss_qvalue", "loss_alpha", "alpha", "entropy"]

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        qvalue_network: TensorDictModuleBase,
        value_network: Optional[TensorDictModuleBase] = None,
        *,
        num_qvalue_nets: int = 2,
        loss_function: str = "l2",
        alpha_init: float = 1.0,
        min_alpha: Optional[float] = None,
        max_alpha: Optional[float] = None,
        fixed_alpha: bool = False,
        target_entropy: Union[str, float] = "auto",
        delay_actor: bool = False,
        delay_qvalue: bool = True,
        gamma: Optional[float] = None,
        priority_key: Optional[str] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network", create_target_params=delay_actor)
        self.convert_to_functional(
            qvalue_network,
            "qvalue_network",
            expand_dim=num_qvalue_nets,
            create_target_params=delay_qvalue,
        )
        self.num_qvalue_nets = num_qvalue_nets
        self.loss_function = loss_function
        self.delay_qvalue = delay_qvalue
        self.reduction = reduction
        self._gamma_init = gamma
        self.fixed_alpha = fixed_alpha
        self.register_buffer("log_alpha", torch.tensor(float(np.log(alpha_init))))
        if not fixed_alpha:
            self.log_alpha = torch.nn.Parameter(self.log_alpha)
        self.min_log_alpha = math.log(min_alpha) if min_alpha else None
        self.max_log_alpha = math.log(max_alpha) if max_alpha else None
        self._target_entropy_spec = target_entropy
        self._target_entropy_val: Optional[float] = None
        if priority_key is not None:
            self._tensor_keys.priority = priority_key

    @property
    def target_entropy(self) -> float:
        if self._target_entropy_val is None:
            te = self._target_entropy_spec
            if te == "auto":
                # -dim(A): probe the actor's action spec if available
                spec = getattr(self.actor_network, "spec", None)
                if spec is not None and hasattr(spec, "shape") and len(spec.shape):
                    te = -float(np.prod(spec.shape[-1:]))
                else:
                    # matching the reference (torchrl/objectives/sac.py):
                    # silently guessing -1.0 mis-tunes alpha for
                    # multi-dim actions, so refuse instead
                    raise RuntimeError(
                        "target_entropy='auto' requires the actor to expose an "
                        "action spec (actor_network.spec with a non-empty shape). "
                        "Pass a numeric target_entropy instead."
                    )
            self._target_entropy_val = float(te)
        return self._target_entropy_val

    @property
    def alpha(self) -> torch.Tensor:
        la = self.log_alpha
        if self.min_log_alpha is not None or self.max_log_alpha is not None:
            la = la.clamp(self.min_log_alpha, self.max_log_alpha)
        return la.detach().exp()

    def make_value_estimator(self, value_type=None, **hyperparams):
        if self._gamma_init is not None:
            hyperparams.setdefault("gamma", self._gamma_init)
        out = super().make_value_estimator(value_type, **hyperparams)
        out.value_network = None
        return out

    def _reduce(self, x):
        if self.reduction == "mean":
            return x.mean()
        if self.reduction == "sum":
            return x.sum()
        return x

    def _qvalues(self, net, td: TensorDictBase, detach_params: bool = False) -> torch.Tensor:
        """[num_nets, *batch, 1] state-action values.

        ``detach_params`` mirrors the reference's
        ``qvalue_network_params.detach()`` for the actor loss: the
        gradient flows through the ACTION, not the Q weights."""
        from .common import _EnsembleModule

        if detach_params and isinstance(net, _EnsembleModule):
            out = net(td, detach_params=True)
        else:
            out = net(td)
        return out.get(self.tensor_keys.state_action_value)

    def _actor_loss(self, td: TensorDictBase):
        d = td.clone(False)
        dist = self.actor_network.get_dist(d)
        action = dist.rsample()
        log_prob = dist.log_prob(action)
        d.set(self.tensor_keys.action, action)
        q = self._qvalues(self.qvalue_network, d.clone(False), detach_params=True)
        min_q = q.min(0).values.squeeze(-1)
        loss = self.alpha * log_prob - min_q
        return loss, log_prob.detach()

    def _qvalue_loss(self, td: TensorDictBase):
        keys = self.tensor_keys
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            next_dist = self.actor_network.get_dist(nxt)
            next_action = next_dist.rsample()
            next_log_prob = next_dist.log_prob(next_action)
            nxt.set(keys.action, next_action)
            q_target_net = (
                self.qvalue_network_target if self.delay_qvalue else self.qvalue_network
            )
            next_q = self._qvalues(q_target_net, nxt)
            next_min_q = next_q.min(0).values
            next_value = next_min_q - self.alpha * next_log_prob.unsqueeze(-1)
            if self.value_estimator is None:
                self.make_value_estimator()
            target = self.value_estimator.value_estimate(td, next_value=next_value)
        q_pred = self._qvalues(self.qvalue_network, td.clone(False))
        td_error = (q_pred - target.unsqueeze(0)).abs().detach().max(0).values
        loss = distance_loss(
            q_pred, target.unsqueeze(0).expand_as(q_pred), self.loss_function
        ).sum(0)
        return loss, td_error

    def _alpha_loss(self, log_prob: torch.Tensor) -> torch.Tensor:
        if self.fixed_alpha:
            return torch.zeros_like(log_prob)
        return -self.log_alpha.exp() * (log_prob + self.target_entropy)

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        loss_actor, log_prob = self._actor_loss(td)
        loss_q, td_error = self._qvalue_loss(td)
        loss_alpha = self._alpha_loss(log_prob)
        tensordict.set(self.tensor_keys.priority, td_error)
        return TensorDict(
            {
                "loss_actor": self._reduce(loss_actor),
                "loss_qvalue": self._reduce(loss_q),
                "loss_alpha": self._reduce(loss_alpha),
                "alpha": self.alpha,
                "entropy": -log_prob.mean(),
            },
            batch_size=[],
        )


class DiscreteSACLoss(LossModule):
    """SAC for discrete action spaces (reference sac.py:979): expectation
    over all actions instead of reparametrized samples."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        action_value: str = "action_value"
        log_prob: str = "log_prob"
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
        action_space: str = "one_hot",
        num_actions: Optional[int] = None,
        num_qvalue_nets: int = 2,
        loss_function: str = "l2",
        alpha_init: float = 1.0,
        fixed_alpha: bool = False,
        target_entropy_weight: float = 0.98,
        target_entropy: Union[str, float] = "auto",
        delay_qvalue: bool = True,
        gamma: Optional[float] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        self.convert_to_functional(
            qvalue_network,
            "qvalue_network",
            expand_dim=num_qvalue_nets,
            create_target_params=delay_qvalue,
        )
        self.action_space = action_space
        self.num_actions = num_actions
        self.loss_function = loss_function
        self.delay_qvalue = delay_qvalue
        self.reduction = reduction
        self._gamma_init = gamma
        self.fixed_alpha = fixed_alpha
        self.register_buffer("log_alpha", torch.tensor(float(np.log(alpha_init))))
        if not fixed_alpha:
            self.log_alpha = torch.nn.Parameter(self.log_alpha)
        if target_entropy == "auto":
            if num_actions is None:
                raise ValueError("num_actions needed for auto target entropy")
            target_entropy = -target_entropy_weight * math.log(1.0 / num_actions) * -1.0
            # reference: -weight * log(1/n) (positive entropy target)
            target_entropy = target_entropy_weight * math.log(num_actions)
        self.target_entropy = float(target_entropy)

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
        if self.reduction == "mean":
            return x.mean()
        if self.reduction == "sum":
            return x.sum()
        return x

    def _probs(self, td):
        dist = self.actor_network.get_dist(td.clone(False))
        logits = dist.logits
        probs = logits.softmax(-1)
        log_probs = logits.log_softmax(-1)
        return probs, log_probs

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        # --- qvalue loss
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            next_probs, next_log_probs = self._probs(nxt)
            q_target_net = (
                self.qvalue_network_target if self.delay_qvalue else self.qvalue_network
            )
            next_q = q_target_net(nxt.clone(False)).get(keys.action_value)
            next_v = (
                next_probs.unsqueeze(0) * (next_q - self.alpha * next_log_probs.unsqueeze(0))
            ).sum(-1, keepdim=True)
            next_value = next_v.min(0).values
            if self.value_estimator is None:
                self.make_value_estimator()
            target = self.value_estimator.value_estimate(td, next_value=next_value)
        q_all = self.qvalue_network(td.clone(False)).get(keys.action_value)
        action = td.get(keys.action)
        if action.dtype == torch.int64 and action.dim() < q_all.dim() - 1:
            idx = action.unsqueeze(-1)
        elif action.dtype == torch.int64:
            idx = action
        else:
            idx = action.argmax(-1, keepdim=True)
        q_chosen = q_all.gather(-1, idx.unsqueeze(0).expand(q_all.shape[0], *idx.shape))
        td_error = (q_chosen - target.unsqueeze(0)).abs().detach().max(0).values
        loss_q = distance_loss(
            q_chosen, target.unsqueeze(0).expand_as(q_chosen), self.loss_function
        ).sum(0)
        # --- actor loss
        probs, log_probs = self._probs(td)
        with torch.no_grad():
            q_min = self.qvalue_network(td.clone(False)).get(keys.action_value).min(0).values
        loss_actor = (probs * (self.alpha * log_probs - q_min)).sum(-1)
        entropy = -(probs * log_probs).sum(-1).detach()
        # --- alpha loss
        if self.fixed_alpha:
            loss_alpha = torch.zeros_like(loss_actor)
        else:
            loss_alpha = self.log_alpha.exp() * (entropy - self.target_entropy).detach()
        tensordict.set(keys.priority, td_error)
        return TensorDict(
            {
                "loss_actor": self._reduce(loss_actor),
                "loss_qvalue": self._reduce(loss_q),
                "loss_alpha": self._reduce(loss_alpha),
                "alpha": self.alpha,
                "entropy": entropy.mean(),
            },
            batch_size=[],
        )
