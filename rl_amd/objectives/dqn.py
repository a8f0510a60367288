"""DQN losses.

Reference: pytorch/rl torchrl/objectives/dqn.py (DQNLoss:34,
DistributionalDQNLoss:389).
"""
from __future__ import annotations

import dataclasses
from typing import Optional, Union

import torch

from ..data.tensor_specs import OneHot
from ..modules.tensordict_module.actors import QValueActor
from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import ValueEstimators, distance_loss

__all__ = ["DQNLoss", "DistributionalDQNLoss"]


class DQNLoss(LossModule):
    """TD0 Q-learning loss with optional double-DQN
    (reference dqn.py:34)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        advantage: str = "advantage"
        value_target: str = "value_target"
        value: str = "chosen_action_value"
        action_value: str = "action_value"
        action: str = "action"
        priority: str = "td_error"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.TD0
    out_keys = ["loss"]

    def __init__(
        self,
        value_network: Union[QValueActor, TensorDictModuleBase],
        *,
        loss_function: str = "l2",
        delay_value: bool = True,
        double_dqn: bool = False,
        gamma: Optional[float] = None,
        action_space: Optional[str] = None,
        priority_key: Optional[str] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.loss_function = loss_function
        self.delay_value = delay_value
        self.double_dqn = double_dqn
        self.reduction = reduction
        if action_space is None:
            spec = getattr(value_network, "spec", None)
            action_space = "one_hot" if spec is None or isinstance(spec, OneHot) else "categorical"
        self.action_space = action_space
        self.convert_to_functional(
            value_network, "value_network", create_target_params=delay_value
        )
        self._gamma_init = gamma
        if priority_key is not None:
            self._tensor_keys.priority = priority_key

    def make_value_estimator(self, value_type=None, **hyperparams):
        if self._gamma_init is not None:
            hyperparams.setdefault("gamma", self._gamma_init)
        out = super().make_value_estimator(value_type, **hyperparams)
        # the estimator must read Q(s',·)max as the next value: we feed it
        # pre-computed next values, so no value_network inside
        out.value_network = None
        return out

    def _reduce(self, loss: torch.Tensor) -> torch.Tensor:
        if self.reduction == "mean":
            return loss.mean()
        if self.reduction == "sum":
            return loss.sum()
        return loss

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        keys = self.tensor_keys
        # online Q(s, a)
        td_q = self.value_network(td.clone(False))
        action_value = td_q.get(keys.action_value)
        action = td.get(keys.action)
        if self.action_space == "categorical":
            idx = action.long()
            if idx.shape != action_value.shape[:-1]:
                idx = idx.squeeze(-1)
            chosen = action_value.gather(-1, idx.unsqueeze(-1))
        else:
            chosen = (action_value * action.to(action_value.dtype)).sum(-1, keepdim=True)

        # target: max_a Q_target(s', a)  (double-DQN: argmax from online)
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            target_net = (
                self.value_network_target if self.delay_value else self.value_network
            )
            next_q_td = target_net(nxt.clone(False))
            next_av = next_q_td.get(keys.action_value)
            if self.double_dqn:
                online_next = self.value_network(nxt.clone(False))
                next_sel = online_next.get(keys.action_value).argmax(-1, keepdim=True)
                next_value = next_av.gather(-1, next_sel)
            else:
                next_value = next_av.max(-1, keepdim=True).values
            if self.value_estimator is None:
                self.make_value_estimator()
            # write pred next value for the estimator
            target = self.value_estimator.value_estimate(
                td, next_value=next_value
            )
        td_error = (chosen - target).abs().detach()
        tensordict.set(keys.priority, td_error)
        loss = distance_loss(chosen, target, self.loss_function)
        loss = self._reduce(loss)
        return TensorDict(
            {"loss": loss, "td_error": td_error.mean()}, batch_size=[]
        )


class DistributionalDQNLoss(LossModule):
    """C51 categorical distributional DQN (reference dqn.py:389).

    ``value_network`` outputs log-probs over atoms:
    ``action_value`` shaped [..., atoms, n_actions]; support is the atom
    grid registered on the actor."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        action_value: str = "action_value"
        action: str = "action"
        priority: str = "td_error"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.TD0

    def __init__(self, value_network, *, gamma: float = 0.99, delay_value: bool = True, priority_key=None, reduction: str = "mean"):
        super().__init__()
        self.register_buffer("gamma", torch.as_tensor(gamma))
        self.delay_value = delay_value
        self.reduction = reduction
        self.convert_to_functional(
            value_network, "value_network", create_target_params=delay_value
        )

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        support = None
        for m in self.value_network.modules():
            if hasattr(m, "support") and isinstance(getattr(m, "support"), torch.Tensor):
                support = m.support
                break
        if support is None:
            raise RuntimeError("value_network must carry an atom `support` buffer")
        n_atoms = support.numel()
        delta_z = (support[-1] - support[0]) / (n_atoms - 1)

        td_q = self.value_network(td.clone(False))
        logits = td_q.get(keys.action_value)  # [..., atoms, actions]
        log_p = logits.log_softmax(-2)
        action = td.get(keys.action)
        if action.dtype == torch.int64 and action.shape == log_p.shape[:-2]:
            idx = action
        elif action.dtype == torch.int64:
            idx = action.squeeze(-1)
        else:
            idx = action.argmax(-1)
        idx_exp = idx.unsqueeze(-1).unsqueeze(-1).expand(*idx.shape, n_atoms, 1)
        log_p_a = log_p.gather(-1, idx_exp).squeeze(-1)  # [..., atoms]

        with torch.no_grad():
            nxt = td.get("next").clone(False)
            target_net = (
                self.value_network_target if self.delay_value else self.value_network
            )
            next_td = target_net(nxt)
            next_logits = next_td.get(keys.action_value)
            next_p = next_logits.softmax(-2)
            next_q = (next_p * support.unsqueeze(-1)).sum(-2)
            next_a = next_q.argmax(-1)
            na_exp = next_a.unsqueeze(-1).unsqueeze(-1).expand(*next_a.shape, n_atoms, 1)
            next_p_a = next_p.gather(-1, na_exp).squeeze(-1)  # [..., atoms]
            reward = td.get(keys.reward)
            terminated = td.get(keys.terminated, td.get(keys.done))
            not_term = (~terminated).to(reward.dtype)
            Tz = (reward + self.gamma * not_term * support).clamp(
                support[0], support[-1]
            )  # [..., atoms]
            b = (Tz - support[0]) / delta_z
            lo = b.floor().long().clamp(0, n_atoms - 1)
            hi = b.ceil().long().clamp(0, n_atoms - 1)
            # distribute probability mass
            m = torch.zeros_like(next_p_a)
            lo_w = (hi.to(b.dtype) - b).where(lo != hi, torch.ones_like(b))
            hi_w = b - lo.to(b.dtype)
            m.scatter_add_(-1, lo, next_p_a * lo_w)
            m.scatter_add_(-1, hi, next_p_a * hi_w)
        loss = -(m * log_p_a).sum(-1)
        tensordict.set(keys.priority, loss.detach().unsqueeze(-1))
        if self.reduction == "mean":
            loss = loss.mean()
        elif self.reduction == "sum":
            loss = loss.sum()
        return TensorDict({"loss": loss}, batch_size=[])
