"""TD3 and TD3+BC losses.

Reference: pytorch/rl torchrl/objectives/td3.py:27, td3_bc.py:27.
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import ValueEstimators, distance_loss

__all__ = ["TD3Loss", "TD3BCLoss"]


class TD3Loss(LossModule):
    """Twin-delayed DDPG (reference td3.py:27): twin Q nets, target policy
    smoothing, delayed actor updates handled by the trainer."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        state_action_value: str = "state_action_value"
        priority: str = "td_error"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.TD0
    out_keys = ["loss_actor", "loss_qvalue"]

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        qvalue_network: TensorDictModuleBase,
        *,
        action_spec=None,
        bounds: Optional[tuple] = None,
        num_qvalue_nets: int = 2,
        policy_noise: float = 0.2,
        noise_clip: float = 0.5,
        loss_function: str = "smooth_l1",
        delay_actor: bool = True,
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
        self.policy_noise = policy_noise
        self.noise_clip = noise_clip
        self.loss_function = loss_function
        self.delay_actor = delay_actor
        self.delay_qvalue = delay_qvalue
        self.reduction = reduction
        self._gamma_init = gamma
        if action_spec is not None and hasattr(action_spec, "low"):
            self.register_buffer("_low", action_spec.low.clone().detach().float())
            self.register_buffer("_high", action_spec.high.clone().detach().float())
        elif bounds is not None:
            self.register_buffer("_low", torch.as_tensor(bounds[0], dtype=torch.float))
            self.register_buffer("_high", torch.as_tensor(bounds[1], dtype=torch.float))
        else:
            self.register_buffer("_low", torch.tensor(-1.0))
            self.register_buffer("_high", torch.tensor(1.0))

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

    def _q(self, net, td):
        return net(td).get(self.tensor_keys.state_action_value)

    def actor_loss(self, td: TensorDictBase) -> torch.Tensor:
        d = td.clone(False)
        d = self.actor_network(d)
        q = self._q(self.qvalue_network, d.clone(False))
        # use the FIRST q net for the actor objective (reference behavior)
        return -q[0].squeeze(-1)

    def qvalue_loss(self, td: TensorDictBase):
        keys = self.tensor_keys
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            actor_t = self.actor_network_target if self.delay_actor else self.actor_network
            nxt = actor_t(nxt)
            next_action = nxt.get(keys.action)
            noise = (torch.randn_like(next_action) * self.policy_noise).clamp(
                -self.noise_clip, self.noise_clip
            )
            next_action = (next_action + noise).clamp(
                self._low.to(next_action.device), self._high.to(next_action.device)
            )
            nxt.set(keys.action, next_action)
            q_t_net = self.qvalue_network_target if self.delay_qvalue else self.qvalue_network
            next_q = self._q(q_t_net, nxt).min(0).values
            if self.value_estimator is None:
                self.make_value_estimator()
            target = self.value_estimator.value_estimate(td, next_value=next_q)
        q_pred = self._q(self.qvalue_network, td.clone(False))
        td_error = (q_pred - target.unsqueeze(0)).abs().detach().max(0).values
        loss = distance_loss(
            q_pred, target.unsqueeze(0).expand_as(q_pred), self.loss_function
        ).sum(0)
        return loss, td_error

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        loss_q, td_error = self.qvalue_loss(td)
        loss_actor = self.actor_loss(td)
        tensordict.set(self.tensor_keys.priority, td_error)
        return TensorDict(
            {
                "loss_actor": self._reduce(loss_actor),
                "loss_qvalue": self._reduce(loss_q),
            },
            batch_size=[],
        )


class TD3BCLoss(TD3Loss):
    """TD3 + behavior-cloning regularizer for offline RL
    (reference td3_bc.py:27)."""

    def __init__(self, *args, alpha: float = 2.5, **kwargs):
        super().__init__(*args, **kwargs)
        self.alpha_bc = alpha

    def actor_loss(self, td: TensorDictBase) -> torch.Tensor:
        keys = self.tensor_keys
        d = td.clone(False)
        behavior_action = d.get(keys.action)
        d = self.actor_network(d)
        pi_action = d.get(keys.action)
        q = self._q(self.qvalue_network, d.clone(False))[0].squeeze(-1)
        lam = self.alpha_bc / q.abs().mean().detach().clamp_min(1e-6)
        bc = (pi_action - behavior_action).pow(2).sum(-1)
        return -lam * q + bc
