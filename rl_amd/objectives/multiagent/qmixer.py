"""Multi-agent value-mixing loss (QMIX / VDN).

Reference: pytorch/rl torchrl/objectives/multiagent/qmixer.py:34
(QMixerLoss); mixers in torchrl/modules/models/multiagent.py:756-1008.
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import torch

from ...tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from ..common import LossModule
from ..utils import ValueEstimators, distance_loss

__all__ = ["QMixerLoss"]


class QMixerLoss(LossModule):
    """DQN-style TD loss on the MIXED value: per-agent chosen Q-values →
    mixer → global Q_tot; target from target copies of both."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: tuple = ("agents", "action")
        action_value: tuple = ("agents", "action_value")
        local_value: tuple = ("agents", "chosen_action_value")
        global_value: str = "chosen_action_value"
        priority: str = "td_error"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.TD0

    def __init__(
        self,
        local_value_network: TensorDictModuleBase,
        mixer_network: TensorDictModuleBase,
        *,
        loss_function: str = "l2",
        delay_value: bool = True,
        double_dqn: bool = False,
        action_space: Optional[str] = None,
        gamma: Optional[float] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.convert_to_functional(
            local_value_network, "local_value_network", create_target_params=delay_value
        )
        self.convert_to_functional(
            mixer_network, "mixer_network", create_target_params=delay_value
        )
        self.loss_function = loss_function
        self.delay_value = delay_value
        self.double_dqn = double_dqn
        self.action_space = action_space or "one_hot"
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

    def _chosen(self, av: torch.Tensor, action: torch.Tensor) -> torch.Tensor:
        if action.dim() == av.dim() and action.shape[-1] == av.shape[-1]:
            # one-hot (any dtype)
            return (av * action.to(av.dtype)).sum(-1, keepdim=True)
        if action.dim() == av.dim() and action.shape[-1] == 1:
            return av.gather(-1, action.long())
        return av.gather(-1, action.long().unsqueeze(-1))

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        # local Q(s, a) per agent → mix
        td_local = self.local_value_network(td.clone(False))
        av = td_local.get(keys.action_value)
        action = td.get(keys.action)
        chosen_local = self._chosen(av, action)
        td_local.set(keys.local_value, chosen_local)
        mixed = self.mixer_network(td_local).get(keys.global_value)
        with torch.no_grad():
            nxt = td.get("next").clone(False)
            tgt_local_net = (
                self.local_value_network_target
                if self.delay_value
                else self.local_value_network
            )
            tgt_mixer = (
                self.mixer_network_target if self.delay_value else self.mixer_network
            )
            next_local_td = tgt_local_net(nxt.clone(False))
            next_av = next_local_td.get(keys.action_value)
            if self.double_dqn:
                online_next = self.local_value_network(nxt.clone(False)).get(
                    keys.action_value
                )
                best = online_next.argmax(-1, keepdim=True)
            else:
                best = next_av.argmax(-1, keepdim=True)
            next_chosen = next_av.gather(-1, best)
            next_local_td.set(keys.local_value, next_chosen)
            next_mixed = tgt_mixer(next_local_td).get(keys.global_value)
            if self.value_estimator is None:
                self.make_value_estimator()
            target = self.value_estimator.value_estimate(td, next_value=next_mixed)
        td_error = (mixed - target).abs().detach()
        tensordict.set(keys.priority, td_error)
        loss = distance_loss(mixed, target, self.loss_function)
        return TensorDict({"loss": self._reduce(loss)}, batch_size=[])
