"""Actor / critic wrappers — the public policy API.

Reference: pytorch/rl torchrl/modules/tensordict_module/actors.py
(Actor:36, ProbabilisticActor:146, ValueOperator:427, QValueModule:500,
QValueActor:1108, ActorValueOperator:1415, ActorCriticOperator:1564,
ActorCriticWrapper:1725, TanhModule:2066).
"""
from __future__ import annotations

from typing import Callable, List, Optional, Sequence, Union

import torch
from torch import nn

from ...data.tensor_specs import Composite, OneHot, TensorSpec
from ...tensordict import (
    InteractionType,
    ProbabilisticTensorDictModule,
    ProbabilisticTensorDictSequential,
    TensorDict,
    TensorDictBase,
    TensorDictModule,
    TensorDictModuleBase,
    TensorDictSequential,
    unravel_key,
)

__all__ = [
    "Actor",
    "ProbabilisticActor",
    "ValueOperator",
    "QValueModule",
    "QValueActor",
    "DistributionalQValueModule",
    "DistributionalQValueActor",
    "ActorValueOperator",
    "ActorCriticOperator",
    "ActorCriticWrapper",
    "TanhModule",
    "MultiStepActorWrapper",
]


class Actor(TensorDictModule):
    """Deterministic actor: obs → action, with optional spec projection
    (reference actors.py:36)."""

    def __init__(
        self,
        module: nn.Module,
        in_keys: Sequence = ("observation",),
        out_keys: Sequence = ("action",),
        spec: Optional[TensorSpec] = None,
        safe: bool = False,
    ):
        super().__init__(module, in_keys=list(in_keys), out_keys=list(out_keys))
        self.spec = spec
        self.safe = safe

    def forward(self, td=None, *args, **kwargs):
        out = super().forward(td, *args, **kwargs)
        if self.safe and self.spec is not None and isinstance(out, TensorDictBase):
            out.set(self.out_keys[0], self.spec.project(out.get(self.out_keys[0])))
        return out


class ProbabilisticActor(ProbabilisticTensorDictSequential):
    """Stochastic actor: net → distribution → sample
    (reference actors.py:146)."""

    def __init__(
        self,
        module: TensorDictModuleBase,
        in_keys: Union[str, Sequence],
        out_keys: Optional[Sequence] = None,
        spec: Optional[TensorSpec] = None,
        safe: bool = False,
        distribution_class=None,
        distribution_kwargs: Optional[dict] = None,
        default_interaction_type: InteractionType = InteractionType.RANDOM,
        return_log_prob: bool = False,
        log_prob_key: str = "sample_log_prob",
        cache_dist: bool = False,
        n_empirical_estimate: int = 1000,
    ):
        if distribution_class is None:
            from ..distributions.continuous import TanhNormal

            distribution_class = TanhNormal
        if out_keys is None:
            out_keys = ["action"]
        if isinstance(in_keys, (str, tuple)) and (
            isinstance(in_keys, str) or all(isinstance(k, str) for k in in_keys) and False
        ):
            in_keys = [in_keys]
        prob = ProbabilisticTensorDictModule(
            in_keys=in_keys,
            out_keys=list(out_keys),
            distribution_class=distribution_class,
            distribution_kwargs=distribution_kwargs,
            default_interaction_type=default_interaction_type,
            return_log_prob=return_log_prob,
            log_prob_key=log_prob_key,
        )
        super().__init__(module, prob)
        self.spec = spec
        self.safe = safe

    def forward(self, td=None, *args, **kwargs):
        out = super().forward(td, *args, **kwargs)
        if self.safe and self.spec is not None and isinstance(out, TensorDictBase):
            key = self._prob_module.out_keys[0]
            out.set(key, self.spec.project(out.get(key)))
        return out


class ValueOperator(TensorDictModule):
    """Critic wrapper: obs (+action) → state_value / state_action_value
    (reference actors.py:427)."""

    def __init__(
        self,
        module: nn.Module,
        in_keys: Sequence = ("observation",),
        out_keys: Optional[Sequence] = None,
    ):
        if out_keys is None:
            out_keys = (
                ["state_value"]
                if "action" not in [unravel_key(k) for k in in_keys]
                else ["state_action_value"]
            )
        super().__init__(module, in_keys=list(in_keys), out_keys=list(out_keys))


class QValueModule(TensorDictModuleBase):
    """argmax over action values → (one-hot or categorical) action +
    chosen_action_value (reference actors.py:500)."""

    def __init__(
        self,
        action_space: Optional[str] = None,
        action_value_key: str = "action_value",
        action_mask_key: Optional[str] = None,
        out_keys: Optional[Sequence] = None,
        spec: Optional[TensorSpec] = None,
        safe: bool = False,
    ):
        super().__init__()
        if action_space is None:
            action_space = (
                "categorical"
                if spec is not None and not isinstance(spec, OneHot)
                else "one_hot"
            )
        self.action_space = action_space
        self.action_value_key = action_value_key
        self.action_mask_key = action_mask_key
        self.in_keys = [action_value_key] + (
            [action_mask_key] if action_mask_key else []
        )
        self.out_keys = (
            list(out_keys)
            if out_keys is not None
            else ["action", action_value_key, "chosen_action_value"]
        )
        self.spec = spec

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        values = td.get(self.action_value_key)
        if self.action_mask_key is not None:
            mask = td.get(self.action_mask_key, None)
            if mask is not None:
                values = values.masked_fill(~mask, -3.4e38)
        idx = values.argmax(-1)
        if self.action_space == "one_hot":
            action = torch.nn.functional.one_hot(idx, values.shape[-1])
            # match the env's declared action dtype (OneHot defaults to bool)
            action = action.to(
                self.spec.dtype if self.spec is not None else torch.int64
            )
        else:
            action = idx
        chosen = values.gather(-1, idx.unsqueeze(-1))
        td.set(self.out_keys[0], action)
        td.set(self.out_keys[2], chosen)
        return td


class QValueActor(TensorDictSequential):
    """net → action_value → argmax action (reference actors.py:1108)."""

    def __init__(
        self,
        module: nn.Module,
        in_keys: Sequence = ("observation",),
        spec: Optional[TensorSpec] = None,
        safe: bool = False,
        action_space: Optional[str] = None,
        action_value_key: str = "action_value",
        action_mask_key: Optional[str] = None,
    ):
        if action_space is None:
            action_space = (
                "categorical"
                if spec is not None and not isinstance(spec, OneHot)
                else "one_hot"
            )
        if isinstance(module, TensorDictModuleBase):
            net = module
        else:
            net = TensorDictModule(
                module, in_keys=list(in_keys), out_keys=[action_value_key]
            )
        qvalue = QValueModule(
            action_space=action_space,
            action_value_key=action_value_key,
            action_mask_key=action_mask_key,
            spec=spec,
        )
        super().__init__(net, qvalue)
        self.spec = spec


class DistributionalQValueModule(QValueModule):
    """argmax over E[Z] of a categorical value distribution
    (reference actors.py:750).  ``action_value`` holds log-softmax logits
    over (atoms, actions); support is the atom grid."""

    def __init__(
        self,
        action_space: Optional[str] = None,
        support: torch.Tensor = None,
        action_value_key: str = "action_value",
        out_keys: Optional[Sequence] = None,
        spec=None,
        safe: bool = False,
    ):
        super().__init__(
            action_space=action_space,
            action_value_key=action_value_key,
            out_keys=out_keys or ["action", action_value_key],
            spec=spec,
            safe=safe,
        )
        self.register_buffer("support", support)

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        logits = td.get(self.action_value_key)  # [..., atoms, actions]
        probs = logits.softmax(-2)
        qvals = (probs * self.support.unsqueeze(-1)).sum(-2)
        idx = qvals.argmax(-1)
        if self.action_space == "one_hot":
            action = torch.nn.functional.one_hot(idx, qvals.shape[-1]).to(torch.int64)
        else:
            action = idx
        td.set(self.out_keys[0], action)
        return td


class DistributionalQValueActor(TensorDictSequential):
    """net → distributional action_value → argmax (reference actors.py:1259)."""

    def __init__(
        self,
        module: nn.Module,
        support: torch.Tensor,
        in_keys: Sequence = ("observation",),
        spec=None,
        safe: bool = False,
        action_space: Optional[str] = None,
        action_value_key: str = "action_value",
        make_log_softmax: bool = True,
    ):
        if isinstance(module, TensorDictModuleBase):
            net = module
        else:
            net = TensorDictModule(module, in_keys=list(in_keys), out_keys=[action_value_key])
        qvalue = DistributionalQValueModule(
            action_space=action_space,
            support=support,
            action_value_key=action_value_key,
            spec=spec,
        )
        super().__init__(net, qvalue)
        self.spec = spec


class ActorValueOperator(TensorDictSequential):
    """Shared-trunk actor-critic (reference actors.py:1415):
    common net → (policy head, value head)."""

    def __init__(
        self,
        common_operator: TensorDictModuleBase,
        policy_operator: TensorDictModuleBase,
        value_operator: TensorDictModuleBase,
    ):
        super().__init__(common_operator, policy_operator, value_operator)
        self.common_operator = common_operator
        self.policy_operator = policy_operator
        self.value_operator = value_operator

    def get_policy_operator(self) -> TensorDictSequential:
        if isinstance(self.policy_operator, (ProbabilisticTensorDictModule, ProbabilisticTensorDictSequential)):
            return ProbabilisticTensorDictSequential(
                self.common_operator, self.policy_operator
            )
        return TensorDictSequential(self.common_operator, self.policy_operator)

    def get_value_operator(self) -> TensorDictSequential:
        return TensorDictSequential(self.common_operator, self.value_operator)

    def get_policy_head(self):
        return self.policy_operator

    def get_value_head(self):
        return self.value_operator


class ActorCriticOperator(ActorValueOperator):
    """Shared trunk, critic reads the action (Q-critic)
    (reference actors.py:1564)."""

    def get_critic_operator(self):
        return TensorDictSequential(
            self.common_operator, self.policy_operator, self.value_operator
        )

    def get_value_operator(self):
        raise RuntimeError(
            "ActorCriticOperator's critic depends on the action; use "
            "get_critic_operator instead"
        )


class ActorCriticWrapper(TensorDictSequential):
    """Independent actor and critic, no shared trunk
    (reference actors.py:1725)."""

    def __init__(self, policy_operator, value_operator):
        super().__init__(policy_operator, value_operator)
        self.policy_operator = policy_operator
        self.value_operator = value_operator

    def get_policy_operator(self):
        return self.policy_operator

    def get_value_operator(self):
        return self.value_operator


class TanhModule(TensorDictModuleBase):
    """Map unbounded outputs into [low, high] with tanh
    (reference actors.py:2066)."""

    def __init__(
        self,
        in_keys: Sequence,
        out_keys: Optional[Sequence] = None,
        spec: Optional[TensorSpec] = None,
        low: float = -1.0,
        high: float = 1.0,
        clamp: bool = False,
    ):
        super().__init__()
        self.in_keys = [unravel_key(k) for k in in_keys]
        self.out_keys = (
            [unravel_key(k) for k in out_keys] if out_keys else list(self.in_keys)
        )
        if spec is not None and hasattr(spec, "low"):
            self.low = spec.low
            self.high = spec.high
        else:
            self.low = torch.as_tensor(low)
            self.high = torch.as_tensor(high)
        self.clamp = clamp

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        for ik, ok in zip(self.in_keys, self.out_keys):
            x = td.get(ik)
            low = self.low.to(x.device) if isinstance(self.low, torch.Tensor) else self.low
            high = self.high.to(x.device) if isinstance(self.high, torch.Tensor) else self.high
            out = (high + low) / 2 + (high - low) / 2 * x.tanh()
            if self.clamp:
                out = out.clamp(low, high)
            td.set(ok, out)
        return td


class MultiStepActorWrapper(TensorDictModuleBase):
    """Execute macro-actions: the wrapped actor emits an action chunk
    [..., T, A]; this wrapper plays it back one step at a time
    (reference actors.py:2280)."""

    def __init__(
        self,
        actor: TensorDictModuleBase,
        action_steps: int,
        action_key: str = "action",
        init_key: str = "is_init",
    ):
        super().__init__()
        self.actor = actor
        self.action_steps = action_steps
        self.action_key = action_key
        self.init_key = init_key
        self.in_keys = list(actor.in_keys)
        self.out_keys = list(actor.out_keys)
        self._queue: Optional[torch.Tensor] = None
        self._ptr = 0

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        need_new = self._queue is None or self._ptr >= self.action_steps
        is_init = td.get(self.init_key, None)
        if is_init is not None and bool(is_init.any()):
            need_new = True
        if need_new:
            out = self.actor(td)
            chunk = out.get(self.action_key)
            self._queue = chunk
            self._ptr = 0
        act = self._queue[..., self._ptr, :]
        self._ptr += 1
        td.set(self.action_key, act)
        return td


class DecisionTransformerInferenceWrapper(TensorDictModuleBase):
    """Maintain the (R, s, a) context window for DT rollouts
    (reference actors.py:1844): keeps the last ``inference_context`` steps,
    feeds them to the DT actor, emits the action for the current step."""

    def __init__(
        self,
        policy: TensorDictModuleBase,
        *,
        inference_context: int = 5,
        observation_key: str = "observation",
        action_key: str = "action",
        return_to_go_key: str = "return_to_go",
    ):
        super().__init__()
        self.policy = policy
        self.inference_context = inference_context
        self.observation_key = observation_key
        self.action_key = action_key
        self.return_to_go_key = return_to_go_key
        self.in_keys = [observation_key, return_to_go_key]
        self.out_keys = [action_key]
        self._obs_hist = None
        self._act_hist = None
        self._rtg_hist = None

    def reset(self):
        self._obs_hist = self._act_hist = self._rtg_hist = None

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        obs = td.get(self.observation_key).unsqueeze(-2)
        rtg = td.get(self.return_to_go_key).unsqueeze(-2)
        if self._obs_hist is None:
            self._obs_hist = obs
            self._rtg_hist = rtg
            self._act_hist = torch.zeros(
                *obs.shape[:-2], 1, self._infer_act_dim(td), device=obs.device
            )
        else:
            self._obs_hist = torch.cat([self._obs_hist, obs], -2)[..., -self.inference_context :, :]
            self._rtg_hist = torch.cat([self._rtg_hist, rtg], -2)[..., -self.inference_context :, :]
            pad_act = torch.zeros_like(self._act_hist[..., :1, :])
            self._act_hist = torch.cat([self._act_hist, pad_act], -2)[..., -self.inference_context :, :]
        seq = td.clone(False)
        seq.set(self.observation_key, self._obs_hist)
        seq.set(self.action_key, self._act_hist)
        seq.set(self.return_to_go_key, self._rtg_hist)
        out = self.policy(seq)
        act_seq = out.get(self.action_key)
        action = act_seq[..., -1, :]
        self._act_hist = torch.cat([self._act_hist[..., :-1, :], action.unsqueeze(-2)], -2)
        td.set(self.action_key, action)
        return td

    def _infer_act_dim(self, td):
        for m in self.policy.modules():
            if hasattr(m, "action_head"):
                return m.action_head.out_features
        raise RuntimeError("cannot infer action dim; pass a DTActor-style policy")


class LMHeadActorValueOperator(ActorValueOperator):
    """Language-model trunk with separate LM head (policy) and value head
    (reference actors.py:2235)."""

    def __init__(self, base_model: TensorDictModuleBase, lm_head: TensorDictModuleBase, value_head: TensorDictModuleBase):
        super().__init__(base_model, lm_head, value_head)
