"""LLM objectives: GRPO, DAPO/CISPO variants, SFT, MC advantage.

Reference: pytorch/rl torchrl/objectives/llm/grpo.py (GRPOLoss:355,
DAPO:953, CISPOLoss:1004, MCAdvantage:1028), sft.py:104.
"""
from __future__ import annotations

import dataclasses
from typing import Optional

import torch

from ...tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from ..common import LossModule

__all__ = ["GRPOLoss", "DAPO", "CISPOLoss", "SFTLoss", "MCAdvantage"]


class MCAdvantage:
    """Group-relative Monte-Carlo advantage (reference grpo.py:1028):
    within each prompt group of G responses, adv_i = (r_i - mean) / std."""

    def __init__(self, grpo_size: int, reward_key=("next", "reward"), advantage_key: str = "advantage", eps: float = 1e-4):
        self.grpo_size = grpo_size
        self.reward_key = reward_key
        self.advantage_key = advantage_key
        self.eps = eps

    def __call__(self, td: TensorDictBase) -> TensorDictBase:
        r = td.get(self.reward_key)
        flat = r.reshape(-1, self.grpo_size)
        mean = flat.mean(-1, keepdim=True)
        std = flat.std(-1, keepdim=True).clamp_min(self.eps)
        adv = ((flat - mean) / std).reshape(r.shape)
        td.set(self.advantage_key, adv)
        return td


class GRPOLoss(LossModule):
    """Token-level clipped policy-gradient with KL-to-reference penalty
    (reference grpo.py:355; Shao et al. 2024).

    Expects: ``log_probs`` (behavior, from generation), ``advantage``
    (per-sequence), the actor in log-prob mode recomputes ``log_probs``
    of ``tokens_response`` under current weights; ``mask`` marks valid
    response tokens.
    """

    @dataclasses.dataclass
    class _AcceptedKeys:
        sample_log_prob: str = "log_probs"
        advantage: str = "advantage"
        ref_log_prob: str = "ref_log_probs"
        mask: str = "attention_mask_response"

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        *,
        clip_epsilon: float = 0.2,
        kl_to_ref_coeff: Optional[float] = None,
        entropy_coeff: float = 0.0,
        reduction: str = "mean",
    ):
        super().__init__()
        self.actor_network = actor_network
        self.clip_epsilon = clip_epsilon
        self.kl_to_ref_coeff = kl_to_ref_coeff
        self.entropy_coeff = entropy_coeff
        self.reduction = reduction

    def _reduce(self, x, mask=None):
        if mask is not None:
            x = x * mask
            denom = mask.sum().clamp_min(1)
            return x.sum() / denom if self.reduction == "mean" else x.sum()
        return x.mean() if self.reduction == "mean" else x.sum()

    def _ratio_clip_gain(self, log_w, adv):
        ratio = log_w.exp()
        g1 = ratio * adv
        g2 = ratio.clamp(1 - self.clip_epsilon, 1 + self.clip_epsilon) * adv
        return torch.minimum(g1, g2), ratio

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        old_lp = td.get(keys.sample_log_prob)
        adv = td.get(keys.advantage)
        while adv.dim() < old_lp.dim():
            adv = adv.unsqueeze(-1)
        # recompute log-probs under current weights
        self.actor_network.generate = False
        new_td = self.actor_network(td.clone(False))
        new_lp = new_td.get(keys.sample_log_prob)
        log_w = new_lp - old_lp.detach()
        gain, ratio = self._ratio_clip_gain(log_w, adv)
        mask = td.get(keys.mask, None)
        if mask is None:
            resp = td.get("tokens_response", None)
            if resp is not None:
                mask = torch.ones_like(new_lp)
        out = TensorDict(
            {
                "loss_objective": -self._reduce(gain, mask),
                "clip_fraction": ((ratio - 1).abs() > self.clip_epsilon).float().mean(),
                "ESS": (log_w.exp().sum().pow(2) / log_w.mul(2).exp().sum().clamp_min(1e-9) / log_w.numel()),
            },
            batch_size=[],
        )
        if self.kl_to_ref_coeff is not None:
            ref_lp = td.get(keys.ref_log_prob, None)
            if ref_lp is not None:
                # unbiased k3 estimator (Schulman)
                lr = ref_lp - new_lp
                kl = lr.exp() - 1 - lr
                out.set("loss_kl_to_ref", self.kl_to_ref_coeff * self._reduce(kl, mask))
                out.set("kl_to_ref", self._reduce(kl.detach(), mask))
        if self.entropy_coeff:
            ent = -new_lp
            out.set("loss_entropy", -self.entropy_coeff * self._reduce(ent, mask))
        return out


class DAPO(GRPOLoss):
    """Decoupled-clip GRPO (reference grpo.py:953): asymmetric clip range
    (clip-higher)."""

    def __init__(self, actor_network, *, clip_epsilon_low: float = 0.2, clip_epsilon_high: float = 0.28, **kwargs):
        super().__init__(actor_network, clip_epsilon=clip_epsilon_low, **kwargs)
        self.clip_epsilon_high = clip_epsilon_high

    def _ratio_clip_gain(self, log_w, adv):
        ratio = log_w.exp()
        g1 = ratio * adv
        g2 = ratio.clamp(1 - self.clip_epsilon, 1 + self.clip_epsilon_high) * adv
        return torch.minimum(g1, g2), ratio


class CISPOLoss(GRPOLoss):
    """Clipped-importance-sampling PG (reference grpo.py:1004): clip the
    IS weight itself, keep the gradient through log-probs."""

    def _ratio_clip_gain(self, log_w, adv):
        ratio = log_w.exp()
        w = ratio.detach().clamp(1 - self.clip_epsilon, 1 + self.clip_epsilon)
        return w * log_w * adv, ratio


class SFTLoss(LossModule):
    """Supervised fine-tuning NLL over response tokens
    (reference sft.py:104)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        log_probs: str = "log_probs"
        mask: str = "attention_mask_response"

    def __init__(self, actor_network: TensorDictModuleBase, *, reduction: str = "mean"):
        super().__init__()
        self.actor_network = actor_network
        self.reduction = reduction

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        self.actor_network.generate = False
        out = self.actor_network(td)
        lp = out.get(self.tensor_keys.log_probs)
        mask = td.get(self.tensor_keys.mask, None)
        nll = -lp
        if mask is not None:
            nll = nll * mask
            loss = nll.sum() / mask.sum().clamp_min(1)
        else:
            loss = nll.mean() if self.reduction == "mean" else nll.sum()
        return TensorDict({"loss_sft": loss}, batch_size=[])


class DistillationLoss(LossModule):
    """Token-level knowledge distillation for LLM policies
    (reference torchrl/objectives/llm/distillation.py:105).

    Minimizes a KL between student and teacher per-token log-probs with
    the k3 estimator (same approximation as the GRPO/SFT KL
    regularizers).  Student log-probs come from running
    ``actor_network`` in log-prob mode; teacher log-probs are read from
    the tensordict (``ref_log_probs``) — written by an offline scoring
    pass or a teacher wrapper.

    ``kl_direction``:
      * ``"reverse"`` (default): KL(student || teacher) — on-policy
        (tokens sampled from the student).
      * ``"forward"``: KL(teacher || student) — teacher-generated data.

    Gradients flow through the student log-probs only.
    """

    @dataclasses.dataclass
    class _AcceptedKeys:
        log_probs: str = "log_probs"
        ref_log_probs: str = "ref_log_probs"
        mask: str = "attention_mask_response"

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        *,
        kl_direction: str = "reverse",
        coeff: float = 1.0,
    ):
        if kl_direction not in ("reverse", "forward"):
            raise ValueError("kl_direction must be 'reverse' or 'forward'")
        super().__init__()
        self.actor_network = actor_network
        self.kl_direction = kl_direction
        self.coeff = coeff

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = tensordict.clone(False)
        teacher_lp = td.get(keys.ref_log_probs)
        self.actor_network.generate = False
        out = self.actor_network(td)
        student_lp = out.get(keys.log_probs)
        # k3 estimator: KL(p||q) ~ E_p[ exp(lq-lp) - 1 - (lq-lp) ]
        if self.kl_direction == "reverse":
            lr = teacher_lp.detach() - student_lp
        else:
            lr = student_lp - teacher_lp.detach()
            # forward KL on teacher samples: gradient still through student
        kl = lr.exp() - 1 - lr
        mask = td.get(keys.mask, None)
        if mask is not None:
            kl = kl * mask
            loss = kl.sum() / mask.sum().clamp_min(1)
        else:
            loss = kl.mean()
        return TensorDict(
            {
                "loss_distill": self.coeff * loss,
                "kl_to_teacher": loss.detach(),
            },
            batch_size=[],
        )


@dataclasses.dataclass
class LLMLossOutput:
    """Typed loss container base (reference llm losses *LossOutput)."""

    loss_objective: "torch.Tensor" = None


@dataclasses.dataclass
class GRPOLossOutput(LLMLossOutput):
    clip_fraction: "torch.Tensor" = None
    ESS: "torch.Tensor" = None
    loss_kl_to_ref: "torch.Tensor" = None
    kl_to_ref: "torch.Tensor" = None
    loss_entropy: "torch.Tensor" = None


@dataclasses.dataclass
class DAPOLossOutput(GRPOLossOutput):
    pass


@dataclasses.dataclass
class CISPOLossOutput(GRPOLossOutput):
    pass


@dataclasses.dataclass
class SFTLossOutput:
    loss_sft: "torch.Tensor" = None


@dataclasses.dataclass
class DistillationLossOutput:
    loss_distill: "torch.Tensor" = None
    kl_to_teacher: "torch.Tensor" = None
