"""PPO family: PPOLoss, ClipPPOLoss, KLPENPPOLoss.

Reference: pytorch/rl torchrl/objectives/ppo.py (PPOLoss:109,
ClipPPOLoss:1082, KLPENPPOLoss:1458).
"""
from __future__ import annotations

import dataclasses
import math
import os
from typing import Optional, Union

import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import ValueEstimators, distance_loss

__all__ = ["PPOLoss", "ClipPPOLoss", "KLPENPPOLoss"]


class PPOLoss(LossModule):
    """Vanilla policy-gradient surrogate with entropy bonus and critic loss
    (reference ppo.py:109)."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        advantage: str = "advantage"
        value_target: str = "value_target"
        value: str = "state_value"
        sample_log_prob: str = "sample_log_prob"
        action: str = "action"
        reward: tuple = ("next", "reward")
        done: tuple = ("next", "done")
        terminated: tuple = ("next", "terminated")

    default_value_estimator = ValueEstimators.GAE
    out_keys = ["loss_objective", "loss_critic", "loss_entropy", "entropy", "ESS"]

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        critic_network: TensorDictModuleBase,
        *,
        entropy_bonus: bool = True,
        samples_mc_entropy: int = 1,
        entropy_coeff: float = 0.01,
        critic_coeff: float = 1.0,
        loss_critic_type: str = "smooth_l1",
        normalize_advantage: bool = False,
        normalize_advantage_exclude_dims: tuple = (),
        gamma: Optional[float] = None,
        separate_losses: bool = False,
        advantage_key: Optional[str] = None,
        value_target_key: Optional[str] = None,
        reduction: str = "mean",
        clip_value: Optional[float] = None,
    ):
        super().__init__()
        self.convert_to_functional(actor_network, "actor_network")
        self.convert_to_functional(critic_network, "critic_network")
        self.entropy_bonus = entropy_bonus
        self.samples_mc_entropy = samples_mc_entropy
        self.entropy_coeff = entropy_coeff
        self.critic_coeff = critic_coeff
        self.loss_critic_type = loss_critic_type
        self.normalize_advantage = normalize_advantage
        self.reduction = reduction
        self.clip_value = clip_value
        self._gamma_init = gamma
        if advantage_key is not None:
            self._tensor_keys.advantage = advantage_key
        if value_target_key is not None:
            self._tensor_keys.value_target = value_target_key

    @property
    def actor(self):  # legacy alias
        return self.actor_network

    @property
    def critic(self):
        return self.critic_network

    def make_value_estimator(self, value_type=None, **hyperparams):
        if self._gamma_init is not None:
            hyperparams.setdefault("gamma", self._gamma_init)
        return super().make_value_estimator(value_type, **hyperparams)

    def reset(self) -> None:
        pass

    def _log_weight(self, td: TensorDictBase):
        """log π_new(a|s) - log π_old(a|s) plus the distribution."""
        prev_log_prob = td.get(self.tensor_keys.sample_log_prob)
        dist = self.actor_network.get_dist(td.clone(False))
        action = td.get(self.tensor_keys.action)
        log_prob = self._fast_log_prob(dist, action)
        if prev_log_prob.dim() > log_prob.dim():
            prev_log_prob = prev_log_prob.squeeze(-1)
        log_weight = log_prob - prev_log_prob.detach()
        return log_weight, dist, log_prob

    @staticmethod
    def _fast_log_prob(dist, action):
        """Route TanhNormal(-1,1) ratio log-probs through the fused
        HIP kernel (csrc/loss_ops.hip) on GPU — the action carries no
        gradient in the PPO objective."""
        from ..modules.distributions.continuous import TanhNormal

        if (
            action.is_cuda
            and isinstance(dist, TanhNormal)
            and not dist.non_trivial_bounds
        ):
            from .. import ops

            if ops.HAS_HIP_EXT:
                return ops.tanh_normal_logprob(dist.loc, dist.scale, action)
        return dist.log_prob(action)

    def _entropy(self, dist) -> torch.Tensor:
        from ..modules.distributions.continuous import TanhNormal

        if (
            isinstance(dist, TanhNormal)
            and not dist.non_trivial_bounds
            and dist.loc.is_cuda
            and self.samples_mc_entropy == 1
        ):
            from .. import ops

            if ops.HAS_HIP_EXT:
                # fused reparameterized MC estimate (csrc/loss_ops.hip)
                return ops.tanh_normal_entropy(dist.loc, dist.scale)
        try:
            ent = dist.entropy()
        except NotImplementedError:
            x = dist.rsample((self.samples_mc_entropy,))
            ent = -dist.log_prob(x).mean(0)
        return ent

    def loss_critic(self, td: TensorDictBase) -> torch.Tensor:
        target = td.get(self.tensor_keys.value_target)
        td_out = self.critic_network(td.clone(False))
        value = td_out.get(self.tensor_keys.value)
        loss_value = distance_loss(value, target, self.loss_critic_type)
        if self.clip_value is not None:
            old_value = td.get(self.tensor_keys.value, None)
            if old_value is not None:
                value_clipped = old_value + (value - old_value).clamp(
                    -self.clip_value, self.clip_value
                )
                loss_clipped = distance_loss(
                    value_clipped, target, self.loss_critic_type
                )
                loss_value = torch.maximum(loss_value, loss_clipped)
        return loss_value

    def _normalize_adv(self, adv: torch.Tensor) -> torch.Tensor:
        loc = adv.mean()
        scale = adv.std().clamp_min(1e-6)
        return (adv - loc) / scale

    def _loss_critic_reduced(self, td: TensorDictBase, scale: float = 1.0) -> torch.Tensor:
        """``scale * self._reduce(self.loss_critic(td))`` with a fused
        HIP path: smooth-L1 + mean in two launches, one-launch analytic
        backward (csrc/loss_ops.hip) when no value clipping is in play;
        the coefficient is folded into the kernel."""
        if (
            self.clip_value is None
            and self.loss_critic_type in ("smooth_l1", "huber")
            and self.reduction == "mean"
        ):
            target = td.get(self.tensor_keys.value_target)
            value = self.critic_network(td.clone(False)).get(self.tensor_keys.value)
            if (
                value.is_cuda
                and value.dtype in (torch.float32, torch.bfloat16)
                and target.dtype == torch.float32
                and value.numel() == target.numel()
            ):
                from .. import ops

                if ops.HAS_HIP_EXT:
                    return ops.smooth_l1_mean(value, target, scale)
            return scale * self._reduce(
                distance_loss(value, target, self.loss_critic_type)
            )
        return scale * self._reduce(self.loss_critic(td))

    def _reduce(self, x: torch.Tensor) -> torch.Tensor:
        if self.reduction == "mean":
            return x.mean()
        if self.reduction == "sum":
            return x.sum()
        return x

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        advantage = td.get(self.tensor_keys.advantage, None)
        if advantage is None:
            if self.value_estimator is None:
                self.make_value_estimator()
            self.value_estimator(td)
            advantage = td.get(self.tensor_keys.advantage)
        if self.normalize_advantage and advantage.numel() > 1:
            advantage = self._normalize_adv(advantage)
        log_weight, dist, _ = self._log_weight(td)
        lw = log_weight
        if lw.dim() < advantage.dim():
            lw = lw.unsqueeze(-1)
        neg_loss = lw.exp() * advantage
        with torch.no_grad():
            lw_flat = log_weight.reshape(-1)
            ess = lw_flat.exp().sum().pow(2) / lw_flat.mul(2).exp().sum().clamp_min(1e-12)
        out = TensorDict({"loss_objective": -self._reduce(neg_loss)}, batch_size=[])
        if self.entropy_bonus:
            entropy = self._entropy(dist)
            out.set("entropy", entropy.detach().mean())
            out.set("loss_entropy", -self.entropy_coeff * self._reduce(entropy))
        if self.critic_coeff is not None and self.critic_coeff > 0:
            out.set(
                "loss_critic",
                self._loss_critic_reduced(td, float(self.critic_coeff)),
            )
        out.set("ESS", ess / log_weight.numel())
        return out


class ClipPPOLoss(PPOLoss):
    """Clipped-ratio PPO (reference ppo.py:1082)."""

    def __init__(self, actor_network, critic_network, *, clip_epsilon: float = 0.2, **kwargs):
        super().__init__(actor_network, critic_network, **kwargs)
        self.register_buffer("clip_epsilon", torch.as_tensor(clip_epsilon))

    @property
    def _clip_bounds(self):
        return (
            math.log1p(-float(self.clip_epsilon)),
            math.log1p(float(self.clip_epsilon)),
        )

    def _mega_probe(self):
        """One-time introspection of the actor for the mega-fused head
        path (csrc/loss_ops.hip ppo_head kernels): ProbabilisticActor
        over Sequential(body…, NormalParamExtractor(biased_softplus))
        with a TanhNormal(-1, 1) distribution.  Returns a dict of
        pieces, or False when the structure does not match."""
        cached = self.__dict__.get("_mega_info", None)
        if cached is not None:
            return cached
        info = False
        try:
            from ..modules.models.models import NormalParamExtractor
            from ..modules.distributions.continuous import TanhNormal
            from ..tensordict.nn import (
                ProbabilisticTensorDictModule,
                TensorDictModule,
            )

            actor = self.actor_network
            mods = list(actor.module)
            if len(mods) == 2:
                tdm, prob = mods
                dk = getattr(prob, "distribution_kwargs", None) or {}
                low, high = dk.get("low", -1.0), dk.get("high", 1.0)
                if (
                    isinstance(prob, ProbabilisticTensorDictModule)
                    and prob.distribution_class is TanhNormal
                    and float(low) == -1.0
                    and float(high) == 1.0
                    and list(prob.in_keys) == ["loc", "scale"]
                    and isinstance(tdm, TensorDictModule)
                    and list(tdm.out_keys) == ["loc", "scale"]
                    and len(tdm.in_keys) == 1
                    and isinstance(tdm.module, torch.nn.Sequential)
                    and len(tdm.module) >= 2
                    and isinstance(tdm.module[-1], NormalParamExtractor)
                    and tdm.module[-1]._inv_softplus_bias is not None
                ):
                    ext = tdm.module[-1]
                    info = {
                        "body": list(tdm.module)[:-1],
                        "obs_key": tdm.in_keys[0],
                        "sp_bias": float(ext._inv_softplus_bias),
                        "scale_lb": float(ext.scale_lb),
                        "ac_pair": None,
                    }
                    # actor+critic dual-kernel pair: both bodies are
                    # FusedMLP3 over the same observation key
                    try:
                        from ..ops import FusedMLP3

                        crit = self.critic_network
                        body = info["body"]
                        if (
                            len(body) == 1
                            and isinstance(body[0], FusedMLP3)
                            and isinstance(crit, TensorDictModule)
                            and list(crit.in_keys) == [info["obs_key"]]
                            and isinstance(crit.module, FusedMLP3)
                            and crit.module.lin3.out_features == 1
                        ):
                            info["ac_pair"] = (body[0], crit.module)
                    except Exception:
                        pass
        except Exception:
            info = False
        self.__dict__["_mega_info"] = info
        return info

    def _mega_head_loss(self, td, advantage, normalize):
        """Fused actor-loss path: raw head -> (loss_objective,
        loss_entropy, entropy, ESS, clip_fraction) in 3-5 launches +
        1-kernel backward.  Returns None when ineligible."""
        if not (
            self.entropy_bonus
            and self.samples_mc_entropy == 1
            and self.reduction == "mean"
            and not advantage.requires_grad
            and advantage.dtype == torch.float32
        ):
            return None
        from .. import ops

        if not ops.HAS_HIP_EXT:
            return None
        info = self._mega_probe()
        if not info:
            return None
        obs = td.get(info["obs_key"], None)
        action = td.get(self.tensor_keys.action, None)
        prev_lp = td.get(self.tensor_keys.sample_log_prob, None)
        if obs is None or action is None or prev_lp is None or not obs.is_cuda:
            return None
        if action.dtype != torch.float32 or prev_lp.dtype != torch.float32:
            return None
        A = action.shape[-1]
        N = action.numel() // A
        if advantage.numel() != N or prev_lp.numel() != N:
            return None
        value = None
        ac = info.get("ac_pair")
        if (
            ac is not None
            and self.clip_value is None
            and self.loss_critic_type in ("smooth_l1", "huber")
            and self.critic_coeff is not None
            and self.critic_coeff > 0
        ):
            target = td.get(self.tensor_keys.value_target, None)
            O = obs.shape[-1]
            if (
                target is not None
                and target.dtype == torch.float32
                and target.numel() == N
                and ops.actor_critic_mlp3_ok(ac[0], ac[1], O)
                and getattr(ac[0].lin1, "_bf16_cache", False)
                and getattr(ac[1].lin1, "_bf16_cache", False)
            ):
                A = action.shape[-1]
                if (
                    ac[0].lin3.out_features == 2 * A
                    and os.environ.get("RL_AMD_MERGED_LOSS", "1") != "0"
                ):
                    # the fully-merged path: MLPs AND loss in one
                    # launch pair (csrc/fused_mlp.hip acloss kernels)
                    eps = self.__dict__.get("_mega_eps")
                    if eps is None or eps.shape != (N, A):
                        eps = torch.randn(N, A, device=action.device,
                                          dtype=torch.float32)
                    lo, hi = self._clip_bounds
                    (loss_obj, loss_ent, ent_mean, ess, clip_frac,
                     loss_act, loss_crit, loss_total) = ops.actor_critic_loss(
                        obs.reshape(N, O), ac[0], ac[1],
                        action.reshape(N, A).float(),
                        prev_lp.reshape(N), advantage.reshape(N),
                        target.reshape(N), eps,
                        sp_bias=info["sp_bias"], scale_lb=info["scale_lb"],
                        lo=lo, hi=hi,
                        entropy_coeff=float(self.entropy_coeff),
                        critic_scale=float(self.critic_coeff),
                        normalize=normalize,
                        # trainer-precomputed per-minibatch stats (one
                        # batched launch per epoch instead of two per
                        # minibatch) — see GraphedPPO._update_phase
                        stats_in=self.__dict__.get("_mega_stats"),
                        gradsq_out=self.__dict__.get("_gradsq_part"),
                    )
                    if self.__dict__.get("_gradsq_part") is not None:
                        # signals the trainer that this step's sq
                        # partials WILL be written by the backward
                        self.__dict__["_gradsq_armed"] = True
                    return (loss_obj, loss_ent, ent_mean, ess, clip_frac,
                            loss_act, loss_crit, loss_total)
                head, value = ops.actor_critic_mlp3(
                    obs.reshape(N, O), ac[0], ac[1]
                )
        if value is None:
            head = obs
            for m in info["body"]:
                head = m(head)
        if head.dtype not in (torch.float32, torch.bfloat16):
            return None
        if head.shape[-1] != 2 * A:
            return None
        eps = torch.randn(N, A, device=action.device, dtype=torch.float32)
        lo, hi = self._clip_bounds
        (loss_obj, loss_ent, ent_mean, ess, clip_frac, loss_act, loss_crit,
         loss_total) = ops.ppo_head_loss(
            head.reshape(N, 2 * A),
            action.reshape(N, A).float(),
            prev_lp.reshape(N),
            advantage.reshape(N),
            eps,
            sp_bias=info["sp_bias"],
            scale_lb=info["scale_lb"],
            lo=lo,
            hi=hi,
            entropy_coeff=float(self.entropy_coeff),
            normalize=normalize,
            value=value,
            value_target=(td.get(self.tensor_keys.value_target)
                          if value is not None else None),
            critic_scale=float(self.critic_coeff or 0.0),
        )
        if value is None:
            loss_crit = loss_total = None
        return (loss_obj, loss_ent, ent_mean, ess, clip_frac, loss_act,
                loss_crit, loss_total)

    def _loss_critic_side_stream(self, td: TensorDictBase) -> torch.Tensor:
        """Critic loss forward on a side HIP stream: the critic chain
        (MLP forward, smooth-L1, and — because autograd replays each
        node on its forward stream — the whole critic backward and its
        wgrads) runs CONCURRENTLY with the actor chain on the main
        stream.  Inside a hipGraph capture this records as parallel
        graph branches.  Trainers must join ``_side_streams`` after
        backward before touching gradients (GraphedPPO does)."""
        coeff = float(self.critic_coeff)
        n = td.batch_size[0] if td.batch_dims >= 1 else 0
        if not torch.cuda.is_available() or n < 32768:
            # below ~32K rows the two branches' kernels just contend
            # for CUs (measured: -4% at 16K rows, +3% at 64K)
            return self._loss_critic_reduced(td, coeff)
        cs = self.__dict__.get("_crit_stream")
        if cs is None:
            cs = torch.cuda.Stream()
            self.__dict__["_crit_stream"] = cs
            self.__dict__["_side_streams"] = [cs]
        cur = torch.cuda.current_stream()
        cs.wait_stream(cur)
        with torch.cuda.stream(cs):
            loss_critic = self._loss_critic_reduced(td, coeff)
        cur.wait_stream(cs)
        if not torch.cuda.is_current_stream_capturing():
            loss_critic.record_stream(cur)
        return loss_critic

    @property
    def _side_streams(self):
        return self.__dict__.get("_side_streams_list", self.__dict__.get("_side_streams", []))

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        advantage = td.get(self.tensor_keys.advantage, None)
        if advantage is None:
            if self.value_estimator is None:
                self.make_value_estimator()
            self.value_estimator(td)
            advantage = td.get(self.tensor_keys.advantage)
        normalize = self.normalize_advantage and advantage.numel() > 1
        mega = self._mega_head_loss(td, advantage, normalize)
        if mega is not None:
            (loss_obj, loss_ent, ent_mean, ess, clip_frac, loss_act,
             loss_critic, loss_total) = mega
            out = TensorDict(
                {
                    "loss_objective": loss_obj,
                    "clip_fraction": clip_frac,
                    "ESS": ess,
                    "entropy": ent_mean,
                    "loss_entropy": loss_ent,
                    # pre-summed loss_objective + loss_entropy: trainers
                    # that recognize it (GraphedPPO) skip the eager adds
                    "_loss_actor": loss_act,
                },
                batch_size=[],
            )
            if loss_critic is not None:
                out.set("loss_critic", loss_critic)
                # kernel-side total (actor + scaled critic): trainers
                # that recognize it skip every eager add
                out.set("_loss_total", loss_total)
            elif self.critic_coeff is not None and self.critic_coeff > 0:
                out.set("loss_critic", self._loss_critic_side_stream(td))
            return out
        log_weight, dist, _ = self._log_weight(td)
        if log_weight.dim() < advantage.dim():
            log_weight = log_weight.unsqueeze(-1)
        fused = None
        if (
            self.reduction == "mean"
            and log_weight.is_cuda
            and log_weight.dtype == torch.float32
            and advantage.dtype == torch.float32
            and not advantage.requires_grad
            and log_weight.numel() == advantage.numel()
        ):
            from .. import ops

            if ops.HAS_HIP_EXT:
                # fused objective + diagnostics + advantage
                # normalization (csrc/loss_ops.hip)
                fused = ops.ppo_clip_objective(
                    log_weight, advantage, *self._clip_bounds, normalize
                )
        if fused is not None:
            loss_objective, ess_per_sample, clip_fraction = fused
        else:
            if normalize:
                advantage = self._normalize_adv(advantage)
            ratio = log_weight.exp()
            gain1 = ratio * advantage
            ratio_clamped = log_weight.clamp(*self._clip_bounds).exp()
            gain2 = ratio_clamped * advantage
            gain = torch.minimum(gain1, gain2)
            with torch.no_grad():
                lw_flat = log_weight.reshape(-1)
                ess = lw_flat.exp().sum().pow(2) / lw_flat.mul(2).exp().sum().clamp_min(1e-12)
                clip_fraction = (ratio_clamped != ratio).float().mean()
                ess_per_sample = ess / log_weight.numel()
            loss_objective = -self._reduce(gain)
        out = TensorDict(
            {
                "loss_objective": loss_objective,
                "clip_fraction": clip_fraction,
                "ESS": ess_per_sample,
            },
            batch_size=[],
        )
        if self.entropy_bonus:
            entropy = self._entropy(dist)
            out.set("entropy", entropy.detach().mean())
            out.set("loss_entropy", -self.entropy_coeff * self._reduce(entropy))
        if self.critic_coeff is not None and self.critic_coeff > 0:
            out.set(
                "loss_critic",
                self._loss_critic_reduced(td, float(self.critic_coeff)),
            )
        return out


class KLPENPPOLoss(PPOLoss):
    """KL-penalty PPO with adaptive β (reference ppo.py:1458)."""

    def __init__(
        self,
        actor_network,
        critic_network,
        *,
        dtarg: float = 0.01,
        beta: float = 1.0,
        increment: float = 2.0,
        decrement: float = 0.5,
        samples_mc_kl: int = 1,
        **kwargs,
    ):
        super().__init__(actor_network, critic_network, **kwargs)
        self.dtarg = dtarg
        self._beta_init = beta
        self.register_buffer("beta", torch.as_tensor(beta))
        self.increment = increment
        self.decrement = decrement
        self.samples_mc_kl = samples_mc_kl

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        advantage = td.get(self.tensor_keys.advantage, None)
        if advantage is None:
            if self.value_estimator is None:
                self.make_value_estimator()
            self.value_estimator(td)
            advantage = td.get(self.tensor_keys.advantage)
        if self.normalize_advantage and advantage.numel() > 1:
            advantage = self._normalize_adv(advantage)
        log_weight, dist, log_prob = self._log_weight(td)
        lw = log_weight
        if lw.dim() < advantage.dim():
            lw = lw.unsqueeze(-1)
        neg_loss = lw.exp() * advantage
        # MC KL(π_old ‖ π_new) ≈ E_old[log w]⁻
        kl = -log_weight.mean()
        if kl > self.dtarg * 1.5:
            self.beta.data.mul_(self.increment)
        elif kl < self.dtarg / 1.5:
            self.beta.data.mul_(self.decrement)
        out = TensorDict(
            {
                "loss_objective": -self._reduce(neg_loss) + self.beta * kl,
                "kl": kl.detach(),
                "beta": self.beta.clone(),
            },
            batch_size=[],
        )
        if self.entropy_bonus:
            entropy = self._entropy(dist)
            out.set("entropy", entropy.detach().mean())
            out.set("loss_entropy", -self.entropy_coeff * self._reduce(entropy))
        if self.critic_coeff is not None and self.critic_coeff > 0:
            out.set(
                "loss_critic",
                self._loss_critic_reduced(td, float(self.critic_coeff)),
            )
        return out
