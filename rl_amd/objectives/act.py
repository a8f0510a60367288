"""ACT and diffusion-policy losses (+ DDPM actor module).

Reference: pytorch/rl torchrl/objectives/act.py:19 (ACTLoss),
diffusion_bc.py:17 (DiffusionBCLoss); DDPM DiffusionActor in
torchrl/modules/tensordict_module/actors.py:2705-2869; ACT model
torchrl/modules/models/act.py.
"""
from __future__ import annotations

import dataclasses
import math
from typing import Optional

import torch
from torch import nn

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import LossModule
from .utils import distance_loss

__all__ = ["ACTLoss", "DiffusionBCLoss", "DiffusionActor"]


class ACTLoss(LossModule):
    """Action-chunking transformer loss (reference act.py:19): L1 action
    reconstruction + KL of the CVAE latent.

    The actor maps (observation, [action chunk during training]) →
    (action_pred, latent mu, latent logvar).
    """

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        action_pred: str = "action_pred"
        mu: str = "latent_mu"
        logvar: str = "latent_logvar"

    def __init__(self, actor_network: TensorDictModuleBase, *, kl_weight: float = 10.0, reduction: str = "mean"):
        super().__init__()
        self.actor_network = actor_network
        self.kl_weight = kl_weight
        self.reduction = reduction

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        td = self.actor_network(tensordict.clone(False))
        pred = td.get(keys.action_pred)
        target = tensordict.get(keys.action)
        l1 = (pred - target).abs().mean()
        out = TensorDict({"loss_l1": l1}, batch_size=[])
        mu = td.get(keys.mu, None)
        logvar = td.get(keys.logvar, None)
        if mu is not None and logvar is not None:
            kl = (-0.5 * (1 + logvar - mu.pow(2) - logvar.exp())).sum(-1).mean()
            out.set("loss_kl", self.kl_weight * kl)
            out.set("loss", l1 + self.kl_weight * kl)
        else:
            out.set("loss", l1)
        return out


def _cosine_alphas(T: int) -> torch.Tensor:
    s = 0.008
    t = torch.linspace(0, T, T + 1) / T
    f = torch.cos((t + s) / (1 + s) * math.pi / 2).pow(2)
    alpha_bar = (f / f[0]).clamp(1e-5, 1.0)
    return alpha_bar


class DiffusionActor(TensorDictModuleBase):
    """DDPM policy head (reference actors.py:2705 DiffusionActor): a noise
    predictor ε_θ(a_t, t, obs) trained by DiffusionBCLoss; sampling runs
    the reverse process from Gaussian noise — every denoise step is a
    small fused-able MLP call, GPU-friendly at large batch."""

    def __init__(
        self,
        noise_net: nn.Module,
        *,
        action_dim: int,
        num_steps: int = 16,
        observation_key: str = "observation",
        action_key: str = "action",
    ):
        super().__init__()
        self.noise_net = noise_net
        self.action_dim = action_dim
        self.num_steps = num_steps
        self.observation_key = observation_key
        self.action_key = action_key
        alpha_bar = _cosine_alphas(num_steps)
        self.register_buffer("alpha_bar", alpha_bar)
        self.in_keys = [observation_key]
        self.out_keys = [action_key]

    def _eps(self, a_t: torch.Tensor, t: torch.Tensor, obs: torch.Tensor) -> torch.Tensor:
        t_feat = (t.float() / self.num_steps).reshape(-1, 1).expand(a_t.shape[0], 1)
        return self.noise_net(torch.cat([a_t, t_feat, obs], -1))

    @torch.no_grad()
    def forward(self, td: TensorDictBase) -> TensorDictBase:
        obs = td.get(self.observation_key)
        B = obs.shape[0]
        a = torch.randn(B, self.action_dim, device=obs.device)
        ab = self.alpha_bar.to(obs.device)
        for step in reversed(range(1, self.num_steps + 1)):
            t = torch.full((B,), step, device=obs.device)
            eps = self._eps(a, t, obs)
            ab_t = ab[step]
            ab_prev = ab[step - 1]
            a0 = (a - (1 - ab_t).sqrt() * eps) / ab_t.sqrt()
            a0 = a0.clamp(-1, 1)
            if step > 1:
                noise = torch.randn_like(a)
                beta = (1 - ab_t / ab_prev).clamp(1e-6, 0.999)
                a = ab_prev.sqrt() * a0 + (1 - ab_prev - beta).clamp_min(0).sqrt() * eps + beta.sqrt() * noise
            else:
                a = a0
        td.set(self.action_key, a)
        return td

    def training_targets(self, action: torch.Tensor, obs: torch.Tensor):
        """(noised action, timestep, true noise) triple for the BC loss."""
        B = action.shape[0]
        t = torch.randint(1, self.num_steps + 1, (B,), device=action.device)
        ab = self.alpha_bar.to(action.device)[t].unsqueeze(-1)
        noise = torch.randn_like(action)
        a_t = ab.sqrt() * action + (1 - ab).sqrt() * noise
        return a_t, t, noise


class DiffusionBCLoss(LossModule):
    """Denoising-score-matching behavior cloning
    (reference diffusion_bc.py:17): MSE between predicted and true noise
    at a random diffusion timestep."""

    @dataclasses.dataclass
    class _AcceptedKeys:
        action: str = "action"
        observation: str = "observation"

    def __init__(self, actor: DiffusionActor, *, reduction: str = "mean"):
        super().__init__()
        self.actor = actor
        self.reduction = reduction

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        keys = self.tensor_keys
        action = tensordict.get(keys.action)
        obs = tensordict.get(keys.observation)
        a_t, t, noise = self.actor.training_targets(action, obs)
        pred = self.actor._eps(a_t, t, obs)
        loss = distance_loss(pred, noise, "l2").mean()
        return TensorDict({"loss_diffusion": loss}, batch_size=[])
