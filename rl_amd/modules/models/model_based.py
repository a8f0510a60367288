"""World-model components: RSSM prior/posterior, encoders, Dreamer heads.

Reference: pytorch/rl torchrl/modules/models/model_based.py (RSSMPrior,
RSSMPosterior, RSSMRollout, ObsEncoder, ObsDecoder, DreamerActor) and
tensordict_module/world_models.py (WorldModelWrapper:178).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import nn

from ...tensordict import TensorDict, TensorDictBase, TensorDictModuleBase, TensorDictSequential
from .models import MLP

__all__ = [
    "RSSMPrior",
    "RSSMPosterior",
    "RSSMRollout",
    "ObsEncoder",
    "ObsDecoder",
    "DreamerActor",
    "WorldModelWrapper",
]


class ObsEncoder(nn.Module):
    """Pixel encoder (reference ObsEncoder): conv stack → embedding."""

    def __init__(self, channels: int = 32, embed_dim: Optional[int] = None, in_channels: int = 3, device=None):
        super().__init__()
        self.net = nn.Sequential(
            nn.Conv2d(in_channels, channels, 4, 2, device=device),
            nn.ELU(),
            nn.Conv2d(channels, channels * 2, 4, 2, device=device),
            nn.ELU(),
            nn.Conv2d(channels * 2, channels * 4, 4, 2, device=device),
            nn.ELU(),
            nn.Conv2d(channels * 4, channels * 8, 4, 2, device=device),
            nn.ELU(),
        )

    def forward(self, pixels: torch.Tensor) -> torch.Tensor:
        lead = pixels.shape[:-3]
        x = pixels.reshape(-1, *pixels.shape[-3:])
        h = self.net(x)
        h = h.reshape(*lead, -1)
        return h


class ObsDecoder(nn.Module):
    """Latent → pixels (reference ObsDecoder)."""

    def __init__(self, latent_dim: int, channels: int = 32, out_channels: int = 3, device=None):
        super().__init__()
        self.fc = nn.Linear(latent_dim, channels * 8 * 4, device=device)
        self.net = nn.Sequential(
            nn.ConvTranspose2d(channels * 8 * 4, channels * 4, 5, 2, device=device),
            nn.ELU(),
            nn.ConvTranspose2d(channels * 4, channels * 2, 5, 2, device=device),
            nn.ELU(),
            nn.ConvTranspose2d(channels * 2, channels, 6, 2, device=device),
            nn.ELU(),
            nn.ConvTranspose2d(channels, out_channels, 6, 2, device=device),
        )

    def forward(self, latent: torch.Tensor) -> torch.Tensor:
        lead = latent.shape[:-1]
        x = self.fc(latent.reshape(-1, latent.shape[-1]))
        x = x.reshape(-1, x.shape[-1], 1, 1)
        img = self.net(x)
        return img.reshape(*lead, *img.shape[-3:])


class RSSMPrior(nn.Module):
    """Deterministic GRU + stochastic latent prior
    (reference RSSMPrior): (stoch, deter, action) → next (prior stoch,
    deter)."""

    def __init__(
        self,
        action_dim: int,
        stoch_dim: int = 30,
        deter_dim: int = 200,
        hidden_dim: int = 200,
        device=None,
    ):
        super().__init__()
        self.stoch_dim = stoch_dim
        self.deter_dim = deter_dim
        self.pre = nn.Sequential(
            nn.Linear(stoch_dim + action_dim, hidden_dim, device=device), nn.ELU()
        )
        self.gru = nn.GRUCell(hidden_dim, deter_dim, device=device)
        self.post = nn.Sequential(
            nn.Linear(deter_dim, hidden_dim, device=device),
            nn.ELU(),
            nn.Linear(hidden_dim, 2 * stoch_dim, device=device),
        )

    def forward(self, stoch, deter, action):
        lead = stoch.shape[:-1]
        x = self.pre(torch.cat([stoch, action], -1))
        deter_new = self.gru(x.reshape(-1, x.shape[-1]), deter.reshape(-1, self.deter_dim))
        deter_new = deter_new.reshape(*lead, self.deter_dim)
        stats = self.post(deter_new)
        mean, std = stats.chunk(2, -1)
        std = torch.nn.functional.softplus(std) + 0.1
        prior = mean + std * torch.randn_like(std)
        return prior, mean, std, deter_new


class RSSMPosterior(nn.Module):
    """(deter, obs_embedding) → posterior stoch (reference RSSMPosterior)."""

    def __init__(self, deter_dim: int = 200, embed_dim: int = 1024, stoch_dim: int = 30, hidden_dim: int = 200, device=None):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(deter_dim + embed_dim, hidden_dim, device=device),
            nn.ELU(),
            nn.Linear(hidden_dim, 2 * stoch_dim, device=device),
        )

    def forward(self, deter, embed):
        stats = self.net(torch.cat([deter, embed], -1))
        mean, std = stats.chunk(2, -1)
        std = torch.nn.functional.softplus(std) + 0.1
        sample = mean + std * torch.randn_like(std)
        return sample, mean, std


class RSSMRollout(nn.Module):
    """Scan prior+posterior over a [B, T] sequence (reference RSSMRollout):
    produces per-step prior/posterior stats for the KL term and the
    posterior latents for decoding."""

    def __init__(self, prior: RSSMPrior, posterior: RSSMPosterior):
        super().__init__()
        self.prior = prior
        self.posterior = posterior

    def forward(self, embed: torch.Tensor, action: torch.Tensor, is_init: Optional[torch.Tensor] = None):
        B, T = embed.shape[:2]
        device = embed.device
        stoch = torch.zeros(B, self.prior.stoch_dim, device=device)
        deter = torch.zeros(B, self.prior.deter_dim, device=device)
        outs = {k: [] for k in ("prior_mean", "prior_std", "post_mean", "post_std", "stoch", "deter")}
        for t in range(T):
            if is_init is not None:
                m = is_init[:, t].reshape(B, 1).to(embed.dtype)
                stoch = stoch * (1 - m)
                deter = deter * (1 - m)
            _p_sample, p_mean, p_std, deter = self.prior(stoch, deter, action[:, t])
            stoch, q_mean, q_std = self.posterior(deter, embed[:, t])
            outs["prior_mean"].append(p_mean)
            outs["prior_std"].append(p_std)
            outs["post_mean"].append(q_mean)
            outs["post_std"].append(q_std)
            outs["stoch"].append(stoch)
            outs["deter"].append(deter)
        return {k: torch.stack(v, 1) for k, v in outs.items()}


class DreamerActor(nn.Module):
    """Latent-space actor (reference DreamerActor)."""

    def __init__(self, latent_dim: int, action_dim: int, hidden: int = 200, device=None):
        super().__init__()
        self.net = MLP(
            in_features=latent_dim,
            out_features=2 * action_dim,
            num_cells=[hidden, hidden],
            activation_class=nn.ELU,
            device=device,
        )

    def forward(self, latent):
        loc, scale = self.net(latent).chunk(2, -1)
        return loc, torch.nn.functional.softplus(scale) + 1e-4


class WorldModelWrapper(TensorDictSequential):
    """(transition model, reward model) pair
    (reference world_models.py:178)."""

    def __init__(self, transition_model: TensorDictModuleBase, reward_model: TensorDictModuleBase):
        super().__init__(transition_model, reward_model)
        self.transition_model = transition_model
        self.reward_model = reward_model

    def get_transition_model_operator(self):
        return self.transition_model

    def get_reward_operator(self):
        return self.reward_model
