"""ACT — Action Chunking with Transformers backbone.

Reference: pytorch/rl torchrl/modules/models/act.py:14 (ACTModel),
Zhao et al. 2023: CVAE encoder over (obs, action chunk) → style latent
z; DETR-style transformer decoder with learned per-timestep action
queries reconstructs the chunk.  Inference uses z = 0 (prior mean).
Pairs with :class:`~rl_amd.objectives.ACTLoss` (reconstruction + KL).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
from torch import nn

__all__ = ["ACTModel"]


def _sinusoidal_pe(length: int, dim: int, device=None) -> torch.Tensor:
    pos = torch.arange(length, device=device).unsqueeze(1).float()
    div = torch.exp(
        torch.arange(0, dim, 2, device=device).float() * (-math.log(10000.0) / dim)
    )
    pe = torch.zeros(length, dim, device=device)
    pe[:, 0::2] = torch.sin(pos * div)
    pe[:, 1::2] = torch.cos(pos * div)
    return pe


class ACTModel(nn.Module):
    """CVAE + transformer action-chunk predictor.

    ``forward(observation, action_chunk=None)`` returns
    ``(pred_chunk [B, T, A], mu [B, Z], logvar [B, Z])`` — mu/logvar are
    zeros at inference (prior mean latent).
    """

    def __init__(
        self,
        obs_dim: int,
        action_dim: int,
        chunk_size: int,
        hidden_dim: int = 256,
        nheads: int = 8,
        num_encoder_layers: int = 2,
        num_decoder_layers: int = 4,
        latent_dim: int = 32,
        dim_feedforward: int = 1024,
        device=None,
    ):
        super().__init__()
        self.chunk_size = chunk_size
        self.latent_dim = latent_dim
        H = hidden_dim
        self.obs_proj = nn.Linear(obs_dim, H, device=device)
        self.action_proj = nn.Linear(action_dim, H, device=device)
        self.cls_token = nn.Parameter(torch.randn(1, 1, H, device=device) * 0.02)
        enc_layer = nn.TransformerEncoderLayer(
            H, nheads, dim_feedforward, batch_first=True, device=device
        )
        self.cvae_encoder = nn.TransformerEncoder(enc_layer, num_encoder_layers)
        self.latent_head = nn.Linear(H, 2 * latent_dim, device=device)
        self.latent_proj = nn.Linear(latent_dim, H, device=device)
        dec_layer = nn.TransformerDecoderLayer(
            H, nheads, dim_feedforward, batch_first=True, device=device
        )
        self.decoder = nn.TransformerDecoder(dec_layer, num_decoder_layers)
        self.action_queries = nn.Parameter(
            torch.randn(1, chunk_size, H, device=device) * 0.02
        )
        self.action_head = nn.Linear(H, action_dim, device=device)
        self.register_buffer(
            "enc_pe", _sinusoidal_pe(chunk_size + 2, H, device=device)
        )

    def encode(
        self, obs_tok: torch.Tensor, action_chunk: torch.Tensor
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        B = obs_tok.shape[0]
        act_tok = self.action_proj(action_chunk)  # [B, T, H]
        seq = torch.cat(
            [self.cls_token.expand(B, 1, -1), obs_tok.unsqueeze(1), act_tok], dim=1
        )
        seq = seq + self.enc_pe[: seq.shape[1]]
        enc = self.cvae_encoder(seq)
        mu, logvar = self.latent_head(enc[:, 0]).chunk(2, dim=-1)
        return mu, logvar

    def forward(
        self,
        observation: torch.Tensor,
        action_chunk: Optional[torch.Tensor] = None,
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        B = observation.shape[0]
        obs_tok = self.obs_proj(observation)  # [B, H]
        if action_chunk is not None:
            mu, logvar = self.encode(obs_tok, action_chunk)
            z = mu + torch.randn_like(mu) * (0.5 * logvar).exp()
        else:
            mu = torch.zeros(B, self.latent_dim, device=observation.device)
            logvar = torch.zeros_like(mu)
            z = mu
        memory = torch.stack([obs_tok, self.latent_proj(z)], dim=1)  # [B, 2, H]
        queries = self.action_queries.expand(B, -1, -1)
        dec = self.decoder(queries, memory)
        return self.action_head(dec), mu, logvar
