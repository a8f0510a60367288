"""Decision Transformer — GPT-2 backbone over (return, state, action)
tokens.

Reference: pytorch/rl torchrl/modules/models/decision_transformer.py
(GPT2-based, config-driven) and models.py:1507 (DTActor head).
Offline-friendly: built from a transformers GPT2Config (no downloads).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
from torch import nn

__all__ = ["DecisionTransformer", "DTActor"]


class DecisionTransformer(nn.Module):
    """(observation [B,T,O], action [B,T,A], return_to_go [B,T,1]) →
    hidden states for action prediction."""

    @dataclass
    class default_config:
        n_embd: int = 128
        n_layer: int = 3
        n_head: int = 4
        n_positions: int = 1024
        resid_pdrop: float = 0.1
        attn_pdrop: float = 0.1

    def __init__(
        self,
        state_dim: int,
        action_dim: int,
        config: Optional[dict] = None,
        device=None,
    ):
        super().__init__()
        from transformers import GPT2Config, GPT2Model

        cfg = dict(
            n_embd=128, n_layer=3, n_head=4, n_positions=1024,
            resid_pdrop=0.1, attn_pdrop=0.1,
        )
        if config:
            cfg.update(config)
        hidden = cfg["n_embd"]
        gpt_cfg = GPT2Config(
            vocab_size=1,
            n_embd=hidden,
            n_layer=cfg["n_layer"],
            n_head=cfg["n_head"],
            n_positions=cfg["n_positions"],
            resid_pdrop=cfg["resid_pdrop"],
            attn_pdrop=cfg["attn_pdrop"],
        )
        self.transformer = GPT2Model(gpt_cfg)
        self.hidden_dim = hidden
        self.embed_state = nn.Linear(state_dim, hidden, device=device)
        self.embed_action = nn.Linear(action_dim, hidden, device=device)
        self.embed_return = nn.Linear(1, hidden, device=device)
        self.embed_ln = nn.LayerNorm(hidden, device=device)
        if device is not None:
            self.transformer = self.transformer.to(device)

    def forward(
        self,
        observation: torch.Tensor,
        action: torch.Tensor,
        return_to_go: torch.Tensor,
    ) -> torch.Tensor:
        B, T = observation.shape[:2]
        s = self.embed_state(observation)
        a = self.embed_action(action)
        r = self.embed_return(return_to_go)
        # interleave (R, s, a) per step → sequence of 3T tokens
        tokens = torch.stack([r, s, a], dim=2).reshape(B, 3 * T, self.hidden_dim)
        tokens = self.embed_ln(tokens)
        out = self.transformer(inputs_embeds=tokens).last_hidden_state
        # hidden state at the STATE token predicts the action
        return out.reshape(B, T, 3, self.hidden_dim)[:, :, 1]


class DTActor(nn.Module):
    """DT + action head (reference models.py:1507/1609)."""

    def __init__(self, state_dim: int, action_dim: int, transformer_config: Optional[dict] = None, device=None):
        super().__init__()
        self.transformer = DecisionTransformer(
            state_dim, action_dim, transformer_config, device=device
        )
        self.action_head = nn.Linear(
            self.transformer.hidden_dim, action_dim, device=device
        )

    def forward(self, observation, action, return_to_go):
        h = self.transformer(observation, action, return_to_go)
        return torch.tanh(self.action_head(h))


class OnlineDTActor(nn.Module):
    """Online Decision Transformer actor (reference models.py:1507):
    Gaussian head over the DT hidden state — returns (mu, std) for the
    stochastic policy of Zheng et al. 2022."""

    def __init__(self, state_dim: int, action_dim: int, transformer_config: Optional[dict] = None, device=None):
        super().__init__()
        self.transformer = DecisionTransformer(
            state_dim, action_dim, transformer_config, device=device
        )
        h = self.transformer.hidden_dim
        self.mean_head = nn.Linear(h, action_dim, device=device)
        self.logstd_head = nn.Linear(h, action_dim, device=device)
        # logstd bounds (paper appendix): keep std in a sane range
        self.log_std_min = -20.0
        self.log_std_max = 2.0

    def forward(self, observation, action, return_to_go):
        h = self.transformer(observation, action, return_to_go)
        mu = torch.tanh(self.mean_head(h))
        log_std = self.logstd_head(h).clamp(self.log_std_min, self.log_std_max)
        return mu, log_std.exp()
