"""Multi-agent networks: MultiAgentMLP, MultiAgentConvNet, VDN/QMIX mixers.

Reference: pytorch/rl torchrl/modules/models/multiagent.py
(MultiAgentMLP:348, MultiAgentConvNet:542, Mixer:756, VDNMixer:935,
QMixer:1008).
"""
from __future__ import annotations

from typing import Optional, Sequence

import torch
from torch import nn

from .models import MLP, ConvNet

__all__ = ["MultiAgentMLP", "MultiAgentConvNet", "Mixer", "VDNMixer", "QMixer"]


class MultiAgentMLP(nn.Module):
    """Per-agent MLPs with optional parameter sharing and centralized
    inputs (reference multiagent.py:348).

    Input [*, n_agents, F]; output [*, n_agents, out].
    """

    def __init__(
        self,
        n_agent_inputs: int,
        n_agent_outputs: int,
        n_agents: int,
        *,
        centralised: bool = False,
        share_params: bool = True,
        depth: int = 2,
        num_cells: int = 64,
        activation_class=nn.Tanh,
        device=None,
    ):
        super().__init__()
        self.n_agents = n_agents
        self.centralised = centralised
        self.share_params = share_params
        in_features = n_agent_inputs * (n_agents if centralised else 1)
        n_nets = 1 if share_params else n_agents
        self.nets = nn.ModuleList(
            [
                MLP(
                    in_features=in_features,
                    out_features=n_agent_outputs,
                    depth=depth,
                    num_cells=num_cells,
                    activation_class=activation_class,
                    device=device,
                )
                for _ in range(n_nets)
            ]
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # x: [*, n_agents, F]
        if self.centralised:
            flat = x.flatten(-2)  # concat all agents
            ins = [flat] * self.n_agents
        else:
            ins = [x[..., i, :] for i in range(self.n_agents)]
        outs = []
        for i in range(self.n_agents):
            net = self.nets[0] if self.share_params else self.nets[i]
            outs.append(net(ins[i]))
        return torch.stack(outs, -2)


class MultiAgentConvNet(nn.Module):
    """Per-agent CNNs (reference multiagent.py:542)."""

    def __init__(self, n_agents: int, *, centralised: bool = False, share_params: bool = True, device=None, **cnn_kwargs):
        super().__init__()
        self.n_agents = n_agents
        self.centralised = centralised
        self.share_params = share_params
        n_nets = 1 if share_params else n_agents
        self.nets = nn.ModuleList(
            [ConvNet(device=device, **cnn_kwargs) for _ in range(n_nets)]
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        outs = []
        for i in range(self.n_agents):
            net = self.nets[0] if self.share_params else self.nets[i]
            outs.append(net(x[..., i, :, :, :]))
        return torch.stack(outs, -2)


class Mixer(nn.Module):
    """ABC: [*, n_agents, 1] local values (+ optional state) → [*, 1]
    (reference multiagent.py:756)."""

    def __init__(self, n_agents: int, needs_state: bool = False):
        super().__init__()
        self.n_agents = n_agents
        self.needs_state = needs_state

    def mix(self, chosen_action_value: torch.Tensor, state: Optional[torch.Tensor]):
        raise NotImplementedError

    def forward(self, chosen_action_value: torch.Tensor, state: Optional[torch.Tensor] = None):
        if chosen_action_value.shape[-2] != self.n_agents:
            raise ValueError(
                f"expected agent dim {self.n_agents}, got {chosen_action_value.shape}"
            )
        return self.mix(chosen_action_value, state)


class VDNMixer(Mixer):
    """Value decomposition: Q_tot = Σ_i Q_i (reference multiagent.py:935)."""

    def __init__(self, n_agents: int, device=None):
        super().__init__(n_agents, needs_state=False)

    def mix(self, chosen_action_value, state=None):
        return chosen_action_value.sum(-2)


class QMixer(Mixer):
    """Monotonic state-conditioned mixing network
    (reference multiagent.py:1008; Rashid et al. 2018): hypernetworks
    produce |abs| weights so ∂Q_tot/∂Q_i ≥ 0."""

    def __init__(
        self,
        state_shape: Sequence[int],
        mixing_embed_dim: int,
        n_agents: int,
        device=None,
    ):
        super().__init__(n_agents, needs_state=True)
        state_dim = int(torch.tensor(list(state_shape)).prod())
        self.embed_dim = mixing_embed_dim
        self.hyper_w1 = nn.Linear(state_dim, n_agents * mixing_embed_dim, device=device)
        self.hyper_b1 = nn.Linear(state_dim, mixing_embed_dim, device=device)
        self.hyper_w2 = nn.Linear(state_dim, mixing_embed_dim, device=device)
        self.hyper_b2 = nn.Sequential(
            nn.Linear(state_dim, mixing_embed_dim, device=device),
            nn.ReLU(),
            nn.Linear(mixing_embed_dim, 1, device=device),
        )

    def mix(self, chosen_action_value, state):
        lead = chosen_action_value.shape[:-2]
        q = chosen_action_value.reshape(-1, 1, self.n_agents)
        s = state.reshape(q.shape[0], -1)
        w1 = self.hyper_w1(s).abs().reshape(-1, self.n_agents, self.embed_dim)
        b1 = self.hyper_b1(s).reshape(-1, 1, self.embed_dim)
        hidden = torch.nn.functional.elu(torch.bmm(q, w1) + b1)
        w2 = self.hyper_w2(s).abs().reshape(-1, self.embed_dim, 1)
        b2 = self.hyper_b2(s).reshape(-1, 1, 1)
        out = torch.bmm(hidden, w2) + b2
        return out.reshape(*lead, 1)
