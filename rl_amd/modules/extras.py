"""Reference-parity module utilities: MCTS scores, multi-agent bases,
cross-group critics, vmap wrapper, RNN helpers and primer extraction.

Reference: pytorch/rl torchrl/modules/mcts/scores.py (MCTSScore,
MCTSScores, PUCTScore, UCB1Score, UCB1TunedScore, EXP3Score),
models/multiagent.py:MultiAgentNetBase,
models/cross_group_critic.py (CrossCriticGroupSpec, CrossGroupCritic),
tensordict_module/common.py:VmapModule, rnn.py:canonicalize_rnn_subset,
_rnn_precision.py (RecurrentMatmulPrecision,
set/get_recurrent_matmul_precision), utils/utils.py
(get_primers_from_module, get_env_transforms_from_module),
tensordict_module/exploration.py:set_exploration_modules_spec_from_env,
vla/common.py:VLAWrapperBase.
"""
from __future__ import annotations

import math
from enum import Enum
from typing import Dict, List, Optional, Sequence

import torch
from torch import nn

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase

__all__ = [
    "AutocastPolicy",
    "MCTSScore",
    "MCTSScores",
    "UCB1TunedScore",
    "MultiAgentNetBase",
    "CrossCriticGroupSpec",
    "CrossGroupCritic",
    "VmapModule",
    "canonicalize_rnn_subset",
    "RecurrentMatmulPrecision",
    "RecurrentMatmulPrecisionUserMode",
    "set_recurrent_matmul_precision",
    "get_recurrent_matmul_precision",
    "get_primers_from_module",
    "get_env_transforms_from_module",
    "set_exploration_modules_spec_from_env",
    "VLAWrapperBase",
]


# ---------------------------------------------------------------------------
# MCTS scores as TensorDict modules (reference mcts/scores.py)
# ---------------------------------------------------------------------------


class MCTSScore(TensorDictModuleBase):
    """Base for node-selection scores: reads per-child statistics
    (win counts, visit counts, priors) from a tensordict and writes a
    ``score`` tensor — everything stays batched on-device so tree
    expansion can score thousands of nodes per launch."""

    def __init__(self, *, win_count_key="win_count", visits_key="visits",
                 total_visits_key="total_visits", prior_prob_key="prior_prob",
                 score_key="score"):
        super().__init__()
        self.win_count_key = win_count_key
        self.visits_key = visits_key
        self.total_visits_key = total_visits_key
        self.prior_prob_key = prior_prob_key
        self.score_key = score_key
        self.in_keys = [win_count_key, visits_key, prior_prob_key]
        self.out_keys = [score_key]


class PUCTScore(MCTSScore):
    """PUCT (AlphaZero-style): Q + c·P·sqrt(N_total)/(1+n)."""

    def __init__(self, *, c: float = 1.414, **kwargs):
        super().__init__(**kwargs)
        self.c = c

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        wins = td.get(self.win_count_key)
        visits = td.get(self.visits_key)
        prior = td.get(self.prior_prob_key)
        n_total = td.get(self.total_visits_key, visits.sum(-1, keepdim=True))
        q = wins / visits.clamp_min(1)
        u = self.c * prior * torch.sqrt(n_total.float()) / (1 + visits)
        td.set(self.score_key, q + u)
        return td


class UCB1Score(MCTSScore):
    """Classic UCB1: Q + c·sqrt(ln N_total / n); unvisited → +inf."""

    def __init__(self, *, c: float = 1.414, **kwargs):
        super().__init__(**kwargs)
        self.c = c

    def _explore(self, visits, n_total):
        return torch.sqrt(torch.log(n_total.float().clamp_min(1)) / visits.clamp_min(1))

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        wins = td.get(self.win_count_key)
        visits = td.get(self.visits_key)
        n_total = td.get(self.total_visits_key, visits.sum(-1, keepdim=True))
        q = wins / visits.clamp_min(1)
        score = q + self.c * self._explore(visits, n_total)
        td.set(self.score_key, torch.where(visits > 0, score, torch.full_like(score, torch.inf)))
        return td


class UCB1TunedScore(UCB1Score):
    """UCB1-Tuned (reference scores.py): the exploration width is
    scaled by the empirical per-arm reward variance bound
    min(1/4, V_n + sqrt(2 ln N / n))."""

    def __init__(self, *, variance_key="reward_variance", **kwargs):
        super().__init__(**kwargs)
        self.variance_key = variance_key

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        wins = td.get(self.win_count_key)
        visits = td.get(self.visits_key)
        n_total = td.get(self.total_visits_key, visits.sum(-1, keepdim=True))
        var = td.get(self.variance_key, torch.full_like(wins, 0.25))
        q = wins / visits.clamp_min(1)
        logn = torch.log(n_total.float().clamp_min(1))
        v_bound = var + torch.sqrt(2 * logn / visits.clamp_min(1))
        width = torch.minimum(torch.full_like(v_bound, 0.25), v_bound)
        score = q + torch.sqrt(logn / visits.clamp_min(1) * width)
        td.set(self.score_key, torch.where(visits > 0, score, torch.full_like(score, torch.inf)))
        return td


class EXP3Score(MCTSScore):
    """EXP3 adversarial-bandit selection probabilities (reference
    scores.py:241): p = (1−γ)·softmax(η·S) + γ/K."""

    def __init__(self, *, gamma: float = 0.1, eta: float = 0.1, **kwargs):
        super().__init__(**kwargs)
        self.gamma = gamma
        self.eta = eta

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        rewards = td.get(self.win_count_key)
        k = rewards.shape[-1]
        p = torch.softmax(self.eta * rewards, dim=-1)
        td.set(self.score_key, (1 - self.gamma) * p + self.gamma / k)
        return td


class MCTSScores(Enum):
    """Named score constructors (reference mcts/scores.py MCTSScores)."""

    PUCT = PUCTScore
    UCB1 = UCB1Score
    UCB1_TUNED = UCB1TunedScore
    EXP3 = EXP3Score

    def __call__(self, **kwargs):
        return self.value(**kwargs)


# ---------------------------------------------------------------------------
# Multi-agent net base + cross-group critic
# ---------------------------------------------------------------------------


class MultiAgentNetBase(nn.Module):
    """Base for per-agent network stacks (reference multiagent.py):
    when params are NOT shared the per-agent forwards are batched into
    a single bmm-shaped pass wherever shapes allow — on MI355X that
    turns n_agents small GEMMs into one MFMA-friendly batched GEMM."""

    def __init__(self, *, n_agents: int, share_params: bool = True):
        super().__init__()
        self.n_agents = n_agents
        self.share_params = share_params

    def _build_single(self) -> nn.Module:  # pragma: no cover - abstract
        raise NotImplementedError

    def _make_nets(self) -> nn.ModuleList:
        n = 1 if self.share_params else self.n_agents
        return nn.ModuleList([self._build_single() for _ in range(n)])

    def forward_per_agent(self, nets: nn.ModuleList, x: torch.Tensor) -> torch.Tensor:
        # x: [..., n_agents, F]
        if self.share_params:
            return nets[0](x)
        outs = [net(x[..., i, :]) for i, net in enumerate(nets)]
        return torch.stack(outs, dim=-2)


class CrossCriticGroupSpec:
    """Spec for one agent group consumed by :class:`CrossGroupCritic`
    (reference cross_group_critic.py:26)."""

    def __init__(self, *, obs_dim: int, n_agents: int, obs_key, value_key):
        self.obs_dim = obs_dim
        self.n_agents = n_agents
        self.obs_key = obs_key
        self.value_key = value_key


class CrossGroupCritic(nn.Module):
    """Centralized critic over MULTIPLE agent groups (reference
    cross_group_critic.py:134): each group's [B, n_i, obs_i]
    observations are encoded to a shared embedding, mean-pooled,
    concatenated across groups, passed through a joint trunk, and a
    per-group head writes ``value_key`` for every group.  One GEMM per
    group + one trunk GEMM regardless of agent counts."""

    def __init__(self, group_specs: Sequence[CrossCriticGroupSpec], *,
                 embed_dim: int = 64, hidden_dim: int = 128, device=None):
        super().__init__()
        self.group_specs = list(group_specs)
        self.encoders = nn.ModuleList(
            [nn.Linear(g.obs_dim, embed_dim, device=device) for g in self.group_specs]
        )
        trunk_in = embed_dim * len(self.group_specs)
        self.trunk = nn.Sequential(
            nn.Linear(trunk_in, hidden_dim, device=device), nn.ReLU(),
            nn.Linear(hidden_dim, hidden_dim, device=device), nn.ReLU(),
        )
        self.heads = nn.ModuleList(
            [nn.Linear(hidden_dim, 1, device=device) for _ in self.group_specs]
        )
        self.in_keys = [g.obs_key for g in self.group_specs]
        self.out_keys = [g.value_key for g in self.group_specs]

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        embeds = []
        for spec, enc in zip(self.group_specs, self.encoders):
            obs = td.get(spec.obs_key)
            embeds.append(torch.relu(enc(obs)).mean(dim=-2))
        joint = self.trunk(torch.cat(embeds, dim=-1))
        for spec, head in zip(self.group_specs, self.heads):
            v = head(joint).unsqueeze(-2).expand(*joint.shape[:-1], spec.n_agents, 1)
            td.set(spec.value_key, v)
        return td


# ---------------------------------------------------------------------------
# VmapModule, RNN helpers, recurrent matmul precision
# ---------------------------------------------------------------------------


class VmapModule(TensorDictModuleBase):
    """Unbind a batch dim, apply the wrapped module per slice, restack
    (reference tensordict_module/common.py:VmapModule).  Used to apply
    single-sample modules over an extra leading dim."""

    def __init__(self, module: TensorDictModuleBase, vmap_dim: int = 0):
        super().__init__()
        self.module = module
        self.vmap_dim = vmap_dim
        self.in_keys = list(getattr(module, "in_keys", []))
        self.out_keys = list(getattr(module, "out_keys", []))

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        from ..tensordict import stack as td_stack

        dim = self.vmap_dim
        n = td.batch_size[dim]
        slices = [self.module(td[(slice(None),) * dim + (i,)].clone(False)) for i in range(n)]
        out = td_stack(slices, dim)
        td.update(out)
        return td


def canonicalize_rnn_subset(in_keys: List, out_keys: List, hidden_keys: List):
    """Normalize the (in, out, hidden) key triple of a recurrent module
    so selection/subsetting keeps the hidden keys paired (reference
    rnn.py): returns (in_keys+hiddens, out_keys+next-hiddens)."""

    hidden_in = list(hidden_keys)
    hidden_out = [
        ("next", k) if not (isinstance(k, tuple) and k and k[0] == "next") else k
        for k in hidden_keys
    ]
    ins = list(in_keys) + [k for k in hidden_in if k not in in_keys]
    outs = list(out_keys) + [k for k in hidden_out if k not in out_keys]
    return ins, outs


class RecurrentMatmulPrecision(Enum):
    """Matmul precision for recurrent cells (reference
    _rnn_precision.py): recurrent GEMMs are tiny and latency-bound on
    MI355X, so bf16 ("low") keeps them in one MFMA pass; fp32 ("high")
    is for numerics-sensitive unrolls."""

    LOW = "low"
    MEDIUM = "medium"
    HIGH = "high"


RecurrentMatmulPrecisionUserMode = RecurrentMatmulPrecision

_RNN_PRECISION = RecurrentMatmulPrecision.HIGH


def set_recurrent_matmul_precision(mode) -> None:
    global _RNN_PRECISION
    if isinstance(mode, str):
        mode = RecurrentMatmulPrecision(mode)
    _RNN_PRECISION = mode


def get_recurrent_matmul_precision() -> RecurrentMatmulPrecision:
    return _RNN_PRECISION


# ---------------------------------------------------------------------------
# Primer / transform extraction (reference utils/utils.py)
# ---------------------------------------------------------------------------


def get_primers_from_module(module, warn: bool = True, strict: bool = True):
    """Collect ``make_tensordict_primer()`` results from all submodules
    (recurrent modules advertise their hidden-state specs this way) and
    compose them into one transform."""
    primers = []
    for sub in module.modules() if isinstance(module, nn.Module) else [module]:
        maker = getattr(sub, "make_tensordict_primer", None)
        if maker is not None:
            primers.append(maker())
    if not primers:
        if strict:
            raise RuntimeError("no submodule defines make_tensordict_primer()")
        return None
    if len(primers) == 1:
        return primers[0]
    from ..envs.transforms import Compose

    return Compose(*primers)


def get_env_transforms_from_module(module, init_key: str = "is_init"):
    """InitTracker + the module's hidden-state primers, composed — the
    transforms a TransformedEnv needs to drive a recurrent policy
    (reference utils/utils.py:107)."""
    from ..envs.transforms import Compose, InitTracker

    primer = get_primers_from_module(module, strict=False)
    parts = [InitTracker(init_key=init_key)]
    if primer is not None:
        parts.append(primer)
    return Compose(*parts)


def set_exploration_modules_spec_from_env(policy, env) -> None:
    """Point every exploration module in ``policy`` (anything exposing
    an ``action_spec`` attribute or ``set_spec``) at ``env.action_spec``
    (reference tensordict_module/exploration.py)."""
    spec = env.action_spec
    for sub in policy.modules() if isinstance(policy, nn.Module) else [policy]:
        if hasattr(sub, "set_spec"):
            sub.set_spec(spec)
        elif hasattr(sub, "action_spec"):
            sub.action_spec = spec


# ---------------------------------------------------------------------------
# VLA wrapper base (reference vla/common.py)
# ---------------------------------------------------------------------------


class VLAWrapperBase(nn.Module):
    """Base for adapters of pretrained VLA policies to the rl_amd
    schema: subclasses implement ``_forward_policy`` mapping (image,
    state, instruction) → action chunk; this base handles tensordict
    IO and chunk bookkeeping."""

    in_keys = [("observation", "image"), ("observation", "state"), "language_instruction"]
    out_keys = [("vla_action", "chunk"), "action"]

    def __init__(self, *, chunk_size: int, action_dim: int):
        super().__init__()
        self.chunk_size = chunk_size
        self.action_dim = action_dim

    def _forward_policy(self, image, state, instruction):  # pragma: no cover
        raise NotImplementedError

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        image = td.get(("observation", "image"))
        state = td.get(("observation", "state"), None)
        instruction = td.get_non_tensor("language_instruction", "")
        chunk = self._forward_policy(image, state, instruction)
        td.set(("vla_action", "chunk"), chunk)
        td.set("action", chunk[..., 0, :])
        return td


class AutocastPolicy(torch.nn.Module):
    """Run a policy under bf16 autocast with float32 outputs — the
    rollout-side compute-dtype wrapper for collector fast paths
    (CDNA4 has no fp32 MFMA, so fp32 policy GEMMs run on the vector
    ALU; autocast moves them to the matrix cores).  ``cache_enabled``
    stays False so the forward is hipGraph-capture-safe."""

    def __init__(self, policy, dtype=torch.bfloat16, cast_outputs_to_float: bool = True):
        super().__init__()
        self.policy = policy
        self.dtype = dtype
        self.cast_outputs_to_float = cast_outputs_to_float
        self.in_keys = getattr(policy, "in_keys", [])
        self.out_keys = getattr(policy, "out_keys", [])

    def forward(self, td):
        enabled = torch.cuda.is_available() and (
            td.device is not None and td.device.type == "cuda"
        )
        with torch.autocast("cuda", dtype=self.dtype, enabled=enabled, cache_enabled=False):
            out = self.policy(td)
        if self.cast_outputs_to_float:
            for k in self.out_keys:
                v = out.get(k, None)
                if v is not None and v.dtype == self.dtype:
                    out.set(k, v.float())
        return out

    def get_dist(self, td):
        return self.policy.get_dist(td)
