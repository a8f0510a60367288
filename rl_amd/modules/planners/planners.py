"""Model-predictive planners: CEM and MPPI.

Reference: pytorch/rl torchrl/modules/planners/ (MPCPlannerBase
common.py:19, CEMPlanner cem.py:17, MPPIPlanner mppi.py:19).
Both roll candidate action sequences through a (model-based) env and
optimize the first action — all candidates are batched through the env in
one shot (GPU-resident envs evaluate thousands of rollouts per call).
"""
from __future__ import annotations

from typing import Optional

import torch

from ...envs.common import EnvBase
from ...tensordict import TensorDict, TensorDictBase, TensorDictModuleBase

__all__ = ["MPCPlannerBase", "CEMPlanner", "MPPIPlanner"]


class MPCPlannerBase(TensorDictModuleBase):
    """Plans an action by simulating the env (reference common.py:19)."""

    def __init__(self, env: EnvBase, action_key: str = "action"):
        super().__init__()
        self.env = env
        self.action_key = action_key
        self.in_keys = list(env.observation_spec.keys(True, True))
        self.out_keys = [action_key]

    def planning(self, td: TensorDictBase) -> torch.Tensor:
        raise NotImplementedError

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        action = self.planning(td)
        spec = self.env.action_spec
        if hasattr(spec, "project"):
            action = spec.project(action)
        td.set(self.action_key, action)
        return td

    def _rollout_candidates(self, td: TensorDictBase, actions: torch.Tensor):
        """actions: [N, T, *action_shape] → total reward [N].

        Requires a STATELESS env (state carried in the TensorDict, e.g.
        :class:`~rl_amd.envs.ModelBasedEnvBase`) — the candidate batch is
        simulated by expanding the start td to N copies."""
        from ...envs.utils import step_mdp

        N, T = actions.shape[:2]
        rewards = torch.zeros(N, 1, device=actions.device)
        state = td.clone(True).expand(N, *td.batch_size).clone()
        if len(td.batch_size):
            state = state.flatten(0, len(td.batch_size))
        for t in range(T):
            state.set(self.action_key, actions[:, t])
            state = self.env.step(state)
            rewards = rewards + state.get(("next", "reward")).reshape(N, -1).sum(-1, keepdim=True)
            state = step_mdp(state)
        return rewards.squeeze(-1)


class CEMPlanner(MPCPlannerBase):
    """Cross-entropy method (reference cem.py:17): iteratively refit a
    Gaussian over action sequences to the top-k candidates."""

    def __init__(
        self,
        env: EnvBase,
        planning_horizon: int,
        optim_steps: int,
        num_candidates: int,
        top_k: int,
        action_key: str = "action",
    ):
        super().__init__(env, action_key)
        self.planning_horizon = planning_horizon
        self.optim_steps = optim_steps
        self.num_candidates = num_candidates
        self.top_k = top_k

    def planning(self, td: TensorDictBase) -> torch.Tensor:
        spec = self.env.action_spec
        act_shape = spec.shape[len(self.env.batch_size):]
        device = td.device
        T, N, K = self.planning_horizon, self.num_candidates, self.top_k
        mean = torch.zeros(T, *act_shape, device=device)
        std = torch.ones(T, *act_shape, device=device)
        for _ in range(self.optim_steps):
            actions = mean + std * torch.randn(N, T, *act_shape, device=device)
            if hasattr(spec, "low"):
                actions = actions.clamp(
                    spec.low.to(device) if isinstance(spec.low, torch.Tensor) else spec.low,
                    spec.high.to(device) if isinstance(spec.high, torch.Tensor) else spec.high,
                )
            returns = self._rollout_candidates(td, actions)
            top = returns.topk(K).indices
            elite = actions[top]
            mean = elite.mean(0)
            std = elite.std(0).clamp_min(1e-4)
        return mean[0].expand(*self.env.batch_size, *act_shape) if self.env.batch_size else mean[0]


class MPPIPlanner(MPCPlannerBase):
    """Model-predictive path integral (reference mppi.py:19): softmax
    return-weighted average of sampled action sequences."""

    def __init__(
        self,
        env: EnvBase,
        planning_horizon: int,
        optim_steps: int,
        num_candidates: int,
        top_k: int,
        temperature: float = 1.0,
        action_key: str = "action",
    ):
        super().__init__(env, action_key)
        self.planning_horizon = planning_horizon
        self.optim_steps = optim_steps
        self.num_candidates = num_candidates
        self.top_k = top_k
        self.temperature = temperature

    def planning(self, td: TensorDictBase) -> torch.Tensor:
        spec = self.env.action_spec
        act_shape = spec.shape[len(self.env.batch_size):]
        device = td.device
        T, N, K = self.planning_horizon, self.num_candidates, self.top_k
        mean = torch.zeros(T, *act_shape, device=device)
        std = torch.ones(T, *act_shape, device=device)
        for _ in range(self.optim_steps):
            actions = mean + std * torch.randn(N, T, *act_shape, device=device)
            returns = self._rollout_candidates(td, actions)
            top_r, top_i = returns.topk(K)
            w = torch.softmax(top_r / self.temperature, 0).reshape(K, *[1] * (actions.dim() - 1))
            elite = actions[top_i]
            mean = (w * elite).sum(0)
            std = (
                (w * (elite - mean.unsqueeze(0)).pow(2)).sum(0).sqrt().clamp_min(1e-4)
            )
        return mean[0].expand(*self.env.batch_size, *act_shape) if self.env.batch_size else mean[0]
