"""MAPPO / IPPO — multi-agent PPO variants.

Reference: pytorch/rl torchrl/objectives/multiagent/mappo.py
(MAPPOLoss:83, IPPOLoss:213): ClipPPO with multi-agent key layout
(per-agent actions under ("agents", ...)); MAPPO uses a centralized
critic, IPPO per-agent critics — both reduce to the same loss math over
the agent-dim-carrying TensorDict.
"""
from __future__ import annotations

from typing import Optional

from ...tensordict import TensorDictModuleBase
from ..ppo import ClipPPOLoss

__all__ = ["MAPPOLoss", "IPPOLoss"]


class MAPPOLoss(ClipPPOLoss):
    """Centralized-critic multi-agent PPO (reference mappo.py:83)."""

    def __init__(
        self,
        actor_network: TensorDictModuleBase,
        critic_network: TensorDictModuleBase,
        *,
        clip_epsilon: float = 0.2,
        **kwargs,
    ):
        super().__init__(actor_network, critic_network, clip_epsilon=clip_epsilon, **kwargs)
        # multi-agent default key layout
        self.set_keys(
            action=("agents", "action"),
            sample_log_prob=("agents", "sample_log_prob"),
            advantage=("agents", "advantage"),
            value_target=("agents", "value_target"),
            value=("agents", "state_value"),
            reward=("next", "agents", "reward"),
            done=("next", "agents", "done"),
            terminated=("next", "agents", "terminated"),
        )

    def set_keys(self, **kwargs):
        # tolerate tuples in dataclass fields
        for k, v in kwargs.items():
            setattr(self._tensor_keys, k, v)
        return self


class IPPOLoss(MAPPOLoss):
    """Independent-critic multi-agent PPO (reference mappo.py:213) — same
    loss; the difference is the critic module the user passes (per-agent
    value heads instead of a centralized one)."""
