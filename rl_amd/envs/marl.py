"""Multi-agent grouping utilities.

Reference: pytorch/rl torchrl/envs/utils.py:1026 (MarlGroupMapType) and
utils.py:1101 (check_marl_grouping): a group map assigns every agent of
a multi-agent env to exactly one named group; grouped agents share a
stacked entry in the tensordict (one batched tensor per group — the
layout that keeps per-group policy forwards as single fused GEMMs on
the GPU instead of per-agent small launches).
"""
from __future__ import annotations

from enum import Enum
from typing import Dict, List

__all__ = ["MarlGroupMapType", "check_marl_grouping"]


class MarlGroupMapType(Enum):
    """Canonical ways of grouping agents.

    - ``ALL_IN_ONE_GROUP``: every agent in one ``"agents"`` group — one
      stacked tensor, one policy forward for the whole team.
    - ``ONE_GROUP_PER_AGENT``: each agent is its own group (keyed by the
      agent name) — for heterogeneous teams with per-agent specs.
    """

    ALL_IN_ONE_GROUP = 1
    ONE_GROUP_PER_AGENT = 2

    def get_group_map(self, agent_names: List[str]) -> Dict[str, List[str]]:
        if self is MarlGroupMapType.ALL_IN_ONE_GROUP:
            return {"agents": list(agent_names)}
        return {name: [name] for name in agent_names}


def check_marl_grouping(group_map: Dict[str, List[str]], agent_names: List[str]) -> None:
    """Validate a group map: non-empty groups, every agent appears in
    exactly one group, and no unknown agents (reference utils.py:1101).
    Raises ``ValueError`` on any violation."""
    if not isinstance(group_map, dict) or not group_map:
        raise ValueError("group_map must be a non-empty dict[str, list[str]]")
    seen: Dict[str, str] = {}
    for group, members in group_map.items():
        if not members:
            raise ValueError(f"group '{group}' is empty")
        for agent in members:
            if agent not in agent_names:
                raise ValueError(f"agent '{agent}' in group '{group}' is not in agent_names")
            if agent in seen:
                raise ValueError(
                    f"agent '{agent}' appears in groups '{seen[agent]}' and '{group}'"
                )
            seen[agent] = group
    missing = [a for a in agent_names if a not in seen]
    if missing:
        raise ValueError(f"agents {missing} are not assigned to any group")
