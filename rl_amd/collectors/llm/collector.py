"""LLMCollector — rollout = generate (reference:
pytorch/rl torchrl/collectors/llm/base.py:26)."""
from __future__ import annotations

from typing import Callable, Optional

from ...envs.common import EnvBase
from ...tensordict import TensorDictBase, stack as td_stack
from ..collectors import BaseCollector

__all__ = ["LLMCollector"]


class LLMCollector(BaseCollector):
    """Iterate conversations: env.reset → policy.generate → env.step,
    yielding one TensorDict per dialogue batch."""

    def __init__(
        self,
        env: EnvBase,
        policy: Callable[[TensorDictBase], TensorDictBase],
        *,
        dialog_turns_per_batch: int = 1,
        total_dialog_turns: int = -1,
        yield_only_last_steps: bool = False,
    ):
        self.env = env
        self.policy = policy
        self.dialog_turns_per_batch = dialog_turns_per_batch
        self.total_dialog_turns = (
            total_dialog_turns if total_dialog_turns > 0 else float("inf")
        )
        self.yield_only_last_steps = yield_only_last_steps
        self._turns = 0

    def iterator(self):
        while self._turns < self.total_dialog_turns:
            steps = []
            carrier = self.env.reset()
            for _ in range(self.dialog_turns_per_batch):
                carrier = self.policy(carrier)
                carrier, next_root = self.env.step_and_maybe_reset(carrier)
                steps.append(carrier.clone(False))
                carrier = next_root
                self._turns += 1
                if self._turns >= self.total_dialog_turns:
                    break
            if self.yield_only_last_steps:
                yield steps[-1]
            else:
                yield td_stack(steps, len(self.env.batch_size)) if len(steps) > 1 else steps[0]

    def shutdown(self, timeout=None):
        self.env.close()
