"""Evaluator — periodic policy evaluation in a background thread.

Reference: pytorch/rl torchrl/collectors/_evaluator.py:99 (Evaluator,
_ThreadEvalBackend:971).
"""
from __future__ import annotations

import threading
import time
from typing import Callable, Optional

import torch

from ..envs.common import EnvBase
from ..envs.utils import ExplorationType, set_exploration_type
from ..tensordict import TensorDictBase

__all__ = ["Evaluator"]


class Evaluator:
    """Runs deterministic eval rollouts every ``eval_interval`` frames.

    ``evaluate()`` can be called inline, or ``maybe_evaluate(frames)``
    from the training loop; with ``async_eval=True`` the rollout happens
    in a thread on a snapshot of the policy weights.
    """

    def __init__(
        self,
        env: EnvBase,
        policy: Callable,
        *,
        num_episodes: int = 3,
        max_steps: int = 1000,
        eval_interval: int = 10_000,
        exploration_type: ExplorationType = ExplorationType.DETERMINISTIC,
        logger=None,
        log_key: str = "eval/reward",
        async_eval: bool = False,
    ):
        self.env = env
        self.policy = policy
        self.num_episodes = num_episodes
        self.max_steps = max_steps
        self.eval_interval = eval_interval
        self.exploration_type = exploration_type
        self.logger = logger
        self.log_key = log_key
        self.async_eval = async_eval
        self._last_eval = 0
        self._thread: Optional[threading.Thread] = None
        self.last_result: Optional[float] = None
        # async eval runs on a SNAPSHOT of the weights (reference
        # _ThreadEvalBackend): the learner may update mid-rollout
        self._eval_policy = None
        if async_eval and hasattr(policy, "state_dict"):
            import copy

            self._eval_policy = copy.deepcopy(policy)

    def evaluate(self, step: Optional[int] = None, policy=None) -> float:
        policy = policy if policy is not None else self.policy
        rewards = []
        with set_exploration_type(self.exploration_type), torch.no_grad():
            for _ in range(self.num_episodes):
                rollout = self.env.rollout(
                    self.max_steps, policy=policy, break_when_any_done=True
                )
                rewards.append(rollout.get(("next", "reward")).sum().item())
        mean_r = sum(rewards) / len(rewards)
        self.last_result = mean_r
        if self.logger is not None:
            self.logger.log_scalar(self.log_key, mean_r, step=step)
        return mean_r

    def maybe_evaluate(self, collected_frames: int) -> Optional[float]:
        if collected_frames - self._last_eval < self.eval_interval:
            return None
        self._last_eval = collected_frames
        if self.async_eval:
            if self._thread is not None and self._thread.is_alive():
                return self.last_result
            eval_policy = self.policy
            if self._eval_policy is not None:
                self._eval_policy.load_state_dict(self.policy.state_dict())
                eval_policy = self._eval_policy
            self._thread = threading.Thread(
                target=self.evaluate, args=(collected_frames, eval_policy),
                daemon=True,
            )
            self._thread.start()
            return self.last_result
        return self.evaluate(collected_frames)

    def shutdown(self):
        if self._thread is not None:
            self._thread.join(timeout=30)
