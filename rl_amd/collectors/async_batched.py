"""AsyncBatchedCollector — batched policy over an AsyncEnvPool.

Reference: pytorch/rl torchrl/collectors/_async_batched.py:118: env-pool
steps complete out of order; the policy runs on whatever sub-batch is
ready (min_get), maximizing device utilization when env step times vary.
"""
from __future__ import annotations

from typing import Callable, Optional, Sequence

import torch

from ..envs.async_envs import AsyncEnvPool
from ..envs.utils import ExplorationType, set_exploration_type
from ..tensordict import TensorDictBase, stack as td_stack
from .collectors import BaseCollector

__all__ = ["AsyncBatchedCollector"]


class AsyncBatchedCollector(BaseCollector):
    def __init__(
        self,
        env_makers: Sequence[Callable],
        policy: Optional[Callable] = None,
        *,
        frames_per_batch: int,
        total_frames: int = -1,
        min_get: Optional[int] = None,
        exploration_type: ExplorationType = ExplorationType.RANDOM,
        backend: str = "threading",
    ):
        self.pool = AsyncEnvPool(env_makers, backend=backend)
        self.policy = policy
        self.frames_per_batch = frames_per_batch
        self.total_frames = total_frames if total_frames > 0 else float("inf")
        self.min_get = min_get or max(1, len(env_makers) // 2)
        self.exploration_type = exploration_type
        self._frames = 0
        self.closed = False

    def iterator(self):
        n = self.pool.num_envs
        reset_td = self.pool.reset()
        # submit a first step for every env
        def _rand_actions(td, ids):
            acts = td_stack(
                [self.pool._envs[i].rand_action(td[j].clone(False)) for j, i in enumerate(ids)],
                0,
            )
            td.update(acts)
            return td

        with set_exploration_type(self.exploration_type), torch.no_grad():
            if self.policy is not None:
                reset_td = self.policy(reset_td)
            else:
                reset_td = _rand_actions(reset_td, list(range(n)))
            self.pool.async_step_send(reset_td)
            collected = []
            collected_frames_in_batch = 0
            while self._frames < self.total_frames:
                ready = self.pool.async_step_recv(min_get=self.min_get)
                collected.append(ready.clone(False))
                k = ready.batch_size[0]
                collected_frames_in_batch += k
                self._frames += k
                # act on the envs that just finished and resubmit
                ids = ready.get("env_index").reshape(-1).tolist()
                from ..envs.utils import step_mdp

                nxt = td_stack(
                    [self.pool._carriers[i].clone(False) for i in ids], 0
                )
                if self.policy is not None:
                    nxt = self.policy(nxt)
                else:
                    nxt = _rand_actions(nxt, ids)
                self.pool.async_step_send(nxt, env_ids=ids)
                if collected_frames_in_batch >= self.frames_per_batch:
                    from ..tensordict import cat as td_cat

                    yield td_cat(collected, 0)
                    collected = []
                    collected_frames_in_batch = 0
            if collected:
                from ..tensordict import cat as td_cat

                yield td_cat(collected, 0)

    def shutdown(self, timeout=None):
        if not self.closed:
            self.pool.close()
            self.closed = True
