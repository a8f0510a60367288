"""Model-based envs: dream rollouts through a learned world model.

Reference: pytorch/rl torchrl/envs/model_based/ (ModelBasedEnvBase
common.py, DreamerEnv dreamer.py).
"""
from __future__ import annotations

from typing import Optional

import torch

from ..tensordict import TensorDict, TensorDictBase, TensorDictModuleBase
from .common import EnvBase

__all__ = ["ModelBasedEnvBase", "DreamerEnv"]


class ModelBasedEnvBase(EnvBase):
    """Env whose ``_step`` runs a world-model TensorDictModule: the model
    maps (state, action) → (next state, reward) (reference common.py).

    The world model runs entirely on-device — imagination rollouts are
    pure GPU compute at whatever batch size the planner asks for.
    """

    def __init__(
        self,
        world_model: TensorDictModuleBase,
        device=None,
        batch_size=None,
        params=None,
    ):
        super().__init__(device=device, batch_size=batch_size)
        self.world_model = world_model

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        td = tensordict.clone(False)
        td = self.world_model(td)
        # batch-polymorphic: planners expand the batch to N candidates
        bs = tensordict.batch_size
        out = TensorDict({}, batch_size=bs, device=self.device)
        for key in self.full_observation_spec.keys(True, True):
            if key in td:
                out.set(key, td.get(key))
        reward = td.get("reward", None)
        if reward is None:
            reward = torch.zeros((*bs, 1), device=self.device)
        out.set("reward", reward)
        done = td.get("done", None)
        if done is None:
            done = torch.zeros((*bs, 1), dtype=torch.bool, device=self.device)
        out.set("done", done)
        out.set("terminated", td.get("terminated", done))
        return out

    def _reset(self, tensordict: Optional[TensorDictBase] = None, **kwargs) -> TensorDictBase:
        td = self.full_observation_spec.zero()
        td.update(self.full_done_spec.zero())
        return td

    def _set_seed(self, seed):
        return seed


class DreamerEnv(ModelBasedEnvBase):
    """Dream env over an RSSM world model (reference dreamer.py): the
    observation is the latent (deter, stoch) pair; decoding to pixels is a
    transform concern."""

    def __init__(self, world_model, prior_shape=None, belief_shape=None, device=None, batch_size=None):
        super().__init__(world_model, device=device, batch_size=batch_size)
        self.prior_shape = prior_shape
        self.belief_shape = belief_shape
