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


class WorldModelEnv(ModelBasedEnvBase):
    """Generic env around a :class:`~rl_amd.modules.WorldModel`
    (reference model_based/world_model_env.py:20): the wrapped module
    owns prediction (dynamics + reward/done heads); this env owns the
    rollout contract.  ``obs_spec``/``action_spec`` may be passed
    explicitly or inferred from a prototype tensordict."""

    def __init__(self, world_model, *, observation_spec=None, action_spec=None,
                 device=None, batch_size=None, max_steps: Optional[int] = None):
        super().__init__(world_model, device=device, batch_size=batch_size)
        if observation_spec is not None:
            self.observation_spec = observation_spec
        if action_spec is not None:
            self.action_spec = action_spec
        self.max_steps = max_steps
        self._t = 0

    def _reset(self, tensordict=None, **kwargs):
        self._t = 0
        if tensordict is not None:
            out = tensordict.clone(False)
            out.update(self.full_done_spec.zero())
            return out
        return super()._reset(tensordict, **kwargs)

    def _step(self, tensordict):
        out = super()._step(tensordict)
        self._t += 1
        if self.max_steps is not None and self._t >= self.max_steps:
            done = torch.ones_like(out.get("done"))
            out.set("done", done)
            out.set("terminated", done)
        return out


class ImaginedEnv(ModelBasedEnvBase):
    """Imagination env for uncertainty-aware model-based policy search
    (reference model_based/imagined.py:17): observations carry mean AND
    variance — ``("observation", "mean")`` / ``("observation", "var")``
    — so moment-matching controllers (PILCO-style, see
    rl_amd.modules.models.gp) can propagate uncertainty through
    imagined rollouts."""

    def __init__(self, world_model, *, state_dim: int, device=None,
                 batch_size=None, reward_model=None):
        super().__init__(world_model, device=device, batch_size=batch_size)
        self.state_dim = state_dim
        self.reward_model = reward_model
        bs = self.batch_size
        from ..data.tensor_specs import Composite, Unbounded

        self.observation_spec = Composite(
            {
                "observation": Composite(
                    {
                        "mean": Unbounded(shape=(*bs, state_dim), device=self.device),
                        "var": Unbounded(shape=(*bs, state_dim), device=self.device),
                    },
                    shape=bs,
                    device=self.device,
                )
            },
            shape=bs,
            device=self.device,
        )

    def _step(self, tensordict):
        td = self.world_model(tensordict.clone(False))
        bs = tensordict.batch_size
        mean = td.get(("observation", "mean"))
        var = td.get(("observation", "var"))
        reward = td.get("reward", None)
        if reward is None and self.reward_model is not None:
            reward = self.reward_model(mean, var)
        if reward is None:
            reward = torch.zeros((*bs, 1), device=mean.device)
        return TensorDict(
            {
                "observation": TensorDict({"mean": mean, "var": var}, batch_size=bs),
                "reward": reward.reshape(*bs, -1)[..., :1],
                "done": torch.zeros((*bs, 1), dtype=torch.bool, device=mean.device),
                "terminated": torch.zeros((*bs, 1), dtype=torch.bool, device=mean.device),
            },
            batch_size=bs,
            device=self.device,
        )


class DreamerDecoder:
    """Transform that decodes latent states back to pixels during
    imagined eval rollouts (reference model_based/dreamer.py:97).
    Appended to a :class:`DreamerEnv`; calls the env's decoder on the
    latent keys and writes ``reco_pixels``."""

    def __init__(self, decoder=None, latent_keys=("state", "belief"),
                 out_key: str = "reco_pixels"):
        self.decoder = decoder
        self.latent_keys = tuple(latent_keys)
        self.out_key = out_key
        self.parent: Optional[EnvBase] = None

    def __call__(self, td: TensorDictBase) -> TensorDictBase:
        decoder = self.decoder
        if decoder is None and self.parent is not None:
            decoder = getattr(self.parent, "decoder", None)
        if decoder is None:
            raise RuntimeError("DreamerDecoder needs a decoder (own or parent env's)")
        latents = [td.get(k) for k in self.latent_keys if k in td]
        td.set(self.out_key, decoder(torch.cat(latents, dim=-1)))
        return td


__all__ += ["WorldModelEnv", "ImaginedEnv", "DreamerDecoder"]
