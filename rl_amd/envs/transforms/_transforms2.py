"""Transforms batch 2: action-space surgery, reward2go, burn-in, RND,
conditional flow, tokenization.

Reference: pytorch/rl torchrl/envs/transforms/ (_action.py
ActionDiscretizer:300, FlattenAction:1525, MultiAction:662; _env.py
BurnInTransform:1649, BatchSizeTransform:1805, RandomTruncation:1255;
_reward.py Reward2Go:637; rnd.py:80; _misc.py ConditionalSkip:658,
RandomCrop:277; _tensor.py Hash:475, Tokenizer:688).
"""
from __future__ import annotations

from typing import Callable, List, Optional, Sequence, Union

import torch

from ...data.tensor_specs import Bounded, Categorical, Composite, TensorSpec, Unbounded
from ...tensordict import TensorDict, TensorDictBase, unravel_key
from ._base import Transform

__all__ = [
    "ActionDiscretizer",
    "FlattenAction",
    "MultiAction",
    "Reward2GoTransform",
    "BurnInTransform",
    "BatchSizeTransform",
    "RandomTruncation",
    "RNDTransform",
    "ConditionalSkip",
    "RandomCrop",
    "HashTransform",
    "TokenizerTransform",
    "ConditionalPolicySwitch",
    "AutoResetTransform",
    "VecGymEnvTransform",
]


class ActionDiscretizer(Transform):
    """Continuous Bounded action space → per-dim categorical grid
    (reference _action.py:300).  The policy emits integer bins; the
    inverse maps them to the continuous midpoints."""

    def __init__(self, num_intervals: Union[int, Sequence[int]] = 10, action_key: str = "action", out_action_key: Optional[str] = None):
        super().__init__(in_keys_inv=[action_key], out_keys_inv=[action_key])
        self.num_intervals = num_intervals
        self.action_key = action_key
        self._low = None
        self._high = None

    def transform_action_spec(self, spec: Composite) -> Composite:
        base = spec[self.action_key]
        if not isinstance(base, Bounded):
            raise TypeError("ActionDiscretizer needs a Bounded action spec")
        self._low = base.low.clone()
        self._high = base.high.clone()
        n_dims = base.shape[-1]
        nvec = (
            [self.num_intervals] * n_dims
            if isinstance(self.num_intervals, int)
            else list(self.num_intervals)
        )
        from ...data.tensor_specs import MultiCategorical

        spec[self.action_key] = MultiCategorical(
            nvec, shape=(*base.shape[:-1], n_dims), device=base.device
        )
        self._nvec = torch.as_tensor(nvec)
        return spec

    def _inv_apply_transform(self, action: torch.Tensor) -> torch.Tensor:
        if self._low is None:
            raise RuntimeError("attach to an env first (spec not transformed)")
        nvec = self._nvec.to(action.device)
        low = self._low.to(action.device)
        high = self._high.to(action.device)
        frac = (action.to(torch.float32) + 0.5) / nvec
        return low + frac * (high - low)


class FlattenAction(Transform):
    """Flatten trailing action dims (reference _action.py:1525)."""

    def __init__(self, action_key: str = "action", first_dim: int = -2, last_dim: int = -1):
        super().__init__(in_keys_inv=[action_key], out_keys_inv=[action_key])
        self.first_dim = first_dim
        self.last_dim = last_dim
        self._shape = None

    def transform_action_spec(self, spec):
        base = spec["action"]
        self._shape = base.shape
        import numpy as np

        flat = int(np.prod(base.shape[self.first_dim:]))
        new_shape = (*base.shape[: self.first_dim], flat)
        spec["action"] = base.expand(*new_shape) if len(new_shape) else base
        return spec

    def _inv_apply_transform(self, action):
        if self._shape is None:
            return action
        return action.reshape(*action.shape[:-1], *self._shape[self.first_dim:])


class MultiAction(Transform):
    """Execute a sequence of actions per env step (macro-actions,
    reference _action.py:662): action shaped [..., S, A] steps the base
    env S times, accumulating reward."""

    def __init__(self, dim: int = 1):
        super().__init__()
        self.dim = dim

    def _inv_call(self, td):
        return td

    def _step(self, td, next_td):
        parent = self.parent
        action = td.get("action")
        S = action.shape[-2]
        base = parent.base_env
        reward = next_td.get("reward")
        from ..utils import step_mdp

        cur = step_mdp(td, next_tensordict=next_td)
        for s in range(1, S):
            if bool(next_td.get("done").any()):
                break
            cur.set("action", action[..., s, :])
            cur = base.step(cur)
            nxt = cur.get("next")
            reward = reward + nxt.get("reward")
            next_td = nxt
            cur = step_mdp(cur)
        next_td.set("reward", reward)
        return next_td

    def transform_action_spec(self, spec):
        base = spec["action"]
        new_shape = (*base.shape[:-1], self.dim, base.shape[-1])
        spec["action"] = base.expand(*new_shape)
        return spec


class Reward2GoTransform(Transform):
    """Discounted reward-to-go — replay-buffer side transform
    (reference _reward.py:637): apply on sampled [*, T] batches via the
    INVERSE call (it is registered as an RB transform)."""

    invertible = True

    def __init__(self, gamma: float = 1.0, in_keys=(("next", "reward"),), out_keys=("reward_to_go",)):
        super().__init__(
            in_keys_inv=list(in_keys), out_keys_inv=list(out_keys)
        )
        self.gamma = gamma

    def _inv_call(self, td: TensorDictBase) -> TensorDictBase:
        from ...objectives.value.functional import reward2go

        for in_key, out_key in zip(self.in_keys_inv, self.out_keys_inv):
            reward = td.get(in_key, None)
            if reward is None:
                continue
            done = td.get(("next", "done"), None)
            if done is None:
                done = torch.zeros_like(reward, dtype=torch.bool)
            td.set(out_key, reward2go(reward, done, self.gamma))
        return td

    def forward(self, td):
        return self._inv_call(td)


class BurnInTransform(Transform):
    """Recurrent warm-up: run the first ``burn_in`` steps of a sampled
    sequence through the RNN modules without grad, then hand the suffix to
    the loss (reference _env.py:1649)."""

    def __init__(self, modules: Sequence, burn_in: int, out_keys: Optional[Sequence] = None):
        super().__init__()
        self.rnn_modules = list(modules)
        self.burn_in = burn_in

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        if self.burn_in == 0 or td.batch_dims < 2:
            return td
        burn = td[:, : self.burn_in]
        rest = td[:, self.burn_in :]
        from ...modules.tensordict_module.rnn import set_recurrent_mode

        with torch.no_grad(), set_recurrent_mode(True):
            for mod in self.rnn_modules:
                burn = mod(burn)
        # carry the final burn-in state into the suffix: the RNN modules'
        # sequence mode reads h0 from recurrent_state[:, 0]
        B, T_rest = rest.batch_size[0], rest.batch_size[1]
        for mod in self.rnn_modules:
            for key in getattr(mod, "state_keys", []):
                nk = ("next", key) if isinstance(key, str) else ("next", *key)
                state = burn.get(nk, None)
                if state is not None:
                    final = state[:, -1]
                    rest.set(
                        key,
                        final.unsqueeze(1).expand(B, T_rest, *final.shape[1:]).clone(),
                    )
        return rest

    def _call(self, td):
        return self.forward(td)


class BatchSizeTransform(Transform):
    """Reshape the env's batch (reference _env.py:1805)."""

    def __init__(self, batch_size: Optional[Sequence[int]] = None, reshape_fn: Optional[Callable] = None):
        super().__init__()
        self.batch_size = torch.Size(batch_size) if batch_size is not None else None
        self.reshape_fn = reshape_fn

    def _call(self, td):
        if self.reshape_fn is not None:
            return self.reshape_fn(td)
        return td.reshape(self.batch_size)

    def _inv_call(self, td):
        return td


class RandomTruncation(Transform):
    """Randomly truncate episodes with prob ``p`` per step
    (reference _env.py:1255)."""

    def __init__(self, p: float = 0.01):
        super().__init__()
        self.p = p

    def _step(self, td, next_td):
        done = next_td.get("done")
        rand_trunc = torch.rand_like(done, dtype=torch.float32) < self.p
        trunc = next_td.get("truncated", torch.zeros_like(done)) | rand_trunc
        next_td.set("truncated", trunc)
        next_td.set("done", done | trunc)
        return next_td


class RNDTransform(Transform):
    """Random-network-distillation intrinsic reward
    (reference rnd.py:80): reward += η·‖f_pred(s') − f_target(s')‖²."""

    def __init__(
        self,
        predictor: torch.nn.Module,
        target: torch.nn.Module,
        *,
        obs_key: str = "observation",
        reward_key: str = "reward",
        intrinsic_weight: float = 1.0,
        out_key: Optional[str] = "intrinsic_reward",
    ):
        super().__init__()
        self.predictor = predictor
        self.target = target
        for p in self.target.parameters():
            p.requires_grad_(False)
        self.obs_key = obs_key
        self.reward_key = reward_key
        self.intrinsic_weight = intrinsic_weight
        self.out_key = out_key

    def _step(self, td, next_td):
        obs = next_td.get(self.obs_key)
        with torch.no_grad():
            err = (self.predictor(obs) - self.target(obs)).pow(2).mean(-1, keepdim=True)
        if self.out_key:
            next_td.set(self.out_key, err)
        r = next_td.get(self.reward_key)
        next_td.set(self.reward_key, r + self.intrinsic_weight * err)
        return next_td

    def transform_observation_spec(self, spec):
        if self.out_key:
            spec[self.out_key] = Unbounded(shape=(*spec.shape, 1), device=spec.device)
        return spec


class ConditionalSkip(Transform):
    """Skip the base env step when a condition holds — the previous
    observation is carried (reference _misc.py:658)."""

    def __init__(self, cond: Callable[[TensorDictBase], torch.Tensor]):
        super().__init__()
        self.cond = cond

    def _step(self, td, next_td):
        skip = self.cond(td)
        if skip is None:
            return next_td
        m = skip
        for key in next_td.keys(True, True):
            prev = td.get(key, None)
            if prev is None or not isinstance(prev, torch.Tensor):
                continue
            cur = next_td.get(key)
            mm = m
            while mm.dim() < cur.dim():
                mm = mm.unsqueeze(-1)
            next_td.set(key, torch.where(mm.expand_as(cur), prev, cur))
        return next_td


class RandomCrop(Transform):
    """Random spatial crop of image observations (reference _misc.py:277)."""

    def __init__(self, w: int, h: Optional[int] = None, in_keys=("pixels",), out_keys=None):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.w = w
        self.h = h if h is not None else w

    def _apply_transform(self, obs):
        H, W = obs.shape[-2:]
        top = int(torch.randint(0, H - self.w + 1, (1,)).item())
        left = int(torch.randint(0, W - self.h + 1, (1,)).item())
        return obs[..., top : top + self.w, left : left + self.h]


class HashTransform(Transform):
    """Hash selected keys to int64 (reference _tensor.py:475)."""

    def __init__(self, in_keys, out_keys, hash_module: Optional[Callable] = None):
        super().__init__(in_keys=list(in_keys), out_keys=list(out_keys))
        from ...data.map import SipHash

        self.hash_module = hash_module or SipHash()

    def _apply_transform(self, x):
        flat = x.reshape(-1, x.shape[-1]) if x.dim() > 1 else x.reshape(1, -1)
        h = self.hash_module(flat)
        return h.reshape(x.shape[:-1]) if x.dim() > 1 else h.squeeze(0)


class TokenizerTransform(Transform):
    """Tokenize text entries with an HF-style tokenizer
    (reference _tensor.py:688)."""

    def __init__(self, tokenizer, in_keys=("text",), out_keys=("tokens",), max_length: Optional[int] = None):
        super().__init__(in_keys=list(in_keys), out_keys=list(out_keys))
        self.tokenizer = tokenizer
        self.max_length = max_length

    def _call(self, td):
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            text = td.get_non_tensor(in_key, None)
            if text is None:
                continue
            if isinstance(text, str):
                text = [text]
            enc = self.tokenizer(text, return_tensors="pt", padding=True)
            ids = enc["input_ids"]
            if self.max_length:
                ids = ids[..., : self.max_length]
            td.set(out_key, ids.squeeze(0) if not td.batch_size else ids)
        return td


class ConditionalPolicySwitch(Transform):
    """Route steps through an alternative policy when a condition holds
    (reference _misc.py:773)."""

    def __init__(self, policy: Callable, condition: Callable[[TensorDictBase], torch.Tensor]):
        super().__init__()
        self.policy = policy
        self.condition = condition

    def _inv_call(self, td):
        cond = self.condition(td)
        if cond is None or not bool(torch.as_tensor(cond).any()):
            return td
        alt = self.policy(td.clone(False))
        action = td.get("action")
        alt_action = alt.get("action")
        m = torch.as_tensor(cond)
        while m.dim() < action.dim():
            m = m.unsqueeze(-1)
        td.set("action", torch.where(m.expand_as(action), alt_action, action))
        return td


class AutoResetTransform(Transform):
    """Adapter for auto-resetting envs (reference _env.py:2011 +
    _misc.py:383 VecGymEnvTransform): libs that reset themselves on done
    return the POST-reset observation in the done step's `next`; this
    transform stashes it and re-emits it as the reset observation so the
    rl_amd step/reset contract holds."""

    def __init__(self, replace: str = "stash", obs_keys: Sequence = ("observation",)):
        super().__init__()
        self.obs_keys = [unravel_key(k) for k in obs_keys]
        self._stash: dict = {}

    def _step(self, td, next_td):
        done = next_td.get("done", None)
        if done is not None and bool(done.any()):
            for k in self.obs_keys:
                val = next_td.get(k, None)
                if val is not None:
                    self._stash[k] = val.clone()
        return next_td

    def _reset(self, td, td_reset):
        if self._stash:
            for k, v in self._stash.items():
                td_reset.set(k, v)
            self._stash = {}
        return td_reset


VecGymEnvTransform = AutoResetTransform
