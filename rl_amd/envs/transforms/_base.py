"""Transform framework: ``Transform``, ``Compose``, ``TransformedEnv``.

Reference: pytorch/rl torchrl/envs/transforms/_base.py:178 (Transform),
:944 (TransformedEnv), :1650 (Compose).  Forward transforms rewrite
observations/rewards on the way out of the env; inverse transforms rewrite
actions/inputs on the way in; spec transforms keep the contract honest.
"""
from __future__ import annotations

import copy
from typing import Any, Callable, List, Optional, Sequence, Union

import torch
from torch import nn

from ...data.tensor_specs import Composite, TensorSpec
from ...tensordict import TensorDict, TensorDictBase, unravel_key
from ..common import EnvBase

__all__ = ["Transform", "Compose", "TransformedEnv", "AutoResetEnv"]


class Transform(nn.Module):
    """Base transform.

    Concrete transforms override ``_apply_transform`` (per-leaf, forward),
    ``_inv_apply_transform`` (per-leaf, inverse), or the coarser
    ``_call``/``_inv_call`` (whole TensorDict), plus the spec hooks.
    """

    invertible: bool = False

    def __init__(
        self,
        in_keys: Optional[Sequence] = None,
        out_keys: Optional[Sequence] = None,
        in_keys_inv: Optional[Sequence] = None,
        out_keys_inv: Optional[Sequence] = None,
    ):
        super().__init__()
        self.in_keys = [unravel_key(k) for k in in_keys] if in_keys else []
        self.out_keys = (
            [unravel_key(k) for k in out_keys] if out_keys else list(self.in_keys)
        )
        self.in_keys_inv = (
            [unravel_key(k) for k in in_keys_inv] if in_keys_inv else []
        )
        self.out_keys_inv = (
            [unravel_key(k) for k in out_keys_inv] if out_keys_inv else list(self.in_keys_inv)
        )
        self._parent: Optional[EnvBase] = None

    # -- parent plumbing -------------------------------------------------- #
    @property
    def parent(self) -> Optional[EnvBase]:
        return self._parent

    def set_container(self, container) -> "Transform":
        self._parent = container
        return self

    def clone(self) -> "Transform":
        parent = self._parent
        self._parent = None
        out = copy.deepcopy(self)
        self._parent = parent
        return out

    # -- forward path ------------------------------------------------------ #
    def _apply_transform(self, obs: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def _call(self, td: TensorDictBase) -> TensorDictBase:
        """Rewrite env outputs in-place (called on reset output and on the
        ``next`` td after each step)."""
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            val = td.get(in_key, None)
            if val is not None:
                td.set(out_key, self._apply_transform(val))
        return td

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        """Apply the forward pass outside an env (dataset / replay use)."""
        return self._call(td)

    def _reset(
        self, tensordict: Optional[TensorDictBase], tensordict_reset: TensorDictBase
    ) -> TensorDictBase:
        """Hook on reset; default applies ``_call`` to the reset output."""
        return self._call(tensordict_reset)

    def _step(
        self, tensordict: TensorDictBase, next_tensordict: TensorDictBase
    ) -> TensorDictBase:
        """Hook on step; default applies ``_call`` to the ``next`` td."""
        return self._call(next_tensordict)

    # -- inverse path ------------------------------------------------------ #
    def _inv_apply_transform(self, val: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def _inv_call(self, td: TensorDictBase) -> TensorDictBase:
        for in_key, out_key in zip(self.in_keys_inv, self.out_keys_inv):
            val = td.get(in_key, None)
            if val is not None:
                td.set(out_key, self._inv_apply_transform(val))
        return td

    def inv(self, td: TensorDictBase) -> TensorDictBase:
        return self._inv_call(td)

    # -- spec hooks --------------------------------------------------------- #
    def transform_observation_spec(self, observation_spec: Composite) -> Composite:
        return observation_spec

    def transform_action_spec(self, action_spec: Composite) -> Composite:
        return action_spec

    def transform_input_spec(self, input_spec: Composite) -> Composite:
        return input_spec

    def transform_reward_spec(self, reward_spec: Composite) -> Composite:
        return reward_spec

    def transform_done_spec(self, done_spec: Composite) -> Composite:
        return done_spec

    def transform_output_spec(self, output_spec: Composite) -> Composite:
        return output_spec

    def transform_state_spec(self, state_spec: Composite) -> Composite:
        return state_spec

    def init_transform(self, env: EnvBase) -> None:
        """Called when attached to a TransformedEnv."""

    def __repr__(self):
        return f"{type(self).__name__}(in_keys={self.in_keys}, out_keys={self.out_keys})"


class Compose(Transform):
    """Chain of transforms (reference _base.py:1650)."""

    def __init__(self, *transforms: Transform):
        super().__init__()
        if len(transforms) == 1 and isinstance(transforms[0], (list, tuple)):
            transforms = tuple(transforms[0])
        self.transforms = nn.ModuleList(transforms)

    def set_container(self, container):
        super().set_container(container)
        for t in self.transforms:
            t.set_container(container)
        return self

    def append(self, t: Transform) -> "Compose":
        self.transforms.append(t)
        t.set_container(self._parent)
        return self

    def insert(self, index: int, t: Transform) -> "Compose":
        self.transforms.insert(index, t)
        t.set_container(self._parent)
        return self

    def __iter__(self):
        return iter(self.transforms)

    def __len__(self):
        return len(self.transforms)

    def __getitem__(self, idx):
        if isinstance(idx, slice):
            return Compose(*list(self.transforms)[idx])
        return self.transforms[idx]

    def _call(self, td):
        for t in self.transforms:
            td = t._call(td)
        return td

    def forward(self, td):
        for t in self.transforms:
            td = t.forward(td)
        return td

    def _step(self, td, next_td):
        for t in self.transforms:
            next_td = t._step(td, next_td)
        return next_td

    def _reset(self, td, td_reset):
        for t in self.transforms:
            td_reset = t._reset(td, td_reset)
        return td_reset

    def _inv_call(self, td):
        for t in reversed(self.transforms):
            td = t._inv_call(td)
        return td

    def transform_observation_spec(self, spec):
        for t in self.transforms:
            spec = t.transform_observation_spec(spec)
        return spec

    def transform_action_spec(self, spec):
        for t in self.transforms:
            spec = t.transform_action_spec(spec)
        return spec

    def transform_input_spec(self, spec):
        for t in self.transforms:
            spec = t.transform_input_spec(spec)
        return spec

    def transform_reward_spec(self, spec):
        for t in self.transforms:
            spec = t.transform_reward_spec(spec)
        return spec

    def transform_done_spec(self, spec):
        for t in self.transforms:
            spec = t.transform_done_spec(spec)
        return spec

    def transform_state_spec(self, spec):
        for t in self.transforms:
            spec = t.transform_state_spec(spec)
        return spec

    def init_transform(self, env):
        for t in self.transforms:
            t.init_transform(env)

    def __repr__(self):
        inner = ", ".join(repr(t) for t in self.transforms)
        return f"Compose({inner})"


class TransformedEnv(EnvBase):
    """Env wrapper applying a transform stack (reference _base.py:944)."""

    def __init__(
        self,
        env: EnvBase,
        transform: Optional[Transform] = None,
        cache_specs: bool = True,
        auto_unwrap: bool = True,
        device=None,
    ):
        if auto_unwrap and isinstance(env, TransformedEnv):
            inner_t = env.transform
            env = env.base_env
            if transform is None:
                transform = inner_t
            else:
                transform = Compose(inner_t, transform)
        super().__init__(
            device=device if device is not None else env.device,
            batch_size=env.batch_size,
        )
        self.base_env = env
        if transform is None:
            transform = Compose()
        elif not isinstance(transform, Compose):
            transform = Compose(transform)
        self.transform = transform
        self.transform.set_container(self)
        self.transform.init_transform(env)
        self._cache_specs = cache_specs
        self._spec_cache: dict = {}

    # -- spec views (transformed) ------------------------------------------- #
    @property
    def full_observation_spec(self) -> Composite:
        key = "obs"
        if self._cache_specs and key in self._spec_cache:
            return self._spec_cache[key]
        spec = self.transform.transform_observation_spec(
            self.base_env.full_observation_spec.clone()
        )
        if self._cache_specs:
            self._spec_cache[key] = spec
        return spec

    @property
    def full_action_spec(self) -> Composite:
        key = "act"
        if self._cache_specs and key in self._spec_cache:
            return self._spec_cache[key]
        spec = self.transform.transform_action_spec(
            self.base_env.full_action_spec.clone()
        )
        if self._cache_specs:
            self._spec_cache[key] = spec
        return spec

    @property
    def full_reward_spec(self) -> Composite:
        key = "rew"
        if self._cache_specs and key in self._spec_cache:
            return self._spec_cache[key]
        spec = self.transform.transform_reward_spec(
            self.base_env.full_reward_spec.clone()
        )
        if self._cache_specs:
            self._spec_cache[key] = spec
        return spec

    @property
    def full_done_spec(self) -> Composite:
        return self.transform.transform_done_spec(
            self.base_env.full_done_spec.clone()
        )

    @property
    def full_state_spec(self) -> Composite:
        return self.transform.transform_state_spec(
            self.base_env.full_state_spec.clone()
        )

    def empty_cache(self):
        self._spec_cache = {}
        self.base_env.empty_cache()

    # -- step/reset --------------------------------------------------------- #
    def _step(self, tensordict):
        raise RuntimeError("TransformedEnv overrides step directly")

    def _reset(self, tensordict=None, **kwargs):
        raise RuntimeError("TransformedEnv overrides reset directly")

    def step(self, tensordict: TensorDictBase) -> TensorDictBase:
        tensordict = self.transform._inv_call(tensordict)
        tensordict = self.base_env.step(tensordict)
        next_td = tensordict.get("next")
        next_td = self.transform._step(tensordict, next_td)
        tensordict.set("next", next_td)
        return tensordict

    def reset(self, tensordict: Optional[TensorDictBase] = None, **kwargs):
        td_reset = self.base_env.reset(tensordict, **kwargs)
        td_reset = self.transform._reset(tensordict, td_reset)
        return td_reset

    def _set_seed(self, seed):
        return self.base_env.set_seed(seed)

    def set_seed(self, seed, static_seed: bool = False):
        return self.base_env.set_seed(seed, static_seed=static_seed)

    def append_transform(self, t: Transform) -> "TransformedEnv":
        self.transform.append(t)
        self.empty_cache()
        return self

    def insert_transform(self, index: int, t: Transform) -> "TransformedEnv":
        self.transform.insert(index, t)
        self.empty_cache()
        return self

    def close(self, raise_if_closed: bool = False):
        self.base_env.close()
        self.is_closed = True

    def __getattr__(self, name):
        # delegate unknown attributes to the base env (reference behavior)
        try:
            return super().__getattr__(name)
        except AttributeError:
            base = self.__dict__.get("_modules", {}).get("base_env")
            if base is not None:
                return getattr(base, name)
            raise

    def __repr__(self):
        return f"TransformedEnv(env={self.base_env}, transform={self.transform})"


class AutoResetEnv(TransformedEnv):
    """TransformedEnv variant for natively auto-resetting base envs
    (reference _base.py:2094): pairs with
    :class:`~rl_amd.envs.transforms.AutoResetTransform`, which buffers
    the post-done reset observation the lib delivers in-step."""

    def _reset(self, tensordict=None, **kwargs):
        if tensordict is not None:
            tensordict = tensordict.select("_reset", strict=False)
        return super().reset(tensordict, **kwargs)
