"""EnvBase — the environment contract.

MI355X-native re-design of the reference env layer
(pytorch/rl torchrl/envs/common.py:404 ``EnvBase``, :2340 ``step``,
:3108 ``reset``, :3449 ``rollout``, :4090 ``step_and_maybe_reset``).
Environments are nn.Modules whose ``_step``/``_reset`` speak TensorDict;
batched (vectorized) envs are the primary path — a GPU-resident env keeps
the whole rollout loop on-device with no host round-trips.
"""
from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional, Sequence, Union

import torch
from torch import nn

from ..data.tensor_specs import (
    Binary,
    Bounded,
    Categorical,
    Composite,
    TensorSpec,
    Unbounded,
)
from ..tensordict import TensorDict, TensorDictBase, stack
from .utils import step_mdp, terminated_or_truncated

__all__ = ["EnvBase", "EnvMetaData", "make_tensordict"]


class EnvMetaData:
    """Picklable env description for worker spec transfer
    (reference torchrl/envs/common.py:124)."""

    def __init__(self, specs: Composite, batch_size: torch.Size, device, env_str: str = ""):
        self.specs = specs
        self.batch_size = batch_size
        self.device = device
        self.env_str = env_str

    @classmethod
    def build(cls, env: "EnvBase") -> "EnvMetaData":
        specs = Composite(shape=env.batch_size, device=env.device)
        specs["full_observation_spec"] = env.full_observation_spec.clone()
        specs["full_action_spec"] = env.full_action_spec.clone()
        specs["full_reward_spec"] = env.full_reward_spec.clone()
        specs["full_done_spec"] = env.full_done_spec.clone()
        return cls(specs, env.batch_size, env.device, str(type(env).__name__))


class EnvBase(nn.Module):
    """Abstract environment.

    Subclasses implement ``_step(td) -> td_next``, ``_reset(td) -> td`` and
    ``_set_seed(seed)``, and declare specs in ``__init__`` via the
    ``observation_spec`` / ``action_spec`` / ``reward_spec`` / ``done_spec``
    setters (leaf or Composite).
    """

    def __init__(
        self,
        device: Union[str, torch.device, None] = None,
        batch_size: Optional[torch.Size] = None,
        run_type_checks: bool = False,
        allow_done_after_reset: bool = False,
    ):
        super().__init__()
        self.__dict__["_device"] = torch.device(device) if device is not None else torch.device("cpu")
        self.__dict__["_batch_size"] = torch.Size(batch_size if batch_size is not None else [])
        self._run_type_checks = run_type_checks
        self.allow_done_after_reset = allow_done_after_reset
        self._full_observation_spec: Optional[Composite] = None
        self._full_action_spec: Optional[Composite] = None
        self._full_reward_spec: Optional[Composite] = None
        self._full_done_spec: Optional[Composite] = None
        self._full_state_spec: Optional[Composite] = None
        self.is_closed = False
        self._seed: Optional[int] = None

    # ------------------------------------------------------------------ #
    # Device / batch-size plumbing
    # ------------------------------------------------------------------ #
    @property
    def device(self) -> torch.device:
        return self.__dict__["_device"]

    @device.setter
    def device(self, value):
        self.__dict__["_device"] = torch.device(value)

    @property
    def batch_size(self) -> torch.Size:
        return self.__dict__["_batch_size"]

    @batch_size.setter
    def batch_size(self, value):
        self.__dict__["_batch_size"] = torch.Size(value)

    @property
    def batch_dims(self) -> int:
        return len(self.batch_size)

    def _empty_composite(self) -> Composite:
        return Composite(shape=self.batch_size, device=self.device)

    @staticmethod
    def _to_composite(value, batch_size, device, default_key: str) -> Composite:
        if isinstance(value, Composite):
            return value
        comp = Composite(shape=batch_size, device=device)
        comp[default_key] = value
        return comp

    # -- observation ---------------------------------------------------- #
    @property
    def full_observation_spec(self) -> Composite:
        if self._full_observation_spec is None:
            self._full_observation_spec = self._empty_composite()
        return self._full_observation_spec

    @full_observation_spec.setter
    def full_observation_spec(self, value):
        self._full_observation_spec = value

    @property
    def observation_spec(self) -> Composite:
        return self.full_observation_spec

    @observation_spec.setter
    def observation_spec(self, value):
        self._full_observation_spec = self._to_composite(
            value, self.batch_size, self.device, "observation"
        )

    # -- action ---------------------------------------------------------- #
    @property
    def full_action_spec(self) -> Composite:
        if self._full_action_spec is None:
            self._full_action_spec = self._empty_composite()
        return self._full_action_spec

    @full_action_spec.setter
    def full_action_spec(self, value):
        self._full_action_spec = self._to_composite(
            value, self.batch_size, self.device, "action"
        )

    @property
    def action_spec(self) -> TensorSpec:
        spec = self.full_action_spec
        keys = spec.keys(True, True)
        if len(keys) == 1:
            return spec[keys[0]]
        return spec

    @action_spec.setter
    def action_spec(self, value):
        self.full_action_spec = value

    @property
    def action_keys(self) -> List:
        return self.full_action_spec.keys(True, True) or ["action"]

    @property
    def action_key(self):
        keys = self.action_keys
        if len(keys) != 1:
            raise RuntimeError("multiple action keys; use action_keys")
        return keys[0]

    # -- reward ----------------------------------------------------------- #
    @property
    def full_reward_spec(self) -> Composite:
        if self._full_reward_spec is None:
            comp = self._empty_composite()
            comp["reward"] = Unbounded(
                shape=(*self.batch_size, 1), device=self.device
            )
            self._full_reward_spec = comp
        return self._full_reward_spec

    @full_reward_spec.setter
    def full_reward_spec(self, value):
        self._full_reward_spec = self._to_composite(
            value, self.batch_size, self.device, "reward"
        )

    @property
    def reward_spec(self) -> TensorSpec:
        spec = self.full_reward_spec
        keys = spec.keys(True, True)
        if len(keys) == 1:
            return spec[keys[0]]
        return spec

    @reward_spec.setter
    def reward_spec(self, value):
        self.full_reward_spec = value

    @property
    def reward_keys(self) -> List:
        return self.full_reward_spec.keys(True, True) or ["reward"]

    @property
    def reward_key(self):
        keys = self.reward_keys
        if len(keys) != 1:
            raise RuntimeError("multiple reward keys; use reward_keys")
        return keys[0]

    # -- done -------------------------------------------------------------- #
    @property
    def full_done_spec(self) -> Composite:
        if self._full_done_spec is None:
            comp = self._empty_composite()
            comp["done"] = Binary(shape=(*self.batch_size, 1), device=self.device)
            comp["terminated"] = Binary(
                shape=(*self.batch_size, 1), device=self.device
            )
            self._full_done_spec = comp
        return self._full_done_spec

    @full_done_spec.setter
    def full_done_spec(self, value):
        self._full_done_spec = self._to_composite(
            value, self.batch_size, self.device, "done"
        )
        if "terminated" not in self._full_done_spec:
            self._full_done_spec["terminated"] = self._full_done_spec["done"].clone()

    @property
    def done_spec(self) -> TensorSpec:
        return self.full_done_spec["done"]

    @done_spec.setter
    def done_spec(self, value):
        self.full_done_spec = value

    @property
    def done_keys(self) -> List:
        return self.full_done_spec.keys(True, True) or ["done"]

    @property
    def done_key(self):
        return "done"

    # -- state -------------------------------------------------------------- #
    @property
    def full_state_spec(self) -> Composite:
        if self._full_state_spec is None:
            self._full_state_spec = self._empty_composite()
        return self._full_state_spec

    @full_state_spec.setter
    def full_state_spec(self, value):
        self._full_state_spec = value

    state_spec = full_state_spec

    @property
    def input_spec(self) -> Composite:
        comp = self._empty_composite()
        comp["full_action_spec"] = self.full_action_spec
        comp["full_state_spec"] = self.full_state_spec
        return comp

    @property
    def output_spec(self) -> Composite:
        comp = self._empty_composite()
        comp["full_observation_spec"] = self.full_observation_spec
        comp["full_reward_spec"] = self.full_reward_spec
        comp["full_done_spec"] = self.full_done_spec
        return comp

    @property
    def specs(self) -> Composite:
        comp = self._empty_composite()
        comp["input_spec"] = self.input_spec
        comp["output_spec"] = self.output_spec
        return comp

    # ------------------------------------------------------------------ #
    # Abstract API
    # ------------------------------------------------------------------ #
    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        raise NotImplementedError

    def _reset(self, tensordict: Optional[TensorDictBase] = None, **kwargs) -> TensorDictBase:
        raise NotImplementedError

    def _set_seed(self, seed: Optional[int]) -> Optional[int]:
        return seed

    # ------------------------------------------------------------------ #
    # Public API
    # ------------------------------------------------------------------ #
    def set_seed(self, seed: Optional[int], static_seed: bool = False) -> Optional[int]:
        self._seed = seed
        out = self._set_seed(seed)
        if out is None or static_seed:
            return seed
        return out

    def step(self, tensordict: TensorDictBase) -> TensorDictBase:
        """Run one transition.  Writes the result under ``tensordict["next"]``
        and returns the same root (reference torchrl/envs/common.py:2340).

        Pre-existing ``"next"`` entries (e.g. recurrent state written by the
        policy) are preserved and merged."""
        next_td = self._step(tensordict)
        self._complete_done(next_td)
        existing = tensordict.get("next", None)
        if existing is not None and not existing.is_empty():
            existing.update(next_td)
        else:
            tensordict.set("next", next_td)
        return tensordict

    def _complete_done(self, td: TensorDictBase) -> None:
        """Ensure done/terminated(/truncated) leaves all exist."""
        done = td.get("done", None)
        term = td.get("terminated", None)
        trunc = td.get("truncated", None)
        if done is None and term is None and trunc is None:
            shape = (*self.batch_size, 1)
            td.set("done", torch.zeros(shape, dtype=torch.bool, device=self.device))
            td.set("terminated", torch.zeros(shape, dtype=torch.bool, device=self.device))
            return
        if term is None:
            term = done if done is not None else torch.zeros_like(trunc)
            td.set("terminated", term)
        if done is None:
            done = term | trunc if trunc is not None else term
            td.set("done", done)
        elif trunc is not None:
            td.set("done", term | trunc | done)

    def reset(
        self,
        tensordict: Optional[TensorDictBase] = None,
        **kwargs,
    ) -> TensorDictBase:
        """Reset the env (optionally partially via a ``"_reset"`` mask in
        ``tensordict``); returns the initial root TensorDict
        (reference torchrl/envs/common.py:3108)."""
        td = self._reset(tensordict, **kwargs)
        self._complete_done(td)
        if tensordict is not None and "_reset" in tensordict:
            td.pop("_reset", None)
        return td

    def maybe_reset(self, tensordict: TensorDictBase) -> TensorDictBase:
        """Reset sub-envs flagged by ``"_reset"``; pass-through otherwise."""
        if "_reset" in tensordict and tensordict.get("_reset").any():
            return self.reset(tensordict)
        return tensordict

    # Vectorized envs whose _reset applies a "_reset" mask with pure tensor
    # ops set this True: step_and_maybe_reset then resets UNCONDITIONALLY
    # (masked), avoiding the bool(done.any()) host-device sync per step —
    # the difference between a stall and a fully queued HIP stream.
    _supports_masked_reset: bool = False

    def step_and_maybe_reset(self, tensordict: TensorDictBase):
        """step + partial auto-reset; returns ``(td_with_next, next_root)``
        (reference torchrl/envs/common.py:4090)."""
        tensordict = self.step(tensordict)
        next_root = step_mdp(
            tensordict,
            reward_keys=self.reward_keys,
            done_keys=self.done_keys,
            action_keys=self.action_keys,
        )
        if self._supports_masked_reset:
            done = next_root.get("done", None)
            if done is None:
                return tensordict, next_root
            next_root.set("_reset", done)
            next_root = self.reset(next_root)
            return tensordict, next_root
        any_done = terminated_or_truncated(next_root, key="_reset")
        if any_done:
            next_root = self.reset(next_root)
        return tensordict, next_root

    def rand_action(self, tensordict: Optional[TensorDictBase] = None) -> TensorDictBase:
        if tensordict is None:
            tensordict = TensorDict({}, batch_size=self.batch_size, device=self.device)
        for key in self.full_action_spec.keys(True, True):
            tensordict.set(key, self.full_action_spec[key].rand())
        return tensordict

    def rand_step(self, tensordict: Optional[TensorDictBase] = None) -> TensorDictBase:
        if tensordict is None:
            tensordict = self.reset()
        tensordict = self.rand_action(tensordict)
        return self.step(tensordict)

    def rollout(
        self,
        max_steps: int,
        policy: Optional[Callable[[TensorDictBase], TensorDictBase]] = None,
        callback: Optional[Callable] = None,
        auto_reset: bool = True,
        auto_cast_to_device: bool = False,
        break_when_any_done: bool = True,
        break_when_all_done: bool = False,
        return_contiguous: bool = True,
        tensordict: Optional[TensorDictBase] = None,
        set_truncated: bool = False,
        out=None,
    ) -> TensorDictBase:
        """Roll the env for up to ``max_steps`` under ``policy`` (random if
        None); stacks transitions along a trailing time dim
        (reference torchrl/envs/common.py:3449)."""
        if auto_reset and tensordict is None:
            tensordict = self.reset()
        elif tensordict is None:
            raise RuntimeError("rollout needs a tensordict when auto_reset=False")
        policy_device = None
        if policy is not None and auto_cast_to_device:
            try:
                policy_device = next(policy.parameters()).device
            except (StopIteration, AttributeError):
                policy_device = None
        tds = []
        td = tensordict
        for i in range(max_steps):
            if policy is not None:
                if policy_device is not None:
                    td = td.to(policy_device)
                td = policy(td)
                if policy_device is not None:
                    td = td.to(self.device)
            else:
                td = self.rand_action(td)
            td = self.step(td)
            tds.append(td.clone(False))
            done = td.get(("next", "done"))
            if break_when_any_done and bool(done.any()):
                break
            if break_when_all_done and bool(done.all()):
                break
            if break_when_any_done or break_when_all_done:
                td = step_mdp(
                    td,
                    reward_keys=self.reward_keys,
                    done_keys=self.done_keys,
                    action_keys=self.action_keys,
                )
            else:
                td, _next_root = None, None
                last = tds[-1]
                next_root = step_mdp(
                    last,
                    reward_keys=self.reward_keys,
                    done_keys=self.done_keys,
                    action_keys=self.action_keys,
                )
                if terminated_or_truncated(next_root, key="_reset"):
                    next_root = self.reset(next_root)
                td = next_root
            if callback is not None:
                callback(self, tds[-1])
        if set_truncated and tds:
            last_next = tds[-1].get("next")
            done = last_next.get("done")
            last_next.set("truncated", torch.ones_like(done))
            last_next.set("done", torch.ones_like(done))
        out_td = stack(tds, len(self.batch_size))
        if return_contiguous:
            out_td = out_td.contiguous()
        return out_td

    def fake_tensordict(self) -> TensorDictBase:
        """Zero-filled TensorDict matching the env's IO contract."""
        td = TensorDict({}, batch_size=self.batch_size, device=self.device)
        for key in self.full_observation_spec.keys(True, True):
            td.set(key, self.full_observation_spec[key].zero())
        for key in self.full_action_spec.keys(True, True):
            td.set(key, self.full_action_spec[key].zero())
        for key in self.full_done_spec.keys(True, True):
            td.set(key, self.full_done_spec[key].zero())
        nxt = TensorDict({}, batch_size=self.batch_size, device=self.device)
        for key in self.full_observation_spec.keys(True, True):
            nxt.set(key, self.full_observation_spec[key].zero())
        for key in self.full_reward_spec.keys(True, True):
            nxt.set(key, self.full_reward_spec[key].zero())
        for key in self.full_done_spec.keys(True, True):
            nxt.set(key, self.full_done_spec[key].zero())
        td.set("next", nxt)
        return td

    def close(self, raise_if_closed: bool = False) -> None:
        self.is_closed = True

    def empty_cache(self):
        pass

    def forward(self, tensordict: TensorDictBase) -> TensorDictBase:
        raise RuntimeError("EnvBase.forward is not the API; use step()/reset()")

    def to(self, device, *args, **kwargs):
        device = torch.device(device)
        out = super().to(device)
        out.__dict__["_device"] = device
        for spec_attr in (
            "_full_observation_spec",
            "_full_action_spec",
            "_full_reward_spec",
            "_full_done_spec",
            "_full_state_spec",
        ):
            spec = getattr(self, spec_attr)
            if spec is not None:
                setattr(self, spec_attr, spec.to(device))
        return out

    def __repr__(self):
        return (
            f"{type(self).__name__}(batch_size={tuple(self.batch_size)}, "
            f"device={self.device})"
        )


def make_tensordict(env: EnvBase) -> TensorDictBase:
    return env.fake_tensordict()
