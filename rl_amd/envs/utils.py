"""Env utilities: ``step_mdp``, done aggregation, ``check_env_specs``,
exploration-type control.

Reference behavior: pytorch/rl torchrl/envs/utils.py:327 (``step_mdp``),
:1142-1393 (done aggregation), :686 (``check_env_specs``).
"""
from __future__ import annotations

import contextlib
from typing import List, Optional, Sequence

import torch

from ..tensordict import TensorDict, TensorDictBase, unravel_key
from ..tensordict.nn import InteractionType, set_interaction_type as _set_it

__all__ = [
    "step_mdp",
    "check_env_specs",
    "terminated_or_truncated",
    "ExplorationType",
    "set_exploration_type",
    "exploration_type",
    "make_composite_from_td",
]

# Exploration types alias the tensordict interaction types (reference:
# torchrl/envs/utils.py ExplorationType).
ExplorationType = InteractionType
set_exploration_type = _set_it


def exploration_type():
    from ..tensordict.nn import interaction_type

    return interaction_type()


def step_mdp(
    tensordict: TensorDictBase,
    next_tensordict: Optional[TensorDictBase] = None,
    keep_other: bool = True,
    exclude_reward: bool = True,
    exclude_done: bool = False,
    exclude_action: bool = True,
    reward_keys: Sequence = ("reward",),
    done_keys: Sequence = ("done", "terminated", "truncated"),
    action_keys: Sequence = ("action",),
) -> TensorDictBase:
    """Promote ``"next"`` to the root: the per-step MDP transition t → t+1.

    Returns a new TensorDict whose root holds what was under ``"next"``,
    minus reward (and optionally done), plus any other root keys carried
    over (policy state like RNN hidden, ``collector`` metadata, …).
    Reference: torchrl/envs/utils.py:327 — the per-step hot path; here a
    flat shallow-copy loop, no key-caching machinery needed.
    """
    nxt = tensordict.get("next") if next_tensordict is None else next_tensordict
    out = nxt.clone(False)
    if exclude_reward:
        for k in reward_keys:
            _pop_nested(out, k)
    if exclude_done:
        for k in done_keys:
            _pop_nested(out, k)
    nested_actions = [
        unravel_key(k) for k in action_keys if isinstance(unravel_key(k), tuple)
    ]
    if keep_other:
        skip = {"next"}
        if exclude_action:
            skip |= {k for k in map(unravel_key, action_keys) if isinstance(k, str)}
        for k, v in tensordict._data.items():
            if k in skip or k in out._data:
                continue
            out._data[k] = v
        if exclude_action:
            for k in nested_actions:
                _pop_nested(out, k)
    elif not exclude_action:
        for k in action_keys:
            if k in tensordict and k not in out._data:
                out.set(k, tensordict.get(k))
    return out


def _pop_nested(td: TensorDictBase, key) -> None:
    key = unravel_key(key)
    try:
        td.del_(key)
    except KeyError:
        pass


def terminated_or_truncated(
    data: TensorDictBase,
    full_done_spec=None,
    key: Optional[str] = "_reset",
    write_full_false: bool = False,
) -> bool:
    """Aggregate ``terminated``/``truncated``/``done`` leaves into a single
    ``_reset`` flag per sub-env; returns True if any env is done.
    Reference: torchrl/envs/utils.py:1142 ``_terminated_or_truncated``."""
    done = None
    for k in ("done", "terminated", "truncated"):
        val = data.get(k, None)
        if val is not None:
            done = val if done is None else (done | val)
    if done is None:
        return False
    any_done = bool(done.any())
    if key is not None and (any_done or write_full_false):
        data.set(key, done)
    return any_done


def make_composite_from_td(td: TensorDictBase, *, unsqueeze_null_shapes: bool = True):
    """Infer a Composite spec from an example TensorDict (reference helper
    used by custom envs)."""
    from ..data.tensor_specs import Composite, NonTensor, Unbounded

    comp = Composite(shape=td.batch_size, device=td.device)
    for k, v in td.items():
        if isinstance(v, TensorDictBase):
            comp[k] = make_composite_from_td(v)
        elif isinstance(v, torch.Tensor):
            comp[k] = Unbounded(shape=v.shape, device=v.device, dtype=v.dtype)
        else:
            comp[k] = NonTensor(example_data=getattr(v, "data", v))
    return comp


def check_env_specs(
    env,
    return_contiguous: bool = True,
    check_dtype: bool = True,
    seed: Optional[int] = None,
    break_when_any_done: bool = False,
) -> None:
    """Roll the env and assert real data matches declared specs — the
    universal env integration test (reference torchrl/envs/utils.py:686)."""
    if seed is not None:
        env.set_seed(seed)
    td = env.reset()
    fake = env.observation_spec.zero()
    # reset output must contain observation + done keys
    for key in env.observation_spec.keys(True, True):
        assert key in td, f"reset output missing observation key {key}"
        if check_dtype:
            real = td.get(key)
            spec = env.observation_spec[key]
            if isinstance(real, torch.Tensor):
                assert real.dtype == spec.dtype, (
                    f"dtype mismatch for {key}: {real.dtype} vs {spec.dtype}"
                )
                assert real.shape == torch.Size([*env.batch_size, *spec.shape[env.batch_dims:]]) or real.shape == spec.shape, (
                    f"shape mismatch for {key}: {tuple(real.shape)} vs spec {tuple(spec.shape)}"
                )
    rollout = env.rollout(3, break_when_any_done=break_when_any_done)
    # action / reward / next-observation checks
    assert "action" in rollout, "rollout must record actions"
    nxt = rollout.get("next")
    for key in ("reward", "done"):
        assert key in nxt, f"rollout['next'] missing {key}"
    r = nxt.get("reward")
    spec = env.reward_spec
    if check_dtype:
        assert r.dtype == spec.dtype, f"reward dtype {r.dtype} != {spec.dtype}"
    for key in env.observation_spec.keys(True, True):
        assert key in nxt, f"rollout['next'] missing observation {key}"
    act_spec = env.full_action_spec
    for key in act_spec.keys(True, True):
        assert key in rollout, f"rollout missing action key {key}"
        assert act_spec[key].is_in(rollout.get(key)), (
            f"rollout action {key} out of spec bounds"
        )
    return None


class _classproperty:
    def __init__(self, fget):
        self.fget = fget

    def __get__(self, obj, owner):
        return self.fget(owner)


class RandomPolicy:
    """Draw random actions from an action spec (reference:
    torchrl/modules/tensordict_module/exploration.py:771)."""

    def __init__(self, action_spec, action_key: str = "action"):
        self.action_spec = action_spec
        self.action_key = action_key

    def __call__(self, td: TensorDictBase) -> TensorDictBase:
        from ..data.tensor_specs import Composite

        # spec shape already includes env batch dims
        if isinstance(self.action_spec, Composite):
            td.update(self.action_spec.rand())
        else:
            td.set(self.action_key, self.action_spec.rand())
        return td
