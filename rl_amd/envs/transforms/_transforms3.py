"""Transforms batch 3: termination predicates, action scaling/chunking/
tokenization, multi-objective rewards, replay-buffer-side transforms,
module/timer/belief-state bridges.

Reference: pytorch/rl torchrl/envs/transforms/ (_env.py
TerminateTransform:1174, gSDENoise:667; _action.py ActionScaling:1004,
ActionChunkTransform:1812, ActionTokenizerTransform:2105; _reward.py
LineariseRewards:881, SuccessReward:997; _keys.py FlattenTensorDict:673;
_clip.py ExpandAs:168; _observation.py NextObservationDelta:1521;
rb_transforms.py MultiStepTransform:25, NextStateReconstructor:230,
PolicyAgeFilter:466; module.py ModuleTransform:123; _timer.py Timer;
gym_transforms.py EndOfLifeTransform:20;
mean_action_selector.py MeanActionSelector).
"""
from __future__ import annotations

import time
import warnings
from typing import Any, Callable, List, Optional, Sequence, Union

import torch

from ...data.tensor_specs import Bounded, Categorical, Composite, TensorSpec, Unbounded
from ...tensordict import TensorDict, TensorDictBase, unravel_key
from ._base import Transform
from ._transforms import TensorDictPrimer

__all__ = [
    "TerminateTransform",
    "gSDENoise",
    "ActionScaling",
    "ActionChunkTransform",
    "ActionTokenizerTransform",
    "LineariseRewards",
    "SuccessReward",
    "FlattenTensorDict",
    "ExpandAs",
    "NextObservationDelta",
    "MultiStepTransform",
    "NextStateReconstructor",
    "PolicyAgeFilter",
    "ModuleTransform",
    "Timer",
    "EndOfLifeTransform",
    "MeanActionSelector",
]


# --------------------------------------------------------------------------- #
# env flow
# --------------------------------------------------------------------------- #
class TerminateTransform(Transform):
    """Terminate a rollout when a predicate on the post-step td is true
    (reference _env.py:1174).  The predicate result is OR-ed into
    ``terminated`` (and ``done`` unless ``write_done=False``)."""

    def __init__(self, stop: Callable[[TensorDictBase], Any], *, write_done: bool = True):
        super().__init__()
        self.stop = stop
        self.write_done = write_done

    def _step(self, tensordict, next_tensordict):
        flag = self.stop(next_tensordict)
        flag = torch.as_tensor(flag, dtype=torch.bool)
        for key in ("terminated",) + (("done",) if self.write_done else ()):
            cur = next_tensordict.get(key, None)
            if cur is not None:
                f = flag
                while f.dim() < cur.dim():
                    f = f.unsqueeze(-1)
                next_tensordict.set(key, cur | f.expand_as(cur))
        return next_tensordict

    def _call(self, td):
        return td


class gSDENoise(TensorDictPrimer):
    """Primer for the gSDE exploration noise ``_eps_gSDE`` (reference
    _env.py:667): zero-filled ``[..., 1]`` when dims are unknown, random
    normal ``[..., action_dim, state_dim]`` otherwise."""

    def __init__(self, state_dim: Optional[int] = None, action_dim: Optional[int] = None, shape=None, **kwargs):
        self.state_dim = state_dim
        self.action_dim = action_dim
        shape = tuple(shape) if shape is not None else ()
        tail = (1,) if state_dim is None or action_dim is None else (action_dim, state_dim)
        random = state_dim is not None and action_dim is not None
        super().__init__(
            primers={"_eps_gSDE": Unbounded(shape=shape + tail)},
            random=random,
            **kwargs,
        )


# --------------------------------------------------------------------------- #
# actions
# --------------------------------------------------------------------------- #
class ActionScaling(Transform):
    """Expose a normalized ``[-1, 1]`` (or ``[0, 1]``) action space and
    affine-map policy actions back to the env's bounded range on the
    inverse path (reference _action.py:1004).

    ``loc = (high+low)/2``, ``scale = (high-low)/2`` when
    ``standard_normal=True``; forward direction normalizes dataset/env
    actions for replay use.
    """

    def __init__(
        self,
        in_keys_inv: Optional[Sequence] = None,
        out_keys_inv: Optional[Sequence] = None,
        *,
        standard_normal: bool = True,
        loc: Optional[torch.Tensor] = None,
        scale: Optional[torch.Tensor] = None,
        in_keys: Optional[Sequence] = None,
        out_keys: Optional[Sequence] = None,
    ):
        if in_keys_inv is None:
            in_keys_inv = ["action"]
        if in_keys is None:
            in_keys = list(in_keys_inv) if in_keys_inv else ["action"]
        super().__init__(
            in_keys=in_keys,
            out_keys=out_keys,
            in_keys_inv=in_keys_inv,
            out_keys_inv=out_keys_inv,
        )
        self.standard_normal = standard_normal
        self._loc = None if loc is None else torch.as_tensor(loc)
        self._scale = None if scale is None else torch.as_tensor(scale)

    def _derive(self, spec: Bounded):
        low, high = spec.low, spec.high
        if self.standard_normal:
            self._loc = (high + low) / 2
            self._scale = (high - low) / 2
        else:
            self._loc = low
            self._scale = high - low

    def transform_action_spec(self, spec: Composite) -> Composite:
        for key in self.in_keys_inv:
            base = spec[key]
            if not isinstance(base, Bounded):
                raise TypeError("ActionScaling needs a Bounded action spec")
            self._derive(base)
            if self.standard_normal:
                spec[key] = Bounded(
                    low=-torch.ones_like(base.low),
                    high=torch.ones_like(base.high),
                    shape=base.shape,
                    device=base.device,
                    dtype=base.dtype,
                )
            else:
                spec[key] = Bounded(
                    low=torch.zeros_like(base.low),
                    high=torch.ones_like(base.high),
                    shape=base.shape,
                    device=base.device,
                    dtype=base.dtype,
                )
        return spec

    def _check_stats(self):
        if self._loc is None or self._scale is None:
            raise RuntimeError(
                "ActionScaling: attach to an env (spec-derived bounds) or pass loc/scale"
            )

    def _inv_apply_transform(self, action: torch.Tensor) -> torch.Tensor:
        self._check_stats()
        loc = self._loc.to(action.device)
        scale = self._scale.to(action.device)
        return action * scale + loc

    def _apply_transform(self, action: torch.Tensor) -> torch.Tensor:
        self._check_stats()
        loc = self._loc.to(action.device)
        scale = self._scale.to(action.device)
        return (action - loc) / scale


class ActionChunkTransform(Transform):
    """Build fixed-length VLA action chunks from a trajectory window
    (reference _action.py:1812): ``action [*B, T, A]`` →
    ``("vla_action", "chunk") [*B, T, H, A]`` gathering
    ``a[t..t+H-1]`` per step, plus a boolean ``action_is_pad``
    ``[*B, T, H]`` marking positions past the window end (filled by
    repeating the last action).  When ``done_key`` data is present,
    chunks are boundary-aware: positions past a done flag are padded.
    This is a pure data transform (replay-buffer / dataset side)."""

    def __init__(
        self,
        chunk_size: int,
        action_key: str = "action",
        out_key=("vla_action", "chunk"),
        pad_key: str = "action_is_pad",
        done_key=("next", "done"),
    ):
        super().__init__(in_keys=[action_key], out_keys=[unravel_key(out_key)])
        self.chunk_size = chunk_size
        self.action_key = action_key
        self.out_key = unravel_key(out_key)
        self.pad_key = pad_key
        self.done_key = unravel_key(done_key)

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        action = td.get(self.action_key)
        if action.dim() < 2:
            raise RuntimeError("ActionChunkTransform expects [*B, T, A] actions")
        *B, T, A = action.shape
        H = self.chunk_size
        device = action.device
        t_idx = torch.arange(T, device=device).unsqueeze(-1)  # [T, 1]
        h_idx = torch.arange(H, device=device).unsqueeze(0)  # [1, H]
        gather = (t_idx + h_idx).clamp_max(T - 1)  # [T, H]
        chunk = action[..., gather, :]  # [*B, T, H, A]
        is_pad = (t_idx + h_idx >= T).expand(*B, T, H).clone()
        done = td.get(self.done_key, None)
        if done is not None:
            d = done.reshape(*B, T).bool()
            # done_before[t, h] — a done strictly inside the window t..t+h-1
            cum = torch.cumsum(d.to(torch.int64), dim=-1)
            cum_prev = torch.nn.functional.pad(cum[..., :-1], (1, 0))
            # dones seen in steps t .. t+h-1 = cum[t+h-1] - cum_prev[t]
            end = (t_idx + h_idx - 1).clamp(0, T - 1)  # [T, H]
            seen = cum[..., end] - cum_prev[..., t_idx.expand(T, H)]
            is_pad |= (seen > 0) & (h_idx > 0)
        td.set(self.out_key, chunk)
        td.set(self.pad_key, is_pad)
        return td

    _call = forward

    def _inv_call(self, td):
        return td


class ActionTokenizerTransform(Transform):
    """Continuous action <-> token-id codec over an
    :class:`~rl_amd.data.vla.ActionTokenizerBase` (reference
    _action.py:2105).  ``mode="encode"``: forward writes token ids at
    ``out_key`` (training targets); inverse decodes token ids back to a
    continuous action (env action-input path).  ``mode="decode"``:
    forward decodes tokens → actions (policy-side use)."""

    def __init__(
        self,
        tokenizer,
        in_key: str = "action",
        out_key: str = "action_tokens",
        *,
        mode: str = "encode",
        strict: bool = False,
    ):
        if mode not in ("encode", "decode"):
            raise ValueError("mode must be 'encode' or 'decode'")
        super().__init__(in_keys=[in_key], out_keys=[out_key], in_keys_inv=[out_key], out_keys_inv=[in_key])
        self.tokenizer = tokenizer
        self.in_key = unravel_key(in_key)
        self.out_key = unravel_key(out_key)
        self.mode = mode
        self.strict = strict

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        if self.mode == "encode":
            action = td.get(self.in_key, None)
            if action is not None:
                td.set(self.out_key, self.tokenizer.encode(action))
        else:
            tokens = td.get(self.out_key, None)
            if tokens is not None:
                td.set(self.in_key, self.tokenizer.decode(tokens))
        return td

    _call = forward

    def _inv_call(self, td: TensorDictBase) -> TensorDictBase:
        tokens = td.get(self.out_key, None)
        if tokens is None:
            # env path with a rewritten Categorical action spec: the policy
            # emits token ids directly under the action key
            val = td.get(self.in_key, None)
            if val is not None and not val.dtype.is_floating_point:
                td.set(self.in_key, self.tokenizer.decode(val))
                return td
            if self.strict:
                raise KeyError(f"ActionTokenizerTransform: missing {self.out_key}")
            return td
        td.set(self.in_key, self.tokenizer.decode(tokens))
        return td

    def transform_action_spec(self, spec: Composite) -> Composite:
        if self.mode == "encode":
            base = spec[self.in_key]
            spec[self.in_key] = Categorical(
                self.tokenizer.vocab_size,
                shape=base.shape,
                device=base.device,
                dtype=torch.long,
            )
        return spec


# --------------------------------------------------------------------------- #
# rewards
# --------------------------------------------------------------------------- #
class LineariseRewards(Transform):
    """Weighted-sum a multi-objective reward into a scalar one
    (reference _reward.py:881)."""

    def __init__(self, in_keys: Sequence, out_keys: Optional[Sequence] = None, *, weights=None):
        super().__init__(in_keys=in_keys, out_keys=out_keys)
        self.weights = None if weights is None else torch.as_tensor(weights, dtype=torch.float32)

    def _apply_transform(self, reward: torch.Tensor) -> torch.Tensor:
        w = self.weights
        if w is None:
            return reward.sum(-1, keepdim=True)
        return (reward * w.to(reward.device)).sum(-1, keepdim=True)

    def transform_reward_spec(self, spec: Composite) -> Composite:
        for key in self.in_keys:
            base = spec[key]
            spec[key] = Unbounded(
                shape=(*base.shape[:-1], 1), device=base.device, dtype=base.dtype
            )
        return spec


class SuccessReward(Transform):
    """Sparse reward from a binary success signal: ``scale`` on success,
    0 otherwise (reference _reward.py:997)."""

    def __init__(self, success_key="success", reward_key="reward", *, scale: float = 1.0):
        super().__init__(in_keys=[success_key], out_keys=[reward_key])
        self.success_key = unravel_key(success_key)
        self.reward_key = unravel_key(reward_key)
        self.scale = scale

    def _call(self, td: TensorDictBase) -> TensorDictBase:
        success = td.get(self.success_key, None)
        if success is not None:
            td.set(self.reward_key, success.to(torch.float32) * self.scale)
        return td

    def _reset(self, td, td_reset):
        return td_reset  # reward written at step time only

    def transform_reward_spec(self, spec: Composite) -> Composite:
        parent = self.parent
        shape = (*spec.shape, 1)
        if parent is not None:
            try:
                shape = parent.base_env.full_observation_spec[self.success_key].shape
            except KeyError:
                pass
        spec[self.reward_key] = Bounded(
            low=0.0, high=self.scale, shape=shape, device=spec.device
        )
        return spec


# --------------------------------------------------------------------------- #
# keys / shapes
# --------------------------------------------------------------------------- #
class FlattenTensorDict(Transform):
    """Flatten batch dims on the replay-buffer *extend* (inverse) path;
    forward is identity (reference _keys.py:673).  Env use raises."""

    def forward(self, td):
        return td

    def _call(self, td):
        raise RuntimeError("FlattenTensorDict is a replay-buffer transform, not an env transform")

    def _inv_call(self, td: TensorDictBase) -> TensorDictBase:
        return td.reshape(-1)


class ExpandAs(Transform):
    """Expand ``in_key`` to the right to match ``ref_key``'s shape
    (reference _clip.py:168)."""

    def __init__(self, in_key, ref_key, out_key=None):
        out_key = in_key if out_key is None else out_key
        super().__init__(in_keys=[in_key], out_keys=[out_key])
        self.ref_key = unravel_key(ref_key)

    def _call(self, td: TensorDictBase) -> TensorDictBase:
        val = td.get(self.in_keys[0], None)
        ref = td.get(self.ref_key, None)
        if val is None or ref is None:
            return td
        while val.dim() < ref.dim():
            val = val.unsqueeze(-1)
        td.set(self.out_keys[0], val.expand_as(ref))
        return td


class NextObservationDelta(Transform):
    """Store ``("next", k)`` as a low-precision delta vs the root key
    (reference _observation.py:1521).  rl_amd form: the compression
    happens on the replay-buffer *extend* (inverse) path —
    ``("next","delta",k) = (next_k - k).to(delta_dtype)`` and the full
    ``("next", k)`` is dropped — and ``forward`` (sample path)
    reconstructs ``("next", k) = k + delta``.  Unlike
    :class:`NextStateReconstructor`, boundary transitions reconstruct
    exactly to the round-trip precision of ``delta_dtype``."""

    def __init__(
        self,
        in_keys: Sequence,
        *,
        delta_dtype: torch.dtype = torch.float16,
        drop_delta: bool = True,
    ):
        super().__init__(in_keys=in_keys)
        if not delta_dtype.is_floating_point:
            raise TypeError("delta_dtype must be floating point")
        self.delta_dtype = delta_dtype
        self.drop_delta = drop_delta

    def _inv_call(self, td: TensorDictBase) -> TensorDictBase:
        for k in self.in_keys:
            root = td.get(k, None)
            nxt = td.get(("next", *((k,) if isinstance(k, str) else k)), None)
            if root is None or nxt is None:
                continue
            delta = (nxt.float() - root.float()).to(self.delta_dtype)
            key = (k,) if isinstance(k, str) else k
            td.set(("next", "delta", *key), delta)
            td.exclude(("next", *key), inplace=True)
        return td

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        for k in self.in_keys:
            key = (k,) if isinstance(k, str) else k
            root = td.get(k, None)
            delta = td.get(("next", "delta", *key), None)
            if root is None or delta is None:
                continue
            td.set(("next", *key), root + delta.to(root.dtype))
            if self.drop_delta:
                td.exclude(("next", "delta", *key), inplace=True)
        return td

    _call = forward


# --------------------------------------------------------------------------- #
# replay-buffer-side transforms
# --------------------------------------------------------------------------- #
class MultiStepTransform(Transform):
    """n-step return accumulation on the replay-buffer extend path
    (reference rb_transforms.py:25).  Keeps the last ``n_steps`` frames
    in a local buffer so consecutive ``extend`` calls see a continuous
    stream — insensitive to the collector's ``frames_per_batch``."""

    def __init__(self, n_steps: int, gamma: float):
        super().__init__()
        from ...data.postprocs import MultiStep

        self.n_steps = n_steps
        self.gamma = gamma
        self._multistep = MultiStep(gamma=gamma, n_steps=n_steps)
        self._tail: Optional[TensorDictBase] = None

    def forward(self, td):
        return td

    def _inv_call(self, td: TensorDictBase) -> TensorDictBase:
        if td.batch_dims < 2:
            raise RuntimeError("MultiStepTransform expects [B, T] batches on extend")
        from ...tensordict import cat as td_cat

        if self._tail is not None:
            td = td_cat([self._tail, td], dim=-1)
        T = td.batch_size[-1]
        keep = max(0, min(self.n_steps, T))
        # hold back the trailing n_steps frames — their n-step targets
        # need future frames from the NEXT extend call
        self._tail = td[..., T - keep :].clone() if keep else None
        out_T = T - keep
        if out_T <= 0:
            # nothing ready to emit yet: emit an empty slice
            return td[..., :0]
        processed = self._multistep(td)
        return processed[..., :out_T]

    def flush(self) -> Optional[TensorDictBase]:
        """Emit the held-back tail (end of collection)."""
        if self._tail is None:
            return None
        out = self._multistep(self._tail)
        self._tail = None
        return out


class NextStateReconstructor(Transform):
    """Rebuild ``("next", k)`` at sample time by shifting root keys
    along the batch (reference rb_transforms.py:230): position ``i``
    takes ``data[k][i+1]`` when ``i+1`` is in-batch and shares the
    trajectory id; boundary positions become ``fill_value`` (NaN)."""

    def __init__(
        self,
        in_keys: Sequence,
        *,
        traj_key=("collector", "traj_ids"),
        fill_value: float = float("nan"),
    ):
        super().__init__(in_keys=in_keys)
        self.traj_key = unravel_key(traj_key)
        self.fill_value = fill_value

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        if td.batch_dims != 1:
            td = td.reshape(-1)
        traj = td.get(self.traj_key, None)
        n = td.batch_size[0]
        if traj is None:
            same = torch.zeros(n, dtype=torch.bool)
        else:
            traj = traj.reshape(n)
            same = torch.zeros(n, dtype=torch.bool, device=traj.device)
            if n > 1:
                same[:-1] = traj[:-1] == traj[1:]
        for k in self.in_keys:
            key = (k,) if isinstance(k, str) else k
            root = td.get(k, None)
            if root is None:
                continue
            nxt = torch.full_like(root, self.fill_value)
            if n > 1:
                mask = same[:-1]
                nxt[:-1][mask] = root[1:][mask]
            td.set(("next", *key), nxt)
        return td

    _call = forward

    def _inv_call(self, td):
        return td


class PolicyAgeFilter(Transform):
    """Drop data whose stamped behavior-policy version lags the live
    version by more than ``max_policy_lag`` (reference
    rb_transforms.py:466).  Filters on both the extend (inverse) and
    sample (forward) paths of a replay buffer; env use is a no-op."""

    def __init__(
        self,
        current_version: Union[int, Callable[[], int]],
        max_policy_lag: int,
        *,
        policy_version_key="policy_version",
        strict: bool = False,
    ):
        super().__init__()
        self._current = current_version
        self.max_policy_lag = max_policy_lag
        self.policy_version_key = unravel_key(policy_version_key)
        self.strict = strict
        self._warned = False

    def _version(self) -> int:
        return self._current() if callable(self._current) else self._current

    def _filter(self, td: TensorDictBase) -> TensorDictBase:
        stamped = td.get(self.policy_version_key, None)
        if stamped is None:
            if self.strict:
                raise KeyError(f"PolicyAgeFilter: missing {self.policy_version_key}")
            if not self._warned:
                warnings.warn("PolicyAgeFilter: no policy_version key; passing through")
                self._warned = True
            return td
        flat = td.reshape(-1) if td.batch_dims != 1 else td
        v = flat.get(self.policy_version_key).reshape(flat.batch_size[0])
        keep = (self._version() - v) <= self.max_policy_lag
        return flat[keep]

    def forward(self, td):
        return self._filter(td)

    def _inv_call(self, td):
        return self._filter(td)

    def _call(self, td):
        return td  # env data is produced by the live policy


# --------------------------------------------------------------------------- #
# module / timing / belief-state bridges
# --------------------------------------------------------------------------- #
class ModuleTransform(Transform):
    """Run an arbitrary (TensorDict)Module as a transform (reference
    module.py:123) — e.g. a learned world-model step or an encoder on
    the env output path, optionally on the inverse path too."""

    def __init__(self, module, *, inverse: bool = False, no_grad: bool = True):
        super().__init__()
        self.module = module
        self.inverse = inverse
        self.no_grad = no_grad

    def _run(self, td):
        if self.no_grad:
            with torch.no_grad():
                return self.module(td)
        return self.module(td)

    def _call(self, td):
        if self.inverse:
            return td
        return self._run(td)

    forward = _call

    def _inv_call(self, td):
        if not self.inverse:
            return td
        return self._run(td)


class Timer(Transform):
    """Measure wall-clock intervals between the env's ``inv`` (action
    in) and ``call`` (obs out) paths (reference _timer.py): writes
    ``time_step`` (inv→call: env compute) and ``time_policy``
    (call→inv: policy compute) in seconds."""

    def __init__(self, out_keys=("time_policy", "time_step")):
        super().__init__()
        self.time_policy_key, self.time_step_key = out_keys
        self._last_inv: Optional[float] = None
        self._last_call: Optional[float] = None

    def _inv_call(self, td):
        now = time.perf_counter()
        self._last_inv = now
        return td

    def _expand(self, value: float, td) -> torch.Tensor:
        return torch.full(tuple(td.batch_size), value)

    def _step(self, td, next_td):
        now = time.perf_counter()
        if self._last_inv is not None:
            next_td.set(self.time_step_key, self._expand(now - self._last_inv, next_td))
        if self._last_call is not None and self._last_inv is not None:
            next_td.set(
                self.time_policy_key,
                self._expand(self._last_inv - self._last_call, next_td),
            )
        self._last_call = now
        return next_td

    def _call(self, td):
        return td

    def transform_observation_spec(self, spec: Composite) -> Composite:
        for key in (self.time_policy_key, self.time_step_key):
            spec[key] = Unbounded(shape=spec.shape, device=spec.device)
        return spec


class EndOfLifeTransform(Transform):
    """ALE-style end-of-life signal (reference gym_transforms.py:20):
    reads a ``lives`` counter from the base env and writes ``eol`` True
    when a life is lost without the episode ending.  Requires the base
    env to expose a ``lives()`` method or ``lives`` attribute (the
    ALE-style contract); raises otherwise."""

    def __init__(self, eol_key="eol", lives_key="lives", done_key="done"):
        super().__init__(in_keys=[done_key], out_keys=[eol_key, lives_key])
        self.eol_key = unravel_key(eol_key)
        self.lives_key = unravel_key(lives_key)
        self.done_key = unravel_key(done_key)
        self._prev_lives: Optional[torch.Tensor] = None

    def _get_lives(self) -> torch.Tensor:
        base = self.parent.base_env if self.parent is not None else None
        lives = getattr(base, "lives", None)
        if lives is None:
            raise AttributeError(
                "EndOfLifeTransform requires a base env with a `lives` attribute/method "
                "(ALE-style); this env does not expose one"
            )
        if callable(lives):
            lives = lives()
        return torch.as_tensor(lives)

    def _step(self, td, next_td):
        lives = self._get_lives()
        done = next_td.get(self.done_key)
        if self._prev_lives is None:
            eol = torch.zeros_like(done)
        else:
            lost = (lives < self._prev_lives.to(lives.device)).reshape(done.shape)
            eol = lost & ~done
        self._prev_lives = lives
        next_td.set(self.eol_key, eol)
        next_td.set(self.lives_key, lives.expand(tuple(done.shape)).clone())
        return next_td

    def _call(self, td):
        return td

    def _reset(self, td, td_reset):
        self._prev_lives = None
        return td_reset

    def transform_observation_spec(self, spec: Composite) -> Composite:
        shape = (*spec.shape, 1)
        spec[self.eol_key] = Categorical(2, shape=shape, device=spec.device, dtype=torch.bool)
        spec[self.lives_key] = Unbounded(shape=shape, device=spec.device, dtype=torch.int64)
        return spec


class MeanActionSelector(Transform):
    """Bridge Gaussian belief-space policies (PILCO-style) to standard
    envs (reference mean_action_selector.py): forward wraps
    ``observation`` into ``("observation","mean")`` + zero
    ``("observation","var")``; inverse extracts ``("action","mean")``
    as the flat ``action``."""

    def __init__(self, observation_key: str = "observation", action_key: str = "action"):
        super().__init__(in_keys=[observation_key], in_keys_inv=[action_key])
        self.observation_key = observation_key
        self.action_key = action_key

    def _call(self, td: TensorDictBase) -> TensorDictBase:
        obs = td.get(self.observation_key, None)
        if obs is None or not isinstance(obs, torch.Tensor):
            return td
        var = torch.zeros(*obs.shape, obs.shape[-1], device=obs.device, dtype=obs.dtype)
        td.set(self.observation_key, TensorDict(
            {"mean": obs, "var": var}, batch_size=td.batch_size, device=td.device
        ))
        return td

    forward = _call

    def _inv_call(self, td: TensorDictBase) -> TensorDictBase:
        act = td.get(self.action_key, None)
        if act is not None and not isinstance(act, torch.Tensor):
            td.set(self.action_key, act.get("mean"))
        return td

    def transform_observation_spec(self, spec: Composite) -> Composite:
        base = spec[self.observation_key]
        spec[self.observation_key] = Composite(
            {
                "mean": base.clone(),
                "var": Unbounded(
                    shape=(*base.shape, base.shape[-1]), device=base.device, dtype=base.dtype
                ),
            },
            shape=spec.shape,
        )
        return spec
