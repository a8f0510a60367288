"""Concrete transforms — the preprocessing/postprocessing vocabulary.

Reference: pytorch/rl torchrl/envs/transforms/ (_env.py, _observation.py,
_action.py, _reward.py, _tensor.py, _normalization.py, _keys.py,
_device.py, _misc.py, _clip.py).  Same names and semantics, compact
MI355X-first implementations (all pure-tensor ops; GPU-resident envs run
these on-device with no host sync).
"""
from __future__ import annotations

from typing import Any, List, Optional, Sequence, Union

import torch

from ...data.tensor_specs import (
    Binary,
    Bounded,
    Categorical,
    Composite,
    TensorSpec,
    Unbounded,
)
from ...tensordict import TensorDict, TensorDictBase, unravel_key
from ._base import Compose, Transform, TransformedEnv

__all__ = [
    "StepCounter",
    "InitTracker",
    "FrameSkipTransform",
    "TensorDictPrimer",
    "TrajCounter",
    "ObservationNorm",
    "CatFrames",
    "FlattenObservation",
    "UnsqueezeTransform",
    "SqueezeTransform",
    "PermuteTransform",
    "GrayScale",
    "ToTensorImage",
    "Resize",
    "CenterCrop",
    "Crop",
    "RewardClipping",
    "RewardScaling",
    "RewardSum",
    "BinarizeReward",
    "SignTransform",
    "TargetReturn",
    "CatTensors",
    "UnaryTransform",
    "ExcludeTransform",
    "SelectTransform",
    "RenameTransform",
    "DTypeCastTransform",
    "DoubleToFloat",
    "DeviceCastTransform",
    "FiniteTensorDictCheck",
    "TimeMaxPool",
    "PinMemoryTransform",
    "ClipTransform",
    "ActionMask",
    "DiscreteActionProjection",
    "StackTransform",
    "RemoveEmptySpecs",
    "NoopResetEnv",
]


# --------------------------------------------------------------------------- #
# Env-flow transforms (_env.py in reference)
# --------------------------------------------------------------------------- #
class StepCounter(Transform):
    """Count steps since reset under ``step_count``; optionally truncate at
    ``max_steps`` (reference _env.py:693)."""

    def __init__(
        self,
        max_steps: Optional[int] = None,
        truncated_key: str = "truncated",
        step_count_key: str = "step_count",
        update_done: bool = True,
    ):
        super().__init__(in_keys=[], out_keys=[])
        self.max_steps = max_steps
        self.truncated_key = truncated_key
        self.step_count_key = step_count_key
        self.update_done = update_done

    def _reset(self, td, td_reset):
        parent = self.parent
        shape = (*td_reset.batch_size, 1)
        prev = None if td is None else td.get(self.step_count_key, None)
        reset_mask = None if td is None else td.get("_reset", None)
        if prev is not None and reset_mask is not None:
            count = torch.where(
                reset_mask.reshape(shape), torch.zeros_like(prev), prev
            )
        else:
            count = torch.zeros(shape, dtype=torch.int64, device=td_reset.device)
        td_reset.set(self.step_count_key, count)
        return td_reset

    def _step(self, td, next_td):
        count = td.get(self.step_count_key, None)
        if count is None:
            count = torch.zeros(
                (*next_td.batch_size, 1), dtype=torch.int64, device=next_td.device
            )
        count = count + 1
        next_td.set(self.step_count_key, count)
        if self.max_steps is not None:
            trunc = count >= self.max_steps
            prev_trunc = next_td.get(self.truncated_key, None)
            if prev_trunc is not None:
                trunc = trunc | prev_trunc
            next_td.set(self.truncated_key, trunc)
            if self.update_done:
                done = next_td.get("done")
                next_td.set("done", done | trunc)
        return next_td

    def transform_observation_spec(self, spec):
        shape = (*spec.shape, 1)
        spec[self.step_count_key] = Unbounded(
            shape=shape, dtype=torch.int64, device=spec.device
        )
        return spec

    def transform_done_spec(self, spec):
        if self.max_steps is not None and self.truncated_key not in spec:
            spec[self.truncated_key] = Binary(
                shape=(*spec.shape, 1), device=spec.device
            )
        return spec


class InitTracker(Transform):
    """``is_init`` flag: True on the step right after a reset
    (reference _env.py:1497) — consumed by recurrent policies."""

    def __init__(self, init_key: str = "is_init"):
        super().__init__()
        self.init_key = init_key

    def _reset(self, td, td_reset):
        shape = (*td_reset.batch_size, 1)
        reset_mask = None if td is None else td.get("_reset", None)
        if reset_mask is None:
            flag = torch.ones(shape, dtype=torch.bool, device=td_reset.device)
        else:
            flag = reset_mask.reshape(shape).clone()
        td_reset.set(self.init_key, flag)
        return td_reset

    def _step(self, td, next_td):
        next_td.set(
            self.init_key,
            torch.zeros((*next_td.batch_size, 1), dtype=torch.bool, device=next_td.device),
        )
        return next_td

    def transform_observation_spec(self, spec):
        spec[self.init_key] = Binary(shape=(*spec.shape, 1), device=spec.device)
        return spec


class FrameSkipTransform(Transform):
    """Repeat each action ``frame_skip`` times, summing rewards
    (reference _env.py:74)."""

    def __init__(self, frame_skip: int = 1):
        super().__init__()
        if frame_skip < 1:
            raise ValueError("frame_skip must be >= 1")
        self.frame_skip = frame_skip

    def _step(self, td, next_td):
        parent = self.parent
        if parent is None:
            raise RuntimeError("FrameSkipTransform needs a parent env")
        base = parent.base_env
        reward = next_td.get("reward")
        for _ in range(self.frame_skip - 1):
            if bool(next_td.get("done").any()):
                break
            from ..utils import step_mdp

            cur = step_mdp(td, next_tensordict=next_td)
            for k in [k for k in td.keys(True, True) if unravel_key(k) == "action" or (isinstance(k, tuple) and k[-1] == "action")]:
                cur.set(k, td.get(k))
            cur = base.step(cur)
            next_td = cur.get("next")
            reward = reward + next_td.get("reward")
        next_td.set("reward", reward)
        return next_td


class NoopResetEnv(Transform):
    """Perform up to ``noops`` random steps after reset
    (reference _env.py:111)."""

    def __init__(self, noops: int = 30, random: bool = True):
        super().__init__()
        self.noops = noops
        self.random = random

    def _reset(self, td, td_reset):
        parent = self.parent
        if parent is None:
            return td_reset
        base = parent.base_env
        n = (
            int(torch.randint(0, self.noops + 1, (1,)).item())
            if self.random
            else self.noops
        )
        cur = td_reset
        for _ in range(n):
            cur = base.rand_action(cur)
            cur = base.step(cur)
            from ..utils import step_mdp

            cur = step_mdp(cur)
            if bool(cur.get("done", torch.zeros(1, dtype=torch.bool)).any()):
                cur = base.reset()
        return cur


class TensorDictPrimer(Transform):
    """Seed extra keys (e.g. RNN hidden state) into reset output and spec
    (reference _env.py:211)."""

    def __init__(
        self,
        primers: Optional[dict] = None,
        default_value: float = 0.0,
        random: bool = False,
        **kwargs,
    ):
        super().__init__()
        if primers is None:
            primers = kwargs
        self.primers = {unravel_key(k): v for k, v in primers.items()}
        self.default_value = default_value
        self.random = random

    def _expanded_shape(self, spec, td) -> tuple:
        # primer specs are batch-free: prepend the env batch dims
        return (*td.batch_size, *spec.shape)

    def _make_value(self, spec, td):
        shape = self._expanded_shape(spec, td)
        if self.random:
            return torch.randn(shape, device=td.device, dtype=spec.dtype)
        val = torch.zeros(shape, device=td.device, dtype=spec.dtype)
        if self.default_value:
            val = val + self.default_value
        return val

    def _reset(self, td, td_reset):
        for key, spec in self.primers.items():
            if td is not None and key in td and not (
                "_reset" in td and bool(td.get("_reset").all())
            ):
                td_reset.set(key, td.get(key))
            elif key not in td_reset:
                td_reset.set(key, self._make_value(spec, td_reset))
        return td_reset

    def _step(self, td, next_td):
        for key in self.primers:
            if key not in next_td and key in td:
                next_td.set(key, td.get(key))
            elif key not in next_td:
                next_td.set(key, self._make_value(self.primers[key], next_td))
        return next_td

    def transform_observation_spec(self, spec):
        for key, s in self.primers.items():
            try:
                spec[key] = s.expand(*spec.shape, *s.shape)
            except (NotImplementedError, RuntimeError):
                spec[key] = s
        return spec


class TrajCounter(Transform):
    """Global trajectory counter (reference _env.py:2305)."""

    def __init__(self, out_key: str = "traj_count"):
        super().__init__()
        self.out_key = out_key
        self._count = 0

    def _reset(self, td, td_reset):
        shape = (*td_reset.batch_size, 1)
        n = int(torch.tensor(shape[:-1]).prod().item()) if len(shape) > 1 else 1
        vals = torch.arange(
            self._count, self._count + n, device=td_reset.device
        ).reshape(shape)
        self._count += n
        td_reset.set(self.out_key, vals)
        return td_reset

    def _step(self, td, next_td):
        if self.out_key in td:
            next_td.set(self.out_key, td.get(self.out_key))
        return next_td

    def transform_observation_spec(self, spec):
        spec[self.out_key] = Unbounded(
            shape=(*spec.shape, 1), dtype=torch.int64, device=spec.device
        )
        return spec


# --------------------------------------------------------------------------- #
# Observation transforms (_observation.py)
# --------------------------------------------------------------------------- #
class ObservationNorm(Transform):
    """Affine normalization ``(obs - loc) / scale`` (standard_normal mode:
    ``(obs - loc) / scale``); stats can be initialized from rollouts
    (reference _normalization.py:52)."""

    def __init__(
        self,
        loc: Union[float, torch.Tensor, None] = None,
        scale: Union[float, torch.Tensor, None] = None,
        in_keys: Sequence = ("observation",),
        out_keys: Optional[Sequence] = None,
        standard_normal: bool = True,
        eps: float = 1e-6,
    ):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.standard_normal = standard_normal
        self.eps = eps
        self.register_buffer(
            "loc",
            torch.as_tensor(loc, dtype=torch.float32) if loc is not None else None,
        )
        self.register_buffer(
            "scale",
            torch.as_tensor(scale, dtype=torch.float32).clamp_min(eps)
            if scale is not None
            else None,
        )

    @property
    def initialized(self) -> bool:
        return self.loc is not None

    def init_stats(
        self,
        num_iter: int,
        reduce_dim: Union[int, Sequence[int], None] = None,
        cat_dim: int = 0,
        key: Optional[str] = None,
    ) -> None:
        parent = self.parent
        if parent is None:
            raise RuntimeError("init_stats needs a parent env")
        key = key if key is not None else self.in_keys[0]
        datas = []
        collected = 0
        while collected < num_iter:
            roll = parent.base_env.rollout(
                min(64, num_iter - collected), break_when_any_done=False
            )
            datas.append(roll.get(("next", key)))
            collected += roll.batch_size[-1]
        data = torch.cat(datas, cat_dim)
        if reduce_dim is None:
            # reduce every batch/time dim, keep feature dims
            n_feat = data.dim() - len(parent.batch_size) - 1
            reduce_dim = tuple(range(data.dim() - n_feat))
        loc = data.mean(reduce_dim, keepdim=False)
        scale = data.std(reduce_dim, keepdim=False).clamp_min(self.eps)
        self.register_buffer("loc", loc)
        self.register_buffer("scale", scale)

    def _apply_transform(self, obs):
        if self.loc is None:
            raise RuntimeError(
                "ObservationNorm stats not initialized; call init_stats or pass loc/scale"
            )
        loc = self.loc.to(obs.device)
        scale = self.scale.to(obs.device)
        if self.standard_normal:
            return (obs - loc) / scale
        return obs * scale + loc

    def _inv_apply_transform(self, obs):
        loc = self.loc.to(obs.device)
        scale = self.scale.to(obs.device)
        if self.standard_normal:
            return obs * scale + loc
        return (obs - loc) / scale

    def transform_observation_spec(self, spec):
        from ...data.tensor_specs import Unbounded as _U

        for in_key, out_key in zip(self.in_keys, self.out_keys):
            if in_key in spec:
                s = spec[in_key]
                spec[out_key] = _U(shape=s.shape, dtype=s.dtype, device=s.device)
        return spec


class CatFrames(Transform):
    """Stack the last N frames along ``dim`` (reference _observation.py:867).
    Keeps a ring buffer per in_key; reset re-fills with the first frame."""

    def __init__(
        self,
        N: int = 4,
        dim: int = -3,
        in_keys: Sequence = ("pixels",),
        out_keys: Optional[Sequence] = None,
        padding: str = "same",
    ):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.N = N
        self.dim = dim
        self.padding = padding
        self._buffers_map: dict = {}

    def _make_buf(self, key, val):
        reps = [1] * val.dim()
        reps[self.dim] = self.N
        if self.padding == "same":
            buf = val.repeat(*reps)
        else:
            shape = list(val.shape)
            shape[self.dim] *= self.N
            buf = torch.zeros(shape, dtype=val.dtype, device=val.device)
        return buf

    def _update(self, key, val, reset_mask=None):
        buf = self._buffers_map.get(key)
        if buf is None or buf.shape[: self.dim if self.dim >= 0 else val.dim() + self.dim] != val.shape[: self.dim if self.dim >= 0 else val.dim() + self.dim]:
            buf = self._make_buf(key, val)
            self._buffers_map[key] = buf
        d = val.shape[self.dim]
        if reset_mask is not None and reset_mask.any():
            refreshed = self._make_buf(key, val)
            m = reset_mask
            while m.dim() < buf.dim():
                m = m.unsqueeze(-1)
            buf = torch.where(m.expand_as(buf), refreshed, buf)
        buf = buf.roll(-d, dims=self.dim)
        idx = [slice(None)] * buf.dim()
        nd = self.dim if self.dim >= 0 else buf.dim() + self.dim
        idx[nd] = slice(buf.shape[nd] - d, buf.shape[nd])
        buf[tuple(idx)] = val
        self._buffers_map[key] = buf
        return buf.clone()

    def _reset(self, td, td_reset):
        reset_mask = None if td is None else td.get("_reset", None)
        if reset_mask is not None:
            reset_mask = reset_mask.reshape(td_reset.batch_size + (1,)).squeeze(-1)
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            val = td_reset.get(in_key, None)
            if val is None:
                continue
            if reset_mask is None:
                self._buffers_map.pop(in_key, None)
            out = self._update(
                in_key, val, reset_mask=torch.ones_like(val[..., 0], dtype=torch.bool).any(-1) if reset_mask is None and in_key in self._buffers_map else reset_mask
            )
            td_reset.set(out_key, out)
        return td_reset

    def _step(self, td, next_td):
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            val = next_td.get(in_key, None)
            if val is not None:
                next_td.set(out_key, self._update(in_key, val))
        return next_td

    def transform_observation_spec(self, spec):
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            if in_key in spec:
                s = spec[in_key].clone()
                shape = list(s.shape)
                shape[self.dim] *= self.N
                spec[out_key] = s.expand(*shape) if isinstance(s, Unbounded) else Unbounded(shape=shape, dtype=s.dtype, device=s.device)
        return spec


class _PerLeafTransform(Transform):
    """Helper base: forward per-leaf fn + spec shape update."""

    def _spec_shape(self, shape: torch.Size) -> torch.Size:
        return shape

    def transform_observation_spec(self, spec):
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            if in_key in spec:
                s = spec[in_key]
                example = s.zero()
                new = self._apply_transform(example)
                spec[out_key] = Unbounded(shape=new.shape, dtype=new.dtype, device=s.device)
        return spec


class FlattenObservation(_PerLeafTransform):
    """Flatten dims [first_dim, last_dim] of the observation
    (reference _observation.py:402)."""

    def __init__(self, first_dim: int, last_dim: int, in_keys=("observation",), out_keys=None):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.first_dim = first_dim
        self.last_dim = last_dim

    def _apply_transform(self, obs):
        return obs.flatten(self.first_dim, self.last_dim)


class UnsqueezeTransform(_PerLeafTransform):
    def __init__(self, dim: int, in_keys=("observation",), out_keys=None, **kwargs):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.dim = dim

    def _apply_transform(self, obs):
        return obs.unsqueeze(self.dim)

    def _inv_apply_transform(self, obs):
        return obs.squeeze(self.dim)


class SqueezeTransform(_PerLeafTransform):
    def __init__(self, dim: int, in_keys=("observation",), out_keys=None, **kwargs):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.dim = dim

    def _apply_transform(self, obs):
        return obs.squeeze(self.dim)

    def _inv_apply_transform(self, obs):
        return obs.unsqueeze(self.dim)


class PermuteTransform(_PerLeafTransform):
    def __init__(self, dims: Sequence[int], in_keys=("observation",), out_keys=None):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.dims = list(dims)

    def _apply_transform(self, obs):
        n_lead = obs.dim() - len(self.dims)
        dims = list(range(n_lead)) + [
            (d if d >= 0 else len(self.dims) + d) + n_lead for d in self.dims
        ]
        return obs.permute(*dims)


class GrayScale(_PerLeafTransform):
    """RGB → luma (reference _observation.py:828)."""

    def __init__(self, in_keys=("pixels",), out_keys=None):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)

    def _apply_transform(self, obs):
        w = torch.tensor([0.2989, 0.587, 0.114], device=obs.device, dtype=obs.dtype)
        return (obs * w.view(-1, 1, 1)).sum(-3, keepdim=True)


class ToTensorImage(_PerLeafTransform):
    """uint8 HWC [0,255] → float CHW [0,1] (reference _observation.py:56)."""

    def __init__(self, in_keys=("pixels",), out_keys=None, from_int: bool = True, dtype=torch.float32):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.from_int = from_int
        self.dtype = dtype

    def _apply_transform(self, obs):
        obs = obs.permute(*range(obs.dim() - 3), -1, -3, -2)
        obs = obs.to(self.dtype)
        if self.from_int:
            obs = obs / 255.0
        return obs


class Resize(_PerLeafTransform):
    """Bilinear resize of image observations (reference _observation.py:166)."""

    def __init__(self, w: int, h: Optional[int] = None, in_keys=("pixels",), out_keys=None, interpolation: str = "bilinear"):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.w = w
        self.h = h if h is not None else w
        self.interpolation = interpolation

    def _apply_transform(self, obs):
        lead = obs.shape[:-3]
        x = obs.reshape(-1, *obs.shape[-3:])
        x = torch.nn.functional.interpolate(
            x,
            size=(self.w, self.h),
            mode=self.interpolation,
            align_corners=False if self.interpolation in ("bilinear", "bicubic") else None,
        )
        return x.reshape(*lead, *x.shape[-3:])


class CenterCrop(_PerLeafTransform):
    def __init__(self, w: int, h: Optional[int] = None, in_keys=("pixels",), out_keys=None):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.w = w
        self.h = h if h is not None else w

    def _apply_transform(self, obs):
        H, W = obs.shape[-2:]
        top = (H - self.w) // 2
        left = (W - self.h) // 2
        return obs[..., top : top + self.w, left : left + self.h]


class Crop(_PerLeafTransform):
    def __init__(self, w: int, h: Optional[int] = None, top: int = 0, left: int = 0, in_keys=("pixels",), out_keys=None):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.w = w
        self.h = h if h is not None else w
        self.top = top
        self.left = left

    def _apply_transform(self, obs):
        return obs[..., self.top : self.top + self.w, self.left : self.left + self.h]


class TimeMaxPool(Transform):
    """Elementwise max over the last T observations
    (reference _misc.py:93)."""

    def __init__(self, in_keys=("observation",), out_keys=None, T: int = 1):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.T = T
        self._bufs: dict = {}

    def _update(self, key, val):
        buf = self._bufs.get(key)
        if buf is None or buf.shape[1:] != val.shape:
            buf = val.unsqueeze(0).repeat(self.T, *[1] * val.dim())
        buf = torch.roll(buf, -1, dims=0)
        buf[-1] = val
        self._bufs[key] = buf
        return buf.max(0).values

    def _reset(self, td, td_reset):
        self._bufs.clear()
        return self._call(td_reset)

    def _call(self, td):
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            val = td.get(in_key, None)
            if val is not None:
                td.set(out_key, self._update(in_key, val))
        return td


# --------------------------------------------------------------------------- #
# Reward transforms (_reward.py)
# --------------------------------------------------------------------------- #
class RewardClipping(Transform):
    def __init__(self, clamp_min: float = None, clamp_max: float = None, in_keys=("reward",), out_keys=None):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.clamp_min = clamp_min
        self.clamp_max = clamp_max

    def _apply_transform(self, r):
        return r.clamp(self.clamp_min, self.clamp_max)

    def transform_reward_spec(self, spec):
        from ...data.tensor_specs import Bounded as _B

        if self.clamp_min is not None and self.clamp_max is not None:
            for in_key, out_key in zip(self.in_keys, self.out_keys):
                if in_key in spec:
                    s = spec[in_key]
                    spec[out_key] = _B(
                        low=self.clamp_min, high=self.clamp_max,
                        shape=s.shape, dtype=s.dtype, device=s.device,
                    )
        return spec


class RewardScaling(Transform):
    """reward ← reward * scale + loc (reference _normalization.py:366)."""

    def __init__(self, loc: float = 0.0, scale: float = 1.0, in_keys=("reward",), out_keys=None, standard_normal: bool = False):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.loc = loc
        self.scale = scale
        self.standard_normal = standard_normal

    def _apply_transform(self, r):
        if self.standard_normal:
            return (r - self.loc) / self.scale
        return r * self.scale + self.loc

    def transform_reward_spec(self, spec):
        from ...data.tensor_specs import Unbounded as _U

        for in_key, out_key in zip(self.in_keys, self.out_keys):
            if in_key in spec:
                s = spec[in_key]
                spec[out_key] = _U(shape=s.shape, dtype=s.dtype, device=s.device)
        return spec


class RewardSum(Transform):
    """Running episode return under ``episode_reward``
    (reference _reward.py:373)."""

    def __init__(self, in_keys=("reward",), out_keys=("episode_reward",), reset_keys=None):
        super().__init__(in_keys=list(in_keys), out_keys=list(out_keys))

    def _reset(self, td, td_reset):
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            shape = (*td_reset.batch_size, 1)
            prev = None if td is None else td.get(out_key, None)
            reset_mask = None if td is None else td.get("_reset", None)
            if prev is not None and reset_mask is not None:
                acc = torch.where(
                    reset_mask.reshape(prev.shape), torch.zeros_like(prev), prev
                )
            else:
                acc = torch.zeros(shape, device=td_reset.device)
            td_reset.set(out_key, acc)
        return td_reset

    def _step(self, td, next_td):
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            prev = td.get(out_key, None)
            r = next_td.get(in_key)
            if prev is None:
                prev = torch.zeros_like(r)
            next_td.set(out_key, prev + r)
        return next_td

    def transform_observation_spec(self, spec):
        for out_key in self.out_keys:
            spec[out_key] = Unbounded(shape=(*spec.shape, 1), device=spec.device)
        return spec


class BinarizeReward(Transform):
    def __init__(self, in_keys=("reward",), out_keys=None):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)

    def _apply_transform(self, r):
        return (r > 0).to(r.dtype)

    def transform_reward_spec(self, spec):
        from ...data.tensor_specs import Bounded as _B

        for in_key, out_key in zip(self.in_keys, self.out_keys):
            if in_key in spec:
                s = spec[in_key]
                spec[out_key] = _B(low=0.0, high=1.0, shape=s.shape,
                                   dtype=s.dtype, device=s.device)
        return spec


class SignTransform(Transform):
    def __init__(self, in_keys=("reward",), out_keys=None):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)

    def _apply_transform(self, r):
        return r.sign()

    def transform_reward_spec(self, spec):
        from ...data.tensor_specs import Bounded as _B

        for in_key, out_key in zip(self.in_keys, self.out_keys):
            if in_key in spec:
                s = spec[in_key]
                spec[out_key] = _B(low=-1.0, high=1.0, shape=s.shape,
                                   dtype=s.dtype, device=s.device)
        return spec


class TargetReturn(Transform):
    """Decision-Transformer style return-to-go input
    (reference _reward.py:63)."""

    def __init__(self, target_return: float, mode: str = "reduce", in_keys=("reward",), out_keys=("target_return",)):
        super().__init__(in_keys=list(in_keys), out_keys=list(out_keys))
        self.target_return = target_return
        self.mode = mode

    def _reset(self, td, td_reset):
        shape = (*td_reset.batch_size, 1)
        td_reset.set(
            self.out_keys[0],
            torch.full(shape, self.target_return, device=td_reset.device),
        )
        return td_reset

    def _step(self, td, next_td):
        prev = td.get(self.out_keys[0], None)
        if prev is None:
            prev = torch.full(
                (*next_td.batch_size, 1), self.target_return, device=next_td.device
            )
        r = next_td.get(self.in_keys[0])
        if self.mode == "reduce":
            next_td.set(self.out_keys[0], prev - r)
        else:
            next_td.set(self.out_keys[0], prev)
        return next_td

    def transform_observation_spec(self, spec):
        spec[self.out_keys[0]] = Unbounded(shape=(*spec.shape, 1), device=spec.device)
        return spec


# --------------------------------------------------------------------------- #
# Tensor / key transforms
# --------------------------------------------------------------------------- #
class CatTensors(Transform):
    """Concatenate several keys into one along ``dim``
    (reference _tensor.py:45)."""

    def __init__(self, in_keys: Sequence, out_key: str = "observation_vector", dim: int = -1, del_keys: bool = True):
        super().__init__(in_keys=list(in_keys), out_keys=[out_key])
        self.dim = dim
        self.del_keys = del_keys

    def _call(self, td):
        vals = [td.get(k) for k in self.in_keys]
        td.set(self.out_keys[0], torch.cat(vals, self.dim))
        if self.del_keys:
            for k in self.in_keys:
                td.pop(k, None)
        return td

    def transform_observation_spec(self, spec):
        shapes = []
        dtype = None
        dev = spec.device
        for k in self.in_keys:
            s = spec[k]
            shapes.append(s.shape)
            dtype = s.dtype
            if self.del_keys:
                del spec[k]
        total = sum(sh[self.dim] for sh in shapes)
        base = list(shapes[0])
        base[self.dim] = total
        spec[self.out_keys[0]] = Unbounded(shape=base, dtype=dtype, device=dev)
        return spec


class StackTransform(Transform):
    """Stack several keys into one new leading dim (reference _tensor.py:991)."""

    def __init__(self, in_keys: Sequence, out_key: str, dim: int = 0, del_keys: bool = True):
        super().__init__(in_keys=list(in_keys), out_keys=[out_key])
        self.dim = dim
        self.del_keys = del_keys

    def _call(self, td):
        vals = [td.get(k) for k in self.in_keys]
        td.set(self.out_keys[0], torch.stack(vals, self.dim))
        if self.del_keys:
            for k in self.in_keys:
                td.pop(k, None)
        return td


class UnaryTransform(Transform):
    """Apply an arbitrary per-leaf function (reference _tensor.py:230)."""

    def __init__(self, in_keys, out_keys, fn, inv_fn=None):
        super().__init__(in_keys=list(in_keys), out_keys=list(out_keys))
        self.fn = fn
        self.inv_fn = inv_fn

    def _apply_transform(self, x):
        return self.fn(x)

    def _inv_apply_transform(self, x):
        if self.inv_fn is None:
            return x
        return self.inv_fn(x)


class ExcludeTransform(Transform):
    def __init__(self, *keys):
        super().__init__()
        self.keys = [unravel_key(k) for k in keys]

    def _call(self, td):
        for k in self.keys:
            td.pop(k, None)
        return td

    def transform_observation_spec(self, spec):
        for k in self.keys:
            try:
                del spec[k]
            except KeyError:
                pass
        return spec


class SelectTransform(Transform):
    def __init__(self, *keys, keep_dones: bool = True):
        super().__init__()
        self.keys = [unravel_key(k) for k in keys]
        self.keep_dones = keep_dones

    def _call(self, td):
        keep = set(self.keys)
        if self.keep_dones:
            keep |= {"done", "terminated", "truncated", "reward"}
        for k in list(td.keys(True, True)):
            kn = unravel_key(k)
            root = kn if isinstance(kn, str) else kn[0]
            if kn not in keep and root not in keep:
                td.pop(k, None)
        return td


class RenameTransform(Transform):
    """Rename keys forward (and inverse) (reference _keys.py:250)."""

    def __init__(self, in_keys, out_keys, in_keys_inv=None, out_keys_inv=None, create_copy: bool = False):
        super().__init__(
            in_keys=list(in_keys),
            out_keys=list(out_keys),
            in_keys_inv=list(in_keys_inv) if in_keys_inv else None,
            out_keys_inv=list(out_keys_inv) if out_keys_inv else None,
        )
        self.create_copy = create_copy

    def _call(self, td):
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            val = td.get(in_key, None)
            if val is not None:
                td.set(out_key, val)
                if not self.create_copy:
                    td.pop(in_key, None)
        return td

    def _inv_call(self, td):
        for in_key, out_key in zip(self.in_keys_inv, self.out_keys_inv):
            val = td.get(in_key, None)
            if val is not None:
                td.set(out_key, val)
                if not self.create_copy:
                    td.pop(in_key, None)
        return td

    def transform_observation_spec(self, spec):
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            if in_key in spec:
                spec[out_key] = spec[in_key]
                if not self.create_copy:
                    del spec[in_key]
        return spec


class RemoveEmptySpecs(Transform):
    """Drop empty Composite branches (reference _keys.py:453)."""

    def _call(self, td):
        for k in list(td.keys()):
            v = td.get(k)
            if isinstance(v, TensorDictBase) and v.is_empty():
                td.pop(k, None)
        return td

    def transform_observation_spec(self, spec):
        for k in list(spec.keys()):
            v = spec[k]
            if isinstance(v, Composite) and v.is_empty():
                del spec[k]
        return spec


# --------------------------------------------------------------------------- #
# Device / dtype transforms (_device.py)
# --------------------------------------------------------------------------- #
class DTypeCastTransform(Transform):
    """Cast matching leaves dtype_in → dtype_out on the way out (and back on
    the way in) (reference _device.py:43)."""

    def __init__(self, dtype_in: torch.dtype, dtype_out: torch.dtype, in_keys=None, out_keys=None, in_keys_inv=None, out_keys_inv=None):
        super().__init__(
            in_keys=in_keys, out_keys=out_keys, in_keys_inv=in_keys_inv, out_keys_inv=out_keys_inv
        )
        self.dtype_in = dtype_in
        self.dtype_out = dtype_out

    def _call(self, td):
        if self.in_keys:
            return super()._call(td)
        for k in list(td.keys(True, True)):
            v = td.get(k)
            if isinstance(v, torch.Tensor) and v.dtype == self.dtype_in:
                td.set(k, v.to(self.dtype_out))
        return td

    def _inv_call(self, td):
        if self.in_keys_inv:
            return super()._inv_call(td)
        # auto mode: with a parent env, only cast back the inputs the
        # BASE env actually declares as dtype_in (e.g. float64 MuJoCo
        # actions) — a float32-native env must keep float32 actions
        base_in = None
        parent = self.parent
        if parent is not None and hasattr(parent, "base_env"):
            base_in = parent.base_env.full_action_spec
        for k in list(td.keys(True, True)):
            v = td.get(k)
            if isinstance(v, torch.Tensor) and v.dtype == self.dtype_out:
                if base_in is not None:
                    sp = base_in.get(k, None) if hasattr(base_in, "get") else None
                    if sp is None or sp.dtype != self.dtype_in:
                        continue
                td.set(k, v.to(self.dtype_in))
        return td

    def _apply_transform(self, x):
        return x.to(self.dtype_out) if x.dtype == self.dtype_in else x

    def _inv_apply_transform(self, x):
        return x.to(self.dtype_in) if x.dtype == self.dtype_out else x

    def _cast_spec(self, spec):
        if isinstance(spec, Composite):
            for k in spec.keys():
                spec[k] = self._cast_spec(spec[k])
            return spec
        if spec.dtype == self.dtype_in:
            spec = spec.clone()
            spec.dtype = self.dtype_out
            if isinstance(spec, Bounded):
                spec.low = spec.low.to(self.dtype_out)
                spec.high = spec.high.to(self.dtype_out)
        return spec

    def transform_observation_spec(self, spec):
        return self._cast_spec(spec)

    def transform_reward_spec(self, spec):
        return self._cast_spec(spec)


class DoubleToFloat(DTypeCastTransform):
    """float64 → float32 (reference _device.py:415)."""

    def __init__(self, in_keys=None, out_keys=None, in_keys_inv=None, out_keys_inv=None):
        super().__init__(
            torch.float64, torch.float32, in_keys, out_keys, in_keys_inv, out_keys_inv
        )


class DeviceCastTransform(Transform):
    """Move the whole td to a device (reference _device.py:541)."""

    def __init__(self, device, orig_device=None):
        super().__init__()
        self.dest = torch.device(device)
        self.orig_device = torch.device(orig_device) if orig_device else None

    def _call(self, td):
        return td.to(self.dest)

    def _inv_call(self, td):
        if self.orig_device is not None:
            return td.to(self.orig_device)
        return td

    def transform_observation_spec(self, spec):
        return spec.to(self.dest)

    def transform_reward_spec(self, spec):
        return spec.to(self.dest)

    def transform_done_spec(self, spec):
        return spec.to(self.dest)


# --------------------------------------------------------------------------- #
# Misc
# --------------------------------------------------------------------------- #
class FiniteTensorDictCheck(Transform):
    """Raise on NaN/Inf anywhere (reference _misc.py:55)."""

    def _call(self, td):
        for k, v in td.items(True, True):
            if isinstance(v, torch.Tensor) and v.is_floating_point():
                if not torch.isfinite(v).all():
                    raise ValueError(f"found non-finite values in key {k}")
        return td


class PinMemoryTransform(Transform):
    def _call(self, td):
        return td.pin_memory()


class ClipTransform(Transform):
    """Clamp observations/rewards to [low, high] (reference _clip.py:37)."""

    def __init__(self, in_keys=("observation",), out_keys=None, low=None, high=None):
        super().__init__(in_keys=list(in_keys), out_keys=out_keys)
        self.low = low
        self.high = high

    def _apply_transform(self, x):
        return x.clamp(self.low, self.high)

    def _clip_spec(self, spec):
        from ...data.tensor_specs import Bounded as _B

        if self.low is None or self.high is None:
            return spec
        for in_key, out_key in zip(self.in_keys, self.out_keys):
            if in_key in spec:
                s = spec[in_key]
                spec[out_key] = _B(low=self.low, high=self.high,
                                   shape=s.shape, dtype=s.dtype, device=s.device)
        return spec

    def transform_observation_spec(self, spec):
        return self._clip_spec(spec)

    def transform_reward_spec(self, spec):
        return self._clip_spec(spec)


class ActionMask(Transform):
    """Mask invalid actions using a boolean mask key
    (reference _action.py:176)."""

    def __init__(self, action_key: str = "action", mask_key: str = "action_mask"):
        super().__init__()
        self.action_key = action_key
        self.mask_key = mask_key

    def _call(self, td):
        return td

    def _inv_call(self, td):
        mask = td.get(self.mask_key, None)
        action = td.get(self.action_key, None)
        if mask is not None and action is not None and action.dtype != torch.int64:
            td.set(self.action_key, action & mask)
        return td


class DiscreteActionProjection(Transform):
    """Project actions from a larger to a smaller discrete space
    (reference _action.py:64)."""

    def __init__(self, num_actions_effective: int, max_actions: int, action_key: str = "action", include_forward: bool = True):
        super().__init__(in_keys_inv=[action_key])
        self.num_actions_effective = num_actions_effective
        self.max_actions = max_actions

    def _inv_apply_transform(self, action):
        if action.dtype == torch.int64:
            return action.clamp_max(self.num_actions_effective - 1)
        # one-hot: truncate and renormalize
        out = action[..., : self.num_actions_effective]
        if not out.any(-1).all():
            idx = out.sum(-1) == 0
            fill = torch.nn.functional.one_hot(
                torch.zeros(
                    idx.sum(), dtype=torch.long, device=action.device
                ),
                self.num_actions_effective,
            ).to(out.dtype)
            out[idx] = fill
        return out
