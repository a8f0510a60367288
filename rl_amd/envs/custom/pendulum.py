"""Batched pure-torch Pendulum — the canonical GPU-resident env.

Reference: pytorch/rl torchrl/envs/custom/pendulum.py (PendulumEnv) —
classic Gym Pendulum-v1 dynamics re-implemented batched: the whole env
state lives in one tensor on the device, so 4096 envs step in a handful
of fused elementwise kernels with zero host traffic.
"""
from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ...data.tensor_specs import Bounded, Composite, Unbounded
from ...tensordict import TensorDict, TensorDictBase
from ..common import EnvBase

__all__ = ["PendulumEnv"]

DEFAULT_X = np.pi
DEFAULT_Y = 1.0


class PendulumEnv(EnvBase):
    metadata = {"render_modes": []}
    batch_locked = False
    _supports_masked_reset = True

    def __init__(
        self,
        batch_size=(),
        device=None,
        g: float = 10.0,
        m: float = 1.0,
        l: float = 1.0,  # noqa: E741
        dt: float = 0.05,
        max_speed: float = 8.0,
        max_torque: float = 2.0,
    ):
        super().__init__(device=device, batch_size=batch_size)
        self.g = g
        self.m = m
        self.l = l
        self.dt = dt
        self.max_speed = max_speed
        self.max_torque = max_torque
        bs = self.batch_size
        self.observation_spec = Composite(
            {
                "observation": Bounded(
                    low=torch.tensor([-1.0, -1.0, -max_speed]),
                    high=torch.tensor([1.0, 1.0, max_speed]),
                    shape=(*bs, 3),
                    device=self.device,
                )
            },
            shape=bs,
            device=self.device,
        )
        self.action_spec = Bounded(
            low=-max_torque, high=max_torque, shape=(*bs, 1), device=self.device
        )
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)
        self._th: Optional[torch.Tensor] = None
        self._thdot: Optional[torch.Tensor] = None
        self._gen = torch.Generator(device="cpu")
        self._capture_safe = False

    def enable_capture_mode(self, mode: bool = True) -> "PendulumEnv":
        """In-place state updates + device-side RNG so the step/reset
        loop is hipGraph-capturable (no host syncs, stable buffers)."""
        self._capture_safe = mode
        return self

    def _obs(self) -> torch.Tensor:
        return torch.stack(
            [self._th.cos(), self._th.sin(), self._thdot], dim=-1
        )

    def _make_td(self, extra: Optional[dict] = None) -> TensorDictBase:
        bs = self.batch_size
        data = {
            "observation": self._obs(),
            "done": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
            "terminated": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
        }
        if extra:
            data.update(extra)
        return TensorDict(data, batch_size=bs, device=self.device)

    def _reset(self, tensordict: Optional[TensorDictBase] = None, **kwargs) -> TensorDictBase:
        bs = self.batch_size
        high_th = DEFAULT_X
        high_thdot = DEFAULT_Y
        if self.device is not None and self.device.type == "cuda":
            # device-side RNG (default CUDA generator is graph-safe)
            new_th = torch.rand(bs or (), device=self.device) * 2 * high_th - high_th
            new_thdot = (
                torch.rand(bs or (), device=self.device) * 2 * high_thdot - high_thdot
            )
        else:
            new_th = (
                torch.rand(bs or (), generator=self._gen).to(self.device) * 2 * high_th
                - high_th
            )
            new_thdot = (
                torch.rand(bs or (), generator=self._gen).to(self.device) * 2 * high_thdot
                - high_thdot
            )
        if (
            tensordict is not None
            and "_reset" in tensordict
            and self._th is not None
        ):
            mask = tensordict.get("_reset").reshape(bs or ())
            if self._capture_safe:
                self._th.copy_(torch.where(mask, new_th, self._th))
                self._thdot.copy_(torch.where(mask, new_thdot, self._thdot))
            else:
                self._th = torch.where(mask, new_th, self._th)
                self._thdot = torch.where(mask, new_thdot, self._thdot)
        else:
            self._th = new_th
            self._thdot = new_thdot
        return self._make_td()

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        u = tensordict.get("action").squeeze(-1).clamp(
            -self.max_torque, self.max_torque
        )
        th, thdot = self._th, self._thdot
        g, m, l, dt = self.g, self.m, self.l, self.dt
        th_norm = ((th + np.pi) % (2 * np.pi)) - np.pi
        costs = th_norm.pow(2) + 0.1 * thdot.pow(2) + 0.001 * u.pow(2)
        newthdot = thdot + (3 * g / (2 * l) * th.sin() + 3.0 / (m * l**2) * u) * dt
        newthdot = newthdot.clamp(-self.max_speed, self.max_speed)
        newth = th + newthdot * dt
        if self._capture_safe:
            # stable buffers: hipGraph replays rewrite the same memory
            self._th.copy_(newth)
            self._thdot.copy_(newthdot)
        else:
            self._th = newth
            self._thdot = newthdot
        bs = self.batch_size
        return self._make_td({"reward": -costs.reshape(*bs, 1)})

    def _set_seed(self, seed: Optional[int]):
        if seed is not None:
            self._gen.manual_seed(seed)
            if self.device is not None and self.device.type == "cuda":
                torch.cuda.manual_seed(seed)
        return seed
