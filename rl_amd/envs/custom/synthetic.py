"""Synthetic MuJoCo-shaped vectorized envs for benchmarking.

These reproduce the observation/action SHAPES and step cost of the classic
control suite on synthetic dynamics (no MuJoCo in the image; BASELINE.json
specifies synthetic observations).  Reference analog: the pure-torch
physics backends in pytorch/rl torchrl/envs/custom/mujoco/_backends.py:105
(`_TorchBackend`) — same idea (torch-native batched dynamics resident on
the GPU), independent implementation.

Dynamics: a fixed random stable linear system with tanh nonlinearity,
quadratic control cost and a velocity bonus — enough structure for PPO to
make measurable progress while every step stays a handful of fused
elementwise/GEMM kernels in HBM.
"""
from __future__ import annotations

from typing import Optional

import torch

from ...data.tensor_specs import Bounded, Composite, Unbounded
from ...tensordict import TensorDict, TensorDictBase
from ..common import EnvBase

__all__ = ["SyntheticMuJoCoEnv", "HalfCheetahVec", "HumanoidVec", "AntVec"]


class SyntheticMuJoCoEnv(EnvBase):
    """Batched synthetic control env with MuJoCo-style IO shapes."""

    _supports_masked_reset = True

    OBS_DIM = 17
    ACT_DIM = 6

    def __init__(
        self,
        batch_size=(),
        device=None,
        obs_dim: Optional[int] = None,
        act_dim: Optional[int] = None,
        max_steps: int = 1000,
        dtype: torch.dtype = torch.float32,
        seed: int = 0,
    ):
        super().__init__(device=device, batch_size=batch_size)
        self.obs_dim = obs_dim or self.OBS_DIM
        self.act_dim = act_dim or self.ACT_DIM
        self.max_steps = max_steps
        self.dtype = dtype
        bs = self.batch_size
        self.observation_spec = Composite(
            {
                "observation": Unbounded(
                    shape=(*bs, self.obs_dim), device=self.device, dtype=dtype
                )
            },
            shape=bs,
            device=self.device,
        )
        self.action_spec = Bounded(
            low=-1.0, high=1.0, shape=(*bs, self.act_dim), device=self.device, dtype=dtype
        )
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device, dtype=dtype)
        gen = torch.Generator().manual_seed(seed)
        # stable random dynamics (spectral radius < 1)
        A = torch.randn(self.obs_dim, self.obs_dim, generator=gen)
        A = 0.95 * A / torch.linalg.matrix_norm(A, 2)
        B = torch.randn(self.act_dim, self.obs_dim, generator=gen) * 0.2
        self.register_buffer("A", A.to(self.device, dtype))
        self.register_buffer("B", B.to(self.device, dtype))
        self._state: Optional[torch.Tensor] = None
        self._t: Optional[torch.Tensor] = None
        self._gen = torch.Generator(device="cpu")
        self._capture_safe = False
        self._zeros_done: Optional[torch.Tensor] = None

    def enable_capture_mode(self, mode: bool = True) -> "SyntheticMuJoCoEnv":
        """In-place state updates for hipGraph capture."""
        self._capture_safe = mode
        return self

    def _reset(self, tensordict=None, **kwargs) -> TensorDictBase:
        bs = self.batch_size
        if self.device.type == "cuda":
            # device-side RNG: masked per-step resets stay fully on-GPU
            new_state = (
                torch.randn((*bs, self.obs_dim), device=self.device, dtype=self.dtype)
                * 0.1
            )
        else:
            new_state = torch.randn(
                (*bs, self.obs_dim), generator=self._gen
            ).to(self.device, self.dtype) * 0.1
        new_t = torch.zeros((*bs, 1), device=self.device)
        if tensordict is not None and "_reset" in tensordict and self._state is not None:
            mask = tensordict.get("_reset").reshape(*bs, 1)
            if self._capture_safe:
                self._state.copy_(torch.where(mask, new_state, self._state))
                self._t.copy_(torch.where(mask, new_t, self._t))
            else:
                self._state = torch.where(mask, new_state, self._state)
                self._t = torch.where(mask, new_t, self._t)
        else:
            self._state = new_state
            self._t = new_t
        return TensorDict(
            {
                "observation": self._state.clone(),
                "done": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
                "terminated": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
            },
            batch_size=bs,
            device=self.device,
        )

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        if (
            self._capture_safe
            and self.device.type == "cuda"
            and self.dtype == torch.float32
        ):
            fused = self._fused_step(tensordict)
            if fused is not None:
                return fused
        action = tensordict.get("action").to(self.dtype).clamp(-1, 1)
        s = self._state
        drive = action @ self.B
        new_state = torch.tanh(s @ self.A + drive)
        # in-place state update keeps the buffers stable — required for
        # hipGraph capture of the rollout loop (replays rewrite the same
        # memory; a rebinding would read stale state on replay)
        self._state = self._state.copy_(new_state) if self._capture_safe else new_state
        self._t = self._t.add_(1) if self._capture_safe else self._t + 1
        # forward-progress reward: first obs component is "velocity"
        vel = self._state[..., :1]
        ctrl_cost = 0.1 * action.pow(2).sum(-1, keepdim=True)
        reward = vel - ctrl_cost
        truncated = self._t >= self.max_steps
        bs = self.batch_size
        return TensorDict(
            {
                "observation": self._state.clone(),
                "reward": reward.to(self.dtype),
                "done": truncated,
                "terminated": torch.zeros_like(truncated),
                "truncated": truncated,
            },
            batch_size=bs,
            device=self.device,
        )

    def _fused_step(self, tensordict: TensorDictBase) -> Optional[TensorDictBase]:
        """One-kernel transition (csrc/env_step.hip): the eager step is
        ~12 launches of ~4 us inside the captured rollout.  Returns
        ``None`` (eager fallback) when the extension is absent or the
        state dims exceed the 160 KB LDS budget (e.g. Humanoid's
        376-dim state: the A matrix alone is 565 KB)."""
        if getattr(self, "_fused_step_ok", None) is False:
            return None
        from ... import ops

        if not ops.HAS_HIP_EXT:
            self._fused_step_ok = False
            return None
        from ... import _C

        if getattr(self, "_fused_step_ok", None) is None:
            apad = self.obs_dim + 1
            lds = 4 * (self.obs_dim * apad + self.act_dim * apad + 16 * apad
                       + 16 * self.act_dim + 16)
            self._fused_step_ok = lds <= 160 * 1024
            if not self._fused_step_ok:
                return None

        bs = self.batch_size
        action = tensordict.get("action")
        if action.dtype != torch.float32:
            action = action.float()
        obs, reward, done = _C.synthetic_env_step(
            self._state, action.contiguous(), self.A, self.B,
            self._t.reshape(-1), float(self.max_steps),
        )
        if self._zeros_done is None or self._zeros_done.shape[0] != done.shape[0]:
            self._zeros_done = torch.zeros_like(done)
        return TensorDict(
            {
                "observation": obs,
                "reward": reward,
                "done": done,
                "terminated": self._zeros_done,
                "truncated": done,
            },
            batch_size=bs,
            device=self.device,
        )

    def _set_seed(self, seed):
        if seed is not None:
            self._gen.manual_seed(seed)
        return seed

    # ------------------------------------------------------------------ #
    # Whole-rollout mega-kernel protocol (consumed by GraphedRollout /
    # Collector fast path): env rows are independent, so the entire
    # T-step rollout of (fused actor + this env) runs as ONE kernel
    # (csrc/rollout_fused.hip).
    # ------------------------------------------------------------------ #
    def supports_fused_rollout(self, policy) -> bool:
        """True when (policy, self) can run as the single-launch rollout
        mega-kernel: GPU-resident fp32 env, 3-Linear fused TanhNormal
        actor, and the weight set fits the 160 KB LDS budget."""
        if self.device is None or self.device.type != "cuda" or self.dtype != torch.float32:
            return False
        from ... import ops

        if not ops.HAS_HIP_EXT:
            return False
        from ... import _C

        if not hasattr(_C, "fused_rollout"):
            return False
        if not isinstance(policy, ops.FusedTanhNormalActor):
            return False
        linears = policy.linears
        S, H1, H2 = self.obs_dim, linears[0].out_features, linears[1].out_features
        if linears[0].in_features != S or linears[2].out_features != 2 * self.act_dim:
            return False
        return _C.fused_rollout_lds_ok(S, H1, H2, self.act_dim) if hasattr(
            _C, "fused_rollout_lds_ok"
        ) else self._fused_rollout_lds_ok(S, H1, H2, self.act_dim)

    @staticmethod
    def _fused_rollout_lds_ok(S, H1, H2, A):
        # mirror of csrc/rollout_fused.hip fused_rollout_lds_bytes
        RO_ROWS = 8
        w1s, w2s, w3s, as_ = S | 1, H1 | 1, H2 | 1, S | 1
        bufw = max(S, H1, H2, 2 * A)
        floats = (
            H1 * w1s + H1 + H2 * w2s + H2 + 2 * A * w3s + 2 * A
            + S * as_ + A * as_ + RO_ROWS * as_ + RO_ROWS * A
            + 2 * RO_ROWS * bufw + RO_ROWS
        )
        return 4 * floats <= 160 * 1024

    def fused_rollout_into(self, policy, store) -> None:
        """Run one [B, T] rollout with ``policy`` (a FusedTanhNormalActor)
        into the pre-allocated store (keys: observation, action,
        sample_log_prob, next/observation, next/reward, next/done).
        Carried state auto-resets; the caller owns loop closure."""
        from ... import _C

        B, T = store.batch_size
        if self._state is None:
            self.reset()
        w1, w2, w3 = policy.linears
        eps_all = torch.randn(T, B, self.act_dim, device=self.device)
        noise_all = torch.randn(T, B, self.obs_dim, device=self.device) * 0.1
        # bf16 weight caches (when the actor has them): the MFMA rollout
        # variant computes the policy on the matrix cores with the SAME
        # weights the update phase uses
        bf16_w = []
        import os

        if os.environ.get("RL_AMD_ROLLOUT_MFMA", "1") != "0":
            for lin in (w1, w2, w3):
                wb = getattr(lin, "weight_bf16", None)
                bb = getattr(lin, "bias_bf16", None)
                if wb is None or bb is None:
                    bf16_w = []
                    break
                bf16_w += [wb, bb]
        _C.fused_rollout(
            self._state,
            self._t.reshape(-1),
            w1.weight, w1.bias,
            w2.weight, w2.bias,
            w3.weight, w3.bias,
            self.A, self.B,
            eps_all, noise_all,
            store.get("observation"),
            store.get("action"),
            store.get("sample_log_prob"),
            store.get(("next", "observation")),
            store.get(("next", "reward")),
            store.get(("next", "done")),
            float(self.max_steps),
            policy.inv_softplus_bias,
            policy.scale_lb,
            bf16_w,
        )


class HalfCheetahVec(SyntheticMuJoCoEnv):
    """HalfCheetah-v4 shapes: obs 17, act 6."""

    OBS_DIM = 17
    ACT_DIM = 6


class HumanoidVec(SyntheticMuJoCoEnv):
    """Humanoid-v4 shapes: obs 376, act 17."""

    OBS_DIM = 376
    ACT_DIM = 17


class AntVec(SyntheticMuJoCoEnv):
    """Ant-v4 shapes: obs 27, act 8."""

    OBS_DIM = 27
    ACT_DIM = 8
