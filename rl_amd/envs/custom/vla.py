"""ToyVLAEnv — synthetic env speaking the canonical VLA TensorDict schema.

Reference: pytorch/rl torchrl/envs/custom/vla.py:24 (ToyVLAEnv): camera
``("observation", "image")`` + proprioceptive ``("observation", "state")``
+ root ``language_instruction``; echo mode (effort penalty, never done)
and tracking mode (per-episode target in the state, success after k
consecutive in-tolerance steps).  rl_amd form: fully batched tensors on
the env device, no host round-trips in step.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from ...data.tensor_specs import Bounded, Categorical, Composite, Unbounded
from ...tensordict import TensorDict, TensorDictBase
from ..common import EnvBase

__all__ = ["ToyVLAEnv"]


class ToyVLAEnv(EnvBase):
    """Minimal VLA-schema env (see module docstring).

    Echo mode (``success_steps=None``): reward = -||action|| (effort
    penalty), never terminates — a plumbing smoke-test for VLA stacks
    (action chunking, MultiStepActorWrapper, recorders).

    Tracking mode (``success_steps=k``): a per-episode target action is
    sampled at reset and exposed at ``state[..., A:2A]``; ``success``
    turns True (and the episode ends) after ``k`` consecutive steps with
    ``|action - target|_inf < success_tol``; reward = -||action-target||.

    ``group_repeats`` (tracking, single env): replay the same target for
    N consecutive episodes and expose an integer ``group_id`` — the
    init-state control GRPO-style grouped advantages need.
    """

    _supports_masked_reset = True

    def __init__(
        self,
        action_dim: int = 4,
        state_dim: int = 6,
        image_shape: Tuple[int, int, int] = (3, 16, 16),
        instruction: str = "push the T-shaped block onto the target",
        *,
        from_pixels: bool = False,
        render_size: int = 64,
        success_steps: Optional[int] = None,
        success_tol: float = 0.25,
        group_repeats: Optional[int] = None,
        group_id_offset: int = 0,
        batch_size=(),
        device=None,
        seed: Optional[int] = None,
    ):
        super().__init__(device=device, batch_size=batch_size)
        self.action_dim = action_dim
        self.state_dim = state_dim
        self.image_shape = tuple(image_shape)
        self.instruction = instruction
        self.from_pixels = from_pixels
        self.render_size = render_size
        self.success_steps = success_steps
        self.success_tol = success_tol
        self.tracking = success_steps is not None
        if self.tracking and state_dim < 2 * action_dim:
            raise ValueError("tracking mode needs state_dim >= 2 * action_dim")
        if state_dim < action_dim:
            raise ValueError("state_dim must be >= action_dim")
        if group_repeats is not None:
            if not self.tracking:
                raise ValueError("group_repeats requires tracking mode")
            if len(self.batch_size):
                raise ValueError("group_repeats supports single envs only")
        self.group_repeats = group_repeats
        self.group_id_offset = group_id_offset
        self._gen = torch.Generator(device="cpu")
        if seed is not None:
            self._gen.manual_seed(seed)
        bs = self.batch_size

        obs = Composite(
            {
                "image": Bounded(
                    low=0,
                    high=255,
                    shape=(*bs, *self.image_shape),
                    dtype=torch.uint8,
                    device=self.device,
                ),
                "state": Unbounded(shape=(*bs, state_dim), device=self.device),
            },
            shape=bs,
            device=self.device,
        )
        spec = {"observation": obs}
        if self.tracking:
            spec["success"] = Categorical(
                2, shape=(*bs, 1), dtype=torch.bool, device=self.device
            )
        if self.group_repeats is not None:
            spec["group_id"] = Unbounded(
                shape=(*bs, 1), dtype=torch.int64, device=self.device
            )
        if from_pixels:
            spec["pixels"] = Bounded(
                low=0,
                high=255,
                shape=(*bs, render_size, render_size, 3),
                dtype=torch.uint8,
                device=self.device,
            )
        self.observation_spec = Composite(spec, shape=bs, device=self.device)
        self.action_spec = Bounded(
            low=-1.0, high=1.0, shape=(*bs, action_dim), device=self.device
        )
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)
        self._target: Optional[torch.Tensor] = None
        self._streak: Optional[torch.Tensor] = None
        self._last_action: Optional[torch.Tensor] = None
        self._episode_counter = 0

    # -- internals --------------------------------------------------------- #
    def _rand(self, *shape) -> torch.Tensor:
        return torch.rand(shape, generator=self._gen).to(self.device)

    def _sample_target(self) -> torch.Tensor:
        # targets in [-0.5, 0.5]: the tolerance ball fits inside the bounds
        return self._rand(*self.batch_size, self.action_dim) - 0.5

    def _state(self) -> torch.Tensor:
        bs = self.batch_size
        state = torch.zeros(*bs, self.state_dim, device=self.device)
        state[..., : self.action_dim] = self._last_action
        if self.tracking:
            state[..., self.action_dim : 2 * self.action_dim] = self._target
        return state

    def _image(self) -> torch.Tensor:
        # stand-in camera feed: random noise
        return (
            self._rand(*self.batch_size, *self.image_shape) * 255
        ).to(torch.uint8)

    def _render(self) -> torch.Tensor:
        """HWC uint8 frame: executed action red, target green (first two
        action dims mapped from the [-1,1] plane)."""
        S = self.render_size
        bs = self.batch_size
        canvas = torch.zeros(*bs, S, S, 3, dtype=torch.uint8, device=self.device)

        def to_px(coord):
            return ((coord.clamp(-1, 1) + 1) / 2 * (S - 1)).long()

        def draw(canvas, pts, channel):
            y, x = to_px(pts[..., 0]), to_px(pts[..., 1] if pts.shape[-1] > 1 else pts[..., 0])
            flat = canvas.reshape(-1, S, S, 3)
            yy, xx = y.reshape(-1), x.reshape(-1)
            for r in range(-2, 3):
                for c in range(-2, 3):
                    flat[
                        torch.arange(flat.shape[0], device=canvas.device),
                        (yy + r).clamp(0, S - 1),
                        (xx + c).clamp(0, S - 1),
                        channel,
                    ] = 255
            return flat.reshape(canvas.shape)

        canvas = draw(canvas, self._last_action, 0)  # red marker: action
        if self.tracking:
            canvas = draw(canvas, self._target, 1)  # green marker: target
        return canvas

    def _obs_dict(self) -> dict:
        bs = self.batch_size
        out = {
            "observation": TensorDict(
                {"image": self._image(), "state": self._state()},
                batch_size=bs,
                device=self.device,
            ),
            "done": torch.zeros(*bs, 1, dtype=torch.bool, device=self.device),
            "terminated": torch.zeros(*bs, 1, dtype=torch.bool, device=self.device),
        }
        if self.tracking:
            out["success"] = torch.zeros(*bs, 1, dtype=torch.bool, device=self.device)
        if self.group_repeats is not None:
            # counter was already advanced by the reset that started this episode
            gid = self.group_id_offset + max(0, self._episode_counter - 1) // self.group_repeats
            out["group_id"] = torch.full(
                (*bs, 1), gid, dtype=torch.int64, device=self.device
            )
        if self.from_pixels:
            out["pixels"] = self._render()
        return out

    # -- EnvBase ----------------------------------------------------------- #
    def _reset(self, tensordict: Optional[TensorDictBase] = None, **kwargs) -> TensorDictBase:
        bs = self.batch_size
        new_target = self._sample_target() if self.tracking else None
        if self.group_repeats is not None:
            # same target replayed group_repeats episodes in a row
            if self._episode_counter % self.group_repeats != 0 and self._target is not None:
                new_target = self._target
            self._episode_counter += 1
        new_streak = (
            torch.zeros(*bs, 1, device=self.device) if self.tracking else None
        )
        new_action = torch.zeros(*bs, self.action_dim, device=self.device)
        if tensordict is not None and "_reset" in tensordict and self._last_action is not None:
            mask = tensordict.get("_reset").reshape(*bs, 1)
            self._last_action = torch.where(mask, new_action, self._last_action)
            if self.tracking:
                self._target = torch.where(mask, new_target, self._target)
                self._streak = torch.where(mask, new_streak, self._streak)
        else:
            self._last_action = new_action
            if self.tracking:
                self._target = new_target
                self._streak = new_streak
        return TensorDict(self._obs_dict(), batch_size=bs, device=self.device)

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        bs = self.batch_size
        action = tensordict.get("action").clamp(-1, 1)
        self._last_action = action
        if self.tracking:
            err = action - self._target
            reward = -err.norm(dim=-1, keepdim=True)
            hit = (err.abs().amax(-1, keepdim=True) < self.success_tol).float()
            self._streak = (self._streak + 1) * hit
            success = self._streak >= self.success_steps
        else:
            reward = -action.norm(dim=-1, keepdim=True)
            success = None
        out = self._obs_dict()
        out["reward"] = reward
        if success is not None:
            out["success"] = success
            out["terminated"] = success.clone()
            out["done"] = success.clone()
        return TensorDict(out, batch_size=bs, device=self.device)

    def _set_seed(self, seed: Optional[int]):
        if seed is not None:
            self._gen.manual_seed(seed)
        return seed
