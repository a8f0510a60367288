"""Robot macro-action primitives: URScript-style arm control and
satellite attitude macros.

Reference: pytorch/rl torchrl/envs/custom/mujoco/_ur_primitives.py:134
(URScriptPrimitive, RobotMacroActionMode, RobotMacroAction,
CartesianSolver, URScriptPrimitiveTransform) and
_satellite_primitives.py (SatelliteMacroAction,
SatelliteAttitudeTransform).  These extend the generic
WAIT/MOVE macro vocabulary (transforms/_transforms4.py) with joint,
Cartesian, gripper and reset moves; the expansion to low-level action
sequences is pure batched torch, so macro expansion for thousands of
envs stays on-device.
"""
from __future__ import annotations

from enum import IntEnum
from typing import Optional, Protocol, Tuple

import torch

from ..transforms._transforms4 import (
    MacroAction,
    MacroPrimitive,
    MacroPrimitiveTransform,
)

__all__ = [
    "URScriptPrimitive",
    "RobotMacroActionMode",
    "RobotMacroAction",
    "CartesianSolver",
    "URScriptPrimitiveTransform",
    "SatelliteMacroAction",
    "SatelliteAttitudeTransform",
]


class URScriptPrimitive(IntEnum):
    """URScript-style primitive ids: the generic WAIT/MOVE vocabulary
    plus Cartesian moves and a binary gripper."""

    WAIT = int(MacroPrimitive.WAIT)
    MOVEJ = int(MacroPrimitive.MOVE)
    MOVEL = 2
    OPEN_GRIPPER = 3
    CLOSE_GRIPPER = 4

    def __str__(self) -> str:
        return self.name.lower()


class RobotMacroActionMode(IntEnum):
    """Readable modes for :class:`RobotMacroAction`; RESET homes the
    arm via the parent env's ``robot_home_qpos``."""

    WAIT = int(URScriptPrimitive.WAIT)
    REACH_JOINTS = int(URScriptPrimitive.MOVEJ)
    REACH_POSE = int(URScriptPrimitive.MOVEL)
    OPEN_GRIPPER = int(URScriptPrimitive.OPEN_GRIPPER)
    CLOSE_GRIPPER = int(URScriptPrimitive.CLOSE_GRIPPER)
    RESET = len(URScriptPrimitive)


class CartesianSolver(Protocol):
    """Inverse-kinematics callback: maps a Cartesian target
    (position [..., 3], quaternion [..., 4]) plus the current joint
    configuration to a joint-space target."""

    def __call__(
        self,
        position: torch.Tensor,
        quaternion: torch.Tensor,
        current_qpos: torch.Tensor,
    ) -> torch.Tensor: ...


def _identity_quat(pos: torch.Tensor) -> torch.Tensor:
    q = torch.zeros(*pos.shape[:-1], 4, device=pos.device, dtype=pos.dtype)
    q[..., 0] = 1.0
    return q


class RobotMacroAction(MacroAction):
    """Structured arm macro: a mode plus (mode-dependent) joint target,
    Cartesian pose target and gripper command, expanded to a low-level
    joint-action sequence by :class:`URScriptPrimitiveTransform`."""

    def __init__(self, mode, steps, *, joints: Optional[torch.Tensor] = None,
                 position: Optional[torch.Tensor] = None,
                 quaternion: Optional[torch.Tensor] = None,
                 gripper: Optional[int] = None, settle_steps=0):
        super().__init__(torch.as_tensor(int(mode)), torch.as_tensor(steps),
                         torch.as_tensor(settle_steps))
        self.joints = joints
        self.position = position
        self.quaternion = quaternion
        self.gripper = gripper

    @classmethod
    def wait(cls, steps: int = 1):
        return cls(RobotMacroActionMode.WAIT, steps)

    @classmethod
    def reach_joints(cls, joints: torch.Tensor, steps: int = 10, settle_steps: int = 0):
        return cls(RobotMacroActionMode.REACH_JOINTS, steps,
                   joints=torch.as_tensor(joints), settle_steps=settle_steps)

    @classmethod
    def reach_pose(cls, position: torch.Tensor, quaternion: Optional[torch.Tensor] = None,
                   steps: int = 10, settle_steps: int = 0):
        position = torch.as_tensor(position)
        if quaternion is None:
            quaternion = _identity_quat(position)
        return cls(RobotMacroActionMode.REACH_POSE, steps, position=position,
                   quaternion=torch.as_tensor(quaternion), settle_steps=settle_steps)

    @classmethod
    def open_gripper(cls, steps: int = 1):
        return cls(RobotMacroActionMode.OPEN_GRIPPER, steps, gripper=1)

    @classmethod
    def close_gripper(cls, steps: int = 1):
        return cls(RobotMacroActionMode.CLOSE_GRIPPER, steps, gripper=0)

    @classmethod
    def home(cls, home_qpos: torch.Tensor, steps: int = 10):
        return cls(RobotMacroActionMode.REACH_JOINTS, steps,
                   joints=torch.as_tensor(home_qpos))

    @classmethod
    def reset(cls, steps: int = 1):
        return cls(RobotMacroActionMode.RESET, steps)


class URScriptPrimitiveTransform(MacroPrimitiveTransform):
    """Expand :class:`RobotMacroAction` into a ``[..., T, A]`` joint
    sequence (reference _ur_primitives.py:607): joint moves linearly
    interpolate; Cartesian moves run through the ``solver``; gripper
    moves hold the arm and toggle the trailing gripper channel; RESET
    interpolates to ``home_qpos``."""

    def __init__(self, action_key: str = "macro_action", out_key: str = "action",
                 *, solver: Optional[CartesianSolver] = None,
                 home_qpos: Optional[torch.Tensor] = None,
                 gripper_dim: bool = True):
        super().__init__(action_key=action_key, out_key=out_key)
        self.solver = solver
        self.home_qpos = None if home_qpos is None else torch.as_tensor(home_qpos)
        self.gripper_dim = gripper_dim

    def _interp(self, start: torch.Tensor, target: torch.Tensor, steps: int,
                settle: int) -> torch.Tensor:
        w = torch.linspace(1.0 / steps, 1.0, steps, device=start.device)
        w = w.reshape(*(1,) * (start.dim() - 1), steps, 1)
        seq = start.unsqueeze(-2) * (1 - w) + target.unsqueeze(-2) * w
        if settle:
            tail = target.unsqueeze(-2).expand(*target.shape[:-1], settle, target.shape[-1])
            seq = torch.cat([seq, tail], dim=-2)
        return seq

    def expand_macro(self, macro: MacroAction, start: torch.Tensor) -> torch.Tensor:
        if not isinstance(macro, RobotMacroAction):
            return super().expand_macro(macro, start)
        steps = int(macro.steps)
        settle = int(macro.settle_steps)
        mode = RobotMacroActionMode(int(macro.mode))
        arm, grip = (start[..., :-1], start[..., -1:]) if self.gripper_dim else (start, None)
        if mode is RobotMacroActionMode.WAIT:
            seq = start.unsqueeze(-2).expand(*start.shape[:-1], steps + settle,
                                             start.shape[-1]).clone()
            return seq
        if mode is RobotMacroActionMode.REACH_JOINTS:
            target = macro.joints.to(arm.device, arm.dtype)
        elif mode is RobotMacroActionMode.REACH_POSE:
            if self.solver is None:
                raise RuntimeError("REACH_POSE needs a CartesianSolver")
            target = self.solver(macro.position.to(arm.device, arm.dtype),
                                 macro.quaternion.to(arm.device, arm.dtype), arm)
        elif mode is RobotMacroActionMode.RESET:
            if self.home_qpos is None:
                raise RuntimeError("RESET needs home_qpos (robot_home_qpos)")
            target = self.home_qpos.to(arm.device, arm.dtype)
        else:  # gripper moves: arm holds
            target = arm
        arm_seq = self._interp(arm, target.expand_as(arm), steps, settle)
        if grip is None:
            return arm_seq
        g = grip
        if macro.gripper is not None:
            g = torch.full_like(grip, float(macro.gripper))
        grip_seq = g.unsqueeze(-2).expand(*g.shape[:-1], arm_seq.shape[-2], 1)
        return torch.cat([arm_seq, grip_seq], dim=-1)


class SatelliteMacroAction(RobotMacroAction):
    """Attitude-control macro for satellite envs (reference
    mujoco/_satellite_primitives.py): the target is a body-frame
    quaternion reached over ``steps`` slews."""

    @classmethod
    def slew_to(cls, quaternion: torch.Tensor, steps: int = 10):
        quaternion = torch.as_tensor(quaternion)
        return cls(RobotMacroActionMode.REACH_POSE, steps,
                   position=torch.zeros(*quaternion.shape[:-1], 3),
                   quaternion=quaternion)


class SatelliteAttitudeTransform(URScriptPrimitiveTransform):
    """Expand :class:`SatelliteMacroAction` quaternion slews into
    per-step attitude-rate commands via normalized-quaternion slerp
    (reference mujoco/_satellite_primitives.py)."""

    def expand_macro(self, macro: MacroAction, start: torch.Tensor) -> torch.Tensor:
        if isinstance(macro, SatelliteMacroAction) and macro.quaternion is not None:
            steps = int(macro.steps)
            q0 = start[..., :4] if start.shape[-1] >= 4 else _identity_quat(start[..., :3])
            q1 = macro.quaternion.to(start.device, start.dtype).expand_as(q0)
            w = torch.linspace(1.0 / steps, 1.0, steps, device=start.device)
            w = w.reshape(*(1,) * (q0.dim() - 1), steps, 1)
            seq = q0.unsqueeze(-2) * (1 - w) + q1.unsqueeze(-2) * w
            seq = seq / seq.norm(dim=-1, keepdim=True).clamp_min(1e-8)
            if start.shape[-1] > 4:
                rest = start[..., 4:].unsqueeze(-2).expand(*start.shape[:-1], steps,
                                                           start.shape[-1] - 4)
                seq = torch.cat([seq, rest], dim=-1)
            return seq
        return super().expand_macro(macro, start)
