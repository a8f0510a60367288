"""MuJoCo-backed custom envs — gated: `mujoco` is not installed in
this offline image.

Reference: pytorch/rl torchrl/envs/custom/mujoco/ (MujocoEnv base,
ant.py AntEnv, hopper.py HopperEnv, humanoid.py HumanoidEnv,
walker2d.py Walker2dEnv, cube_bowl.py CubeBowlEnv, satellite.py
SatelliteEnv) and mjlab wrappers (MJLabEnv, MJLabWrapper).  The
device-resident locomotion workloads these back are covered offline
by :class:`~rl_amd.envs.custom.synthetic.SyntheticMuJoCoEnv`
(HalfCheetah/Hopper/Walker2d/Ant/Humanoid-shaped, pure torch), which
is what bench.py and the kernels run on.
"""
from __future__ import annotations

import importlib.util

from ..common import EnvBase

__all__ = [
    "MujocoEnv",
    "AntEnv",
    "HopperEnv",
    "HumanoidEnv",
    "Walker2dEnv",
    "CubeBowlEnv",
    "SatelliteEnv",
]


class MujocoEnv(EnvBase):
    """Base for native-MuJoCo envs (reference custom/mujoco/): steps the
    simulator directly (no gym layer) and exposes batched specs."""

    _lib = "mujoco"

    def __init__(self, *args, **kwargs):
        if importlib.util.find_spec(self._lib) is None:
            raise ImportError(
                f"{type(self).__name__} requires the `{self._lib}` package, which is "
                "not installed in this image. Use SyntheticMuJoCoEnv for the "
                "same observation/action shapes offline."
            )
        raise NotImplementedError(f"{type(self).__name__}: simulator backend scaffolding")


class AntEnv(MujocoEnv):
    """Ant locomotion (reference custom/mujoco/ant.py)."""


class HopperEnv(MujocoEnv):
    """Hopper locomotion (reference custom/mujoco/hopper.py)."""


class HumanoidEnv(MujocoEnv):
    """Humanoid locomotion (reference custom/mujoco/humanoid.py)."""


class Walker2dEnv(MujocoEnv):
    """Walker2d locomotion (reference custom/mujoco/walker2d.py)."""


class CubeBowlEnv(MujocoEnv):
    """Cube-in-bowl manipulation (reference custom/mujoco/cube_bowl.py)."""


class SatelliteEnv(MujocoEnv):
    """Satellite attitude control (reference custom/mujoco/satellite.py);
    pairs with SatelliteMacroAction / SatelliteAttitudeTransform."""
