"""Render CLI — replay checkpoints to rollouts/videos.

Reference: pytorch/rl torchrl/render/ (cli.py ``rlrender``, rollout.py,
checkpoint.py): load a saved checkpoint, rebuild the policy, roll the env
and dump trajectories (and pixel videos when the env provides them).
Console entry: ``python -m rl_amd.render <checkpoint> [--steps N]``.
"""
from .cli import main, render_rollout, save_render_checkpoint
from .video import frames_to_uint8, write_gif

__all__ = ["main", "render_rollout", "save_render_checkpoint", "write_gif", "frames_to_uint8"]
