"""Checkpoint → rollout rendering (reference: pytorch/rl torchrl/render/).

``save_render_checkpoint`` stores policy weights + env/policy factory
configs alongside a normal Checkpoint; ``main`` reloads and rolls out.
"""
from __future__ import annotations

import argparse
import json
import os
from typing import Callable, Optional

import torch

from ..checkpoint import Checkpoint
from ..envs.utils import ExplorationType, set_exploration_type
from ..tensordict import TensorDictBase
from ..trainers.configs import instantiate, load_config

__all__ = ["save_render_checkpoint", "render_rollout", "main"]


def save_render_checkpoint(
    path: str,
    policy: torch.nn.Module,
    env_config: dict,
    policy_config: Optional[dict] = None,
) -> str:
    """Persist everything needed to re-render: weights + build configs
    (reference render/checkpoint.py ``save_render_checkpoint``)."""
    ckpt = Checkpoint()
    ckpt.register(policy, "policy")
    out = ckpt.save(path)
    with open(os.path.join(path, "render_config.json"), "w") as f:
        json.dump({"env": env_config, "policy": policy_config}, f, indent=2, default=str)
    return out


def render_rollout(
    env,
    policy,
    steps: int = 200,
    deterministic: bool = True,
    out_path: Optional[str] = None,
    video_path: Optional[str] = None,
    pixel_key: str = "pixels",
) -> TensorDictBase:
    etype = (
        ExplorationType.DETERMINISTIC if deterministic else ExplorationType.RANDOM
    )
    with set_exploration_type(etype), torch.no_grad():
        rollout = env.rollout(steps, policy=policy, break_when_any_done=True)
    if out_path is not None:
        torch.save(rollout, out_path)
    if video_path is not None:
        frames = rollout.get(pixel_key, None)
        if frames is None:
            frames = rollout.get(("next", pixel_key), None)
        if frames is None:
            raise KeyError(
                f"rollout has no {pixel_key!r} frames — add a pixel transform "
                "or PixelRenderTransform to the env"
            )
        from .video import write_gif

        write_gif(frames.reshape(-1, *frames.shape[-3:]), video_path)
    return rollout


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(
        prog="rlrender", description="replay an rl_amd checkpoint"
    )
    parser.add_argument("checkpoint", help="checkpoint directory")
    parser.add_argument("--steps", type=int, default=200)
    parser.add_argument("--episodes", type=int, default=1)
    parser.add_argument("--out", type=str, default=None)
    parser.add_argument("--video", type=str, default=None,
                        help="write an animated GIF of the pixel frames")
    parser.add_argument("--stochastic", action="store_true")
    args = parser.parse_args(argv)

    cfg_path = os.path.join(args.checkpoint, "render_config.json")
    if not os.path.exists(cfg_path):
        raise FileNotFoundError(
            f"{cfg_path} missing — save with save_render_checkpoint"
        )
    with open(cfg_path) as f:
        cfg = json.load(f)
    env = instantiate(cfg["env"])
    policy = instantiate(cfg["policy"]) if cfg.get("policy") else None
    if policy is not None:
        Checkpoint().register(policy, "policy").load(args.checkpoint)
    total = 0.0
    for ep in range(args.episodes):
        rollout = render_rollout(
            env,
            policy,
            steps=args.steps,
            deterministic=not args.stochastic,
            out_path=(
                f"{args.out}_ep{ep}.pt" if args.out and args.episodes > 1 else args.out
            ),
            video_path=(
                f"{args.video}_ep{ep}.gif" if args.video and args.episodes > 1 else args.video
            ),
        )
        r = float(rollout.get(("next", "reward")).sum())
        total += r
        print(f"episode {ep}: steps={rollout.batch_size[-1]} return={r:.3f}")
    print(f"mean return: {total / args.episodes:.3f}")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
