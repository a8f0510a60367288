"""Batched-env throughput: SerialEnv vs ParallelEnv (shared-memory
done-flag handshake) vs ParallelEnv (no-buffer pipe fallback).

Reference analog: pytorch/rl benchmarks/benchmark_batched_envs.py.
CPU-simulator path (the GPU-vectorized envs are the primary MI355X
path and are measured by bench.py); run anywhere.

Usage: python benchmarks/benchmark_batched_envs.py [--steps 200]
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rl_amd.envs import ParallelEnv, SerialEnv
from rl_amd.testing import ContinuousActionVecMockEnv


def make_env():
    return ContinuousActionVecMockEnv(batch_size=[4], max_steps=50)


def run_env(env, steps):
    td = env.reset()
    spec = env.full_action_spec
    t0 = time.perf_counter()
    for _ in range(steps):
        td = td.clone(False)
        for k in spec.keys(True, True):
            td.set(k, spec[k].rand())
        td, root = env.step_and_maybe_reset(td)
        td = root
    dt = time.perf_counter() - t0
    n_envs = int(torch.tensor(env.batch_size).prod()) * 4  # inner batch 4
    return steps * n_envs / dt


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--workers", type=int, nargs="+", default=[4, 8])
    args = p.parse_args()

    for n in args.workers:
        serial = SerialEnv(n, make_env)
        fps_serial = run_env(serial, args.steps)
        serial.close()

        par = ParallelEnv(n, make_env)
        t_start = time.perf_counter()
        fps_shm = run_env(par, args.steps)
        par.close()

        par_nb = ParallelEnv(n, make_env, shared_memory=False)
        fps_pipe = run_env(par_nb, args.steps)
        par_nb.close()

        print(
            json.dumps(
                {
                    "metric": f"batched_env_fps_{n}workers",
                    "workers": n,
                    "steps": args.steps,
                    "serial_fps": round(fps_serial),
                    "parallel_shm_flags_fps": round(fps_shm),
                    "parallel_pipe_fps": round(fps_pipe),
                    "shm_vs_pipe": round(fps_shm / fps_pipe, 2),
                }
            )
        )


if __name__ == "__main__":
    main()
