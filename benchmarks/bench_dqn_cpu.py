"""CartPole DQN CPU plumbing benchmark (BASELINE.json config 1:
"CartPole-v1 DQN, SyncDataCollector + TensorDictReplayBuffer on CPU").

Measures the collector → replay buffer → DQN update loop throughput on
CPU — the plumbing path, not a GPU benchmark.  One step = collect one
frames_per_batch batch + `utd` sampled Q-updates.  Metric: env frames/s.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rl_amd.collectors import Collector
from rl_amd.data import LazyTensorStorage, TensorDictReplayBuffer
from rl_amd.envs.transforms import StepCounter, TransformedEnv
from rl_amd.modules import MLP, EGreedyModule, QValueActor
from rl_amd.objectives import DQNLoss, SoftUpdate
from rl_amd.tensordict import TensorDictSequential
from rl_amd.testing import DiscreteActionVecMockEnv


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--envs", type=int, default=16)
    p.add_argument("--frames-per-batch", type=int, default=256)
    p.add_argument("--utd", type=int, default=4)
    args = p.parse_args()
    torch.manual_seed(0)

    env = TransformedEnv(
        DiscreteActionVecMockEnv(batch_size=[args.envs], max_steps=200), StepCounter()
    )
    n_obs = env.full_observation_spec["observation"].shape[-1]
    n_act = env.full_action_spec["action"].shape[-1]
    value_net = MLP(in_features=n_obs, out_features=n_act, num_cells=[120, 84])
    actor = QValueActor(value_net, spec=env.full_action_spec["action"])
    greedy = EGreedyModule(
        spec=env.full_action_spec["action"], annealing_num_steps=10_000
    )
    policy = TensorDictSequential(actor, greedy)
    col = Collector(env, policy, frames_per_batch=args.frames_per_batch, total_frames=-1)
    rb = TensorDictReplayBuffer(storage=LazyTensorStorage(100_000), batch_size=128)
    loss_mod = DQNLoss(actor, delay_value=True)
    loss_mod.make_value_estimator()
    updater = SoftUpdate(loss_mod, tau=0.02)
    optim = torch.optim.Adam(loss_mod.parameters(), lr=2e-3)
    it = iter(col.iterator())

    def one_step():
        batch = next(it)
        rb.extend(batch.reshape(-1))
        for _ in range(args.utd):
            sample = rb.sample()
            out = loss_mod(sample)
            optim.zero_grad(set_to_none=True)
            out.get("loss").backward()
            optim.step()
            updater.step()
        greedy.step(batch.numel())

    for _ in range(args.warmup):
        one_step()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    dt = time.perf_counter() - t0
    frames = args.steps * args.frames_per_batch
    print(
        json.dumps(
            {
                "metric": "dqn_cartpole_cpu_frames_per_sec",
                "value": frames / dt,
                "unit": "frames/s",
                "n_gpus": 0,
                "steps": args.steps,
                "warmup": args.warmup,
                "ms_per_step": dt / args.steps * 1000,
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": None,
                "dtype": "fp32",
                "data": "synthetic",
                "config": {
                    "model": "dqn_mlp120x84",
                    "global_batch": 128,
                    "frames_per_batch": args.frames_per_batch,
                    "utd": args.utd,
                    "parallelism": "cpu",
                },
            }
        )
    )
    col.shutdown()


if __name__ == "__main__":
    main()
