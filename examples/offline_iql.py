"""Offline IQL from a D4RL-layout dataset (reference
sota-implementations/iql/iql_offline.py shape).

Demonstrates the full offline pipeline: a D4RL-layout source file →
`convert_d4rl_hdf5` → memmap cache → `D4RLExperienceReplay` →
IQLLoss + target updates.  (The source here is synthesized — this image
has no network — but the path is byte-identical to a downloaded file.)

Run: python examples/offline_iql.py [--steps 200]
"""
from __future__ import annotations

import argparse
import os
import sys
import tempfile

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rl_amd.data import Bounded, D4RLExperienceReplay
from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal, ValueOperator
from rl_amd.objectives import IQLLoss, SoftUpdate
from rl_amd.tensordict import TensorDictModule

OBS, ACT = 6, 3


def synthesize_dataset(root: str, dataset_id: str, n: int = 5000) -> None:
    """Write a D4RL-layout .npz a downloader would have produced."""
    rng = np.random.default_rng(0)
    obs = rng.standard_normal((n, OBS)).astype(np.float32)
    act = np.tanh(obs[:, :ACT] + 0.1 * rng.standard_normal((n, ACT))).astype(
        np.float32
    )
    rew = (obs[:, 0] * act[:, 0]).astype(np.float32)
    term = np.zeros(n, bool)
    term[99::100] = True
    np.savez(
        os.path.join(root, f"{dataset_id}.npz"),
        observations=obs,
        actions=act,
        rewards=rew,
        terminals=term,
    )


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--batch", type=int, default=256)
    args = p.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    root = tempfile.mkdtemp(prefix="rl_amd_d4rl_")
    synthesize_dataset(root, "halfcheetah-medium-v2")
    rb = D4RLExperienceReplay(
        "halfcheetah-medium-v2", root=root, batch_size=args.batch, device=device
    )
    print(f"dataset: {len(rb)} transitions (memmap cache under {root})")

    actor = ProbabilisticActor(
        TensorDictModule(
            torch.nn.Sequential(
                MLP(in_features=OBS, out_features=2 * ACT, num_cells=[64, 64], device=device),
                NormalParamExtractor(),
            ),
            in_keys=["observation"],
            out_keys=["loc", "scale"],
        ),
        in_keys=["loc", "scale"],
        distribution_class=TanhNormal,
        return_log_prob=True,
        spec=Bounded(-1.0, 1.0, shape=(ACT,), device=device),
    )
    qvalue = ValueOperator(
        MLP(in_features=OBS + ACT, out_features=1, num_cells=[64, 64], device=device),
        in_keys=["observation", "action"],
    )
    value = ValueOperator(
        MLP(in_features=OBS, out_features=1, num_cells=[64, 64], device=device),
        in_keys=["observation"],
    )
    loss = IQLLoss(actor, qvalue, value_network=value).to(device)
    loss.make_value_estimator()
    updater = SoftUpdate(loss, tau=0.005)
    optim = torch.optim.Adam(loss.parameters(), lr=3e-4)

    for step in range(args.steps):
        batch = rb.sample()
        out = loss(batch)
        total = sum(v for k, v in out.items() if str(k).startswith("loss_"))
        optim.zero_grad()
        total.backward()
        optim.step()
        updater.step()
        if step % 50 == 0:
            print(f"step {step}: loss {float(total):.4f}")
    print("done")


if __name__ == "__main__":
    main()
