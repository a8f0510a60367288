"""Offline Decision Transformer (reference sota-implementations/
decision_transformer/dt.py shape): trajectory slices from a replay
buffer with return-to-go conditioning, DTLoss on the GPT-2-style model.

Run: python examples/decision_transformer.py [--steps 100]
"""
from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rl_amd.data import LazyTensorStorage, SliceSampler, TensorDictReplayBuffer
from rl_amd.modules import DecisionTransformer
from rl_amd.objectives import DTLoss
from rl_amd.objectives.value.functional import reward2go
from rl_amd.tensordict import TensorDict, TensorDictModule

OBS, ACT, CTX = 6, 3, 8


def build_offline_buffer(n_traj=50, T=20, device="cpu"):
    """Synthetic trajectories with return-to-go precomputed."""
    torch.manual_seed(0)
    rows = []
    for _ in range(n_traj):
        obs = torch.randn(T, OBS)
        act = torch.tanh(obs[:, :ACT] + 0.1 * torch.randn(T, ACT))
        rew = (obs[:, :1] * act[:, :1])
        done = torch.zeros(T, 1, dtype=torch.bool)
        done[-1] = True
        rtg = reward2go(rew, done, gamma=1.0)
        rows.append(
            TensorDict(
                {
                    "observation": obs,
                    "action": act,
                    "return_to_go": rtg,
                    "next": {"reward": rew, "done": done, "terminated": done.clone()},
                },
                batch_size=[T],
            )
        )
    from rl_amd.tensordict import cat as td_cat

    data = td_cat(rows, 0)
    rb = TensorDictReplayBuffer(
        storage=LazyTensorStorage(len(data), device=device),
        sampler=SliceSampler(slice_len=CTX),
        batch_size=4 * CTX,
    )
    rb.extend(data.to(device))
    return rb


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=100)
    args = p.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    rb = build_offline_buffer(device=device)

    class DTActor(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.dt = DecisionTransformer(
                state_dim=OBS, action_dim=ACT,
                config={"n_embd": 64, "n_layer": 2, "n_head": 2},
                device=device,
            )
            self.head = torch.nn.Linear(self.dt.hidden_dim, ACT, device=device)

        def forward(self, observation, action, return_to_go):
            h = self.dt(observation, action, return_to_go)
            return torch.tanh(self.head(h))

    model = TensorDictModule(
        DTActor().to(device),
        in_keys=["observation", "action", "return_to_go"],
        out_keys=["action_pred"],
    )
    loss = DTLoss(model)
    loss.set_keys(action_pred="action_pred")
    optim = torch.optim.Adam(loss.parameters(), lr=1e-3)

    for step in range(args.steps):
        batch = rb.sample().reshape(4, CTX)  # [B, context]
        out = loss(batch)
        total = out.get("loss")
        optim.zero_grad()
        total.backward()
        optim.step()
        if step % 25 == 0:
            print(f"step {step}: loss {float(total):.4f}")
    print("done")


if __name__ == "__main__":
    main()
