"""MAPPO on a dense multi-agent counting env (reference
sota-implementations/multiagent/mappo_ippo.py shape).

Centralized critic over the joint observation, per-agent policies with
shared parameters (MultiAgentMLP), MAPPOLoss + GAE.

Run: python examples/mappo_multiagent.py [--iters 20]
"""
from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rl_amd.collectors import Collector
from rl_amd.modules import MultiAgentMLP, NormalParamExtractor, ProbabilisticActor, TanhNormal, ValueOperator
from rl_amd.objectives import MAPPOLoss
from rl_amd.objectives.value.advantages import MultiAgentGAE
from rl_amd.tensordict import TensorDictModule
from rl_amd.testing import MultiAgentCountingEnv


class ContinuousMACounting(MultiAgentCountingEnv):
    """Continuous-action flavor: each agent emits a scalar in [-1, 1];
    positive values count (the loss needs log-probs of the SAME key the
    env consumes)."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        from rl_amd.data.tensor_specs import Bounded, Composite

        bs = self.batch_size
        self.full_action_spec = Composite(
            {("agents", "action"): Bounded(
                low=-1.0, high=1.0, shape=(*bs, self.n_agents, 1),
                device=self.device,
            )},
            shape=bs,
            device=self.device,
        )

    def _step(self, tensordict):
        cont = tensordict.get(("agents", "action"))
        td = tensordict.clone(False)
        td.set(("agents", "action"), cont > 0)
        out = super()._step(td)
        return out


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--agents", type=int, default=3)
    args = p.parse_args()

    n = args.agents
    env = ContinuousMACounting(n_agents=n, max_steps=10, batch_size=[8])

    policy_net = MultiAgentMLP(
        n_agent_inputs=3,
        n_agent_outputs=2,  # loc+scale for 1-dim action per agent
        n_agents=n,
        centralised=False,
        share_params=True,
        num_cells=[32],
    )
    policy = ProbabilisticActor(
        TensorDictModule(
            torch.nn.Sequential(policy_net, NormalParamExtractor()),
            in_keys=[("agents", "observation")],
            out_keys=[("agents", "loc"), ("agents", "scale")],
        ),
        in_keys=[("agents", "loc"), ("agents", "scale")],
        out_keys=[("agents", "action")],
        distribution_class=TanhNormal,
        return_log_prob=True,
        log_prob_key=("agents", "sample_log_prob"),
    )

    critic = ValueOperator(
        MultiAgentMLP(
            n_agent_inputs=3,
            n_agent_outputs=1,
            n_agents=n,
            centralised=True,  # MAPPO: critic sees every agent
            share_params=True,
            num_cells=[32],
        ),
        in_keys=[("agents", "observation")],
        out_keys=[("agents", "state_value")],
    )

    loss = MAPPOLoss(policy, critic)
    loss.set_keys(done=("next", "done"), terminated=("next", "terminated"))
    gae = MultiAgentGAE(gamma=0.9, lmbda=0.9, value_network=critic)
    gae.set_keys(
        value=("agents", "state_value"),
        advantage=("agents", "advantage"),
        value_target=("agents", "value_target"),
        reward=("next", "agents", "reward"),
    )
    optim = torch.optim.Adam(loss.parameters(), lr=3e-4)
    col = Collector(env, policy, frames_per_batch=64, total_frames=64 * args.iters)

    for i, batch in enumerate(col):
        with torch.no_grad():
            gae(batch)
        out = loss(batch.reshape(-1))
        total = sum(v for k, v in out.items() if str(k).startswith("loss_"))
        optim.zero_grad()
        total.backward()
        optim.step()
        if i % 5 == 0:
            r = batch.get(("next", "agents", "reward")).float().mean().item()
            print(f"iter {i}: loss {float(total):.4f} mean agent reward {r:.3f}")
    col.shutdown()
    print("done")


if __name__ == "__main__":
    main()
