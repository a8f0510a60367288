"""Correctness check: full-step-graphed PPO must actually LEARN.

Runs bench.py's captured training structure on the Pendulum env (dense
reward, solvable) and asserts the mean reward improves — guarding
against silent no-op replays (stale weights, broken RNG, etc.).
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from rl_amd.collectors import GraphedRollout
from rl_amd.envs import PendulumEnv
from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal, ValueOperator
from rl_amd.objectives import ClipPPOLoss
from rl_amd.objectives.value.advantages import GAE
from rl_amd.tensordict import TensorDictModule


def main(iters: int = 60, envs: int = 1024, T: int = 32):
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    torch.manual_seed(0)
    env = PendulumEnv(batch_size=[envs], device=device)
    net = torch.nn.Sequential(
        MLP(in_features=3, out_features=2, num_cells=[64, 64], device=device),
        NormalParamExtractor(),
    )
    actor = ProbabilisticActor(
        TensorDictModule(net, in_keys=["observation"], out_keys=["loc", "scale"]),
        in_keys=["loc", "scale"],
        distribution_class=TanhNormal,
        distribution_kwargs={"low": -2.0, "high": 2.0},
        return_log_prob=True,
    )
    critic = ValueOperator(
        MLP(in_features=3, out_features=1, num_cells=[64, 64], device=device),
        in_keys=["observation"],
    )
    loss_mod = ClipPPOLoss(actor, critic, critic_coeff=0.5, normalize_advantage=True)
    gae = GAE(gamma=0.99, lmbda=0.95, value_network=critic)
    optim = torch.optim.Adam(loss_mod.parameters(), lr=1e-3)
    gr = GraphedRollout(env, actor, horizon=T).initialize()
    print("captured:", gr.captured)
    if device.type == "cuda":
        assert gr.captured, "rollout graph capture failed on GPU — the check must exercise the captured path"
    rewards = []
    for i in range(iters):
        batch = gr.collect()
        with torch.no_grad():
            gae(batch)
        flat = batch.reshape(-1)
        n = flat.batch_size[0]
        perm = torch.randperm(n, device=device)
        for k in range(4):
            idx = perm[k * n // 4 : (k + 1) * n // 4]
            sub = flat[idx]
            out = loss_mod(sub)
            total = out.get("loss_objective") + out.get("loss_critic") + out.get("loss_entropy")
            optim.zero_grad(set_to_none=True)
            total.backward()
            torch.nn.utils.clip_grad_norm_(loss_mod.parameters(), 1.0)
            optim.step()
        r = batch.get(("next", "reward")).mean().item()
        rewards.append(r)
        if i % 10 == 0:
            print(f"iter {i}: mean reward {r:.4f}")
    w = max(1, len(rewards) // 6)
    early = sum(rewards[:w]) / w
    late = sum(rewards[-w:]) / w
    print(f"early {early:.4f} -> late {late:.4f}")
    assert late > early + 0.3, "graphed PPO failed to improve Pendulum reward"
    print("LEARNING CHECK PASSED")


if __name__ == "__main__":
    import sys
    kw = {}
    if len(sys.argv) > 1:
        kw = dict(iters=int(sys.argv[1]), envs=int(sys.argv[2]), T=int(sys.argv[3]))
    main(**kw)
