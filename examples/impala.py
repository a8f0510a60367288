"""IMPALA-style async actor-learner with V-trace — milestone M4 config
(BASELINE.json: "IMPALA ... MultiSyncDataCollector sharded across
8xMI355X (RCCL all-gather rollouts + V-trace HIP kernel)").

Single-process form here; the 8-GPU form launches via torch.distributed.run
with DistributedCollector (see rl_amd/collectors/distributed.py).

Reference analog: pytorch/rl sota-implementations/impala/.
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from rl_amd.collectors import Collector
from rl_amd.envs.custom.synthetic import SyntheticMuJoCoEnv
from rl_amd.modules import MLP, ProbabilisticActor, ValueOperator, OneHotCategorical
from rl_amd.objectives import A2CLoss, VTrace
from rl_amd.record import CSVLogger
from rl_amd.tensordict import TensorDictModule


class DiscreteSynthetic(SyntheticMuJoCoEnv):
    """Pong-shaped discrete control over the synthetic dynamics."""

    OBS_DIM = 64
    ACT_DIM = 6

    def __init__(self, **kwargs):
        super().__init__(**kwargs)
        from rl_amd.data import OneHot

        bs = self.batch_size
        self.action_spec = OneHot(6, shape=(*bs, 6), device=self.device)

    def _step(self, td):
        a = td.get("action")
        cont = (a.to(self.dtype) - 0.5) * 0.2
        td2 = td.clone(False)
        td2.set("action", torch.nn.functional.pad(cont, (0, 0))[..., : self.act_dim])
        return super()._step(td2)


def main(total_frames: int = 100_000, n_envs: int = 64, device=None):
    device = device or ("cuda" if torch.cuda.is_available() else "cpu")
    env = DiscreteSynthetic(batch_size=[n_envs], device=device, max_steps=256)
    actor = ProbabilisticActor(
        TensorDictModule(
            MLP(in_features=64, out_features=6, num_cells=[256, 256], device=device),
            in_keys=["observation"], out_keys=["logits"],
        ),
        in_keys=["logits"],
        distribution_class=OneHotCategorical,
        return_log_prob=True,
    )
    critic = ValueOperator(
        MLP(in_features=64, out_features=1, num_cells=[256, 256], device=device),
        in_keys=["observation"],
    )
    vtrace = VTrace(gamma=0.99, value_network=critic, actor_network=actor)
    loss = A2CLoss(actor, critic, entropy_coeff=0.01)
    loss.value_estimator = vtrace
    optim = torch.optim.RMSprop(loss.parameters(), lr=6e-4)
    collector = Collector(env, actor, frames_per_batch=n_envs * 32,
                          total_frames=total_frames, device=device)
    logger = CSVLogger("impala")
    for i, batch in enumerate(collector):
        with torch.no_grad():
            vtrace(batch)
        out = loss(batch.reshape(-1))
        total = out.get("loss_objective") + out.get("loss_critic") + out.get("loss_entropy")
        optim.zero_grad(set_to_none=True)
        total.backward()
        torch.nn.utils.clip_grad_norm_(loss.parameters(), 40.0)
        optim.step()
        if i % 10 == 0:
            r = batch.get(("next", "reward")).mean().item()
            logger.log_scalar("reward", r, step=i)
            print(f"iter {i}: mean reward {r:.4f}")
    collector.shutdown()


if __name__ == "__main__":
    main(total_frames=20_000)
