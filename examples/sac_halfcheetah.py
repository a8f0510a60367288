"""SAC with HBM-resident prioritized replay on HalfCheetah shapes —
milestone M3 config (BASELINE.json: "SAC ... 1M-transition
PrioritizedReplayBuffer resident in HBM").

Reference analog: pytorch/rl sota-implementations/sac/.
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from rl_amd.collectors import Collector
from rl_amd.envs.custom.synthetic import HalfCheetahVec
from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal, ValueOperator
from rl_amd.record import CSVLogger
from rl_amd.tensordict import TensorDictModule
from rl_amd.trainers import SACTrainer


def main(total_frames: int = 100_000, device=None):
    device = device or ("cuda" if torch.cuda.is_available() else "cpu")
    env = HalfCheetahVec(batch_size=[32], device=device)
    obs_dim, act_dim = env.obs_dim, env.act_dim
    net = torch.nn.Sequential(
        MLP(in_features=obs_dim, out_features=2 * act_dim, num_cells=[256, 256], device=device),
        NormalParamExtractor(),
    )
    from rl_amd.data import Bounded

    actor = ProbabilisticActor(
        TensorDictModule(net, in_keys=["observation"], out_keys=["loc", "scale"]),
        in_keys=["loc", "scale"],
        distribution_class=TanhNormal,
        return_log_prob=True,
        spec=Bounded(-1.0, 1.0, shape=(act_dim,), device=device),
    )
    qnet = ValueOperator(
        MLP(in_features=obs_dim + act_dim, out_features=1, num_cells=[256, 256], device=device),
        in_keys=["observation", "action"],
    )
    collector = Collector(env, actor, frames_per_batch=1024, total_frames=total_frames,
                          init_random_frames=5000, device=device)
    trainer = SACTrainer(
        actor=actor, qvalue=qnet, collector=collector,
        total_frames=total_frames,
        buffer_size=1_000_000, batch_size=256,
        prioritized=True, device=device,
        optim_steps_per_batch=32,
        logger=CSVLogger("sac_halfcheetah"), progress_bar=True,
    )
    trainer.train()
    print("final training reward:", trainer._log_cache.get("r_training"))


if __name__ == "__main__":
    main()
