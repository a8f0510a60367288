"""TD3 on the GPU-resident Pendulum env.

Reference analog: pytorch/rl sota-implementations/td3.  Shows the
off-policy trainer stack: Collector → replay buffer → TD3 twin-critic
updates with target-policy smoothing and delayed actor updates.
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from rl_amd.collectors import Collector
from rl_amd.envs import PendulumEnv
from rl_amd.modules import MLP, AdditiveGaussianModule, TanhModule, ValueOperator
from rl_amd.tensordict import TensorDictModule, TensorDictSequential
from rl_amd.trainers import TD3Trainer


def main(total_frames: int = 2000):
    torch.manual_seed(0)
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    env = PendulumEnv(batch_size=[8], device=device)
    actor = TensorDictSequential(
        TensorDictModule(
            MLP(in_features=3, out_features=1, num_cells=[64, 64], device=device),
            in_keys=["observation"],
            out_keys=["action"],
        ),
        TanhModule(in_keys=["action"], low=-2.0, high=2.0),
    )
    explore = TensorDictSequential(
        actor, AdditiveGaussianModule(spec=env.full_action_spec["action"], sigma_init=0.3)
    )
    qvalue = ValueOperator(
        MLP(in_features=4, out_features=1, num_cells=[64, 64], device=device),
        in_keys=["observation", "action"],
    )
    col = Collector(env, explore, frames_per_batch=256, total_frames=total_frames)
    trainer = TD3Trainer(
        actor=actor,
        qvalue=qvalue,
        collector=col,
        total_frames=total_frames,
        batch_size=128,
        optim_steps_per_batch=4,
        device=device,
    )
    trainer.train()
    trainer.shutdown()
    print("done")


if __name__ == "__main__":
    main()
