"""DQN on a CartPole-shaped env — milestone M1 config
(BASELINE.json: "CartPole-v1 DQN, SyncDataCollector + TensorDictReplayBuffer").

Reference analog: pytorch/rl sota-implementations/dqn/dqn_cartpole.py.
Uses gymnasium's CartPole-v1 when installed, else the bundled mock.
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from rl_amd.collectors import Collector
from rl_amd.envs.transforms import StepCounter, TransformedEnv
from rl_amd.modules import MLP, EGreedyModule, QValueActor
from rl_amd.record import CSVLogger
from rl_amd.tensordict import TensorDictSequential
from rl_amd.trainers import DQNTrainer


def make_env():
    try:
        from rl_amd.envs.libs.gym import GymEnv

        return TransformedEnv(GymEnv("CartPole-v1"), StepCounter())
    except ImportError:
        from rl_amd.testing import DiscreteActionVecMockEnv

        return TransformedEnv(DiscreteActionVecMockEnv(max_steps=200), StepCounter())


def main(total_frames: int = 20_000):
    env = make_env()
    obs_dim = env.observation_spec["observation"].shape[-1]
    n_act = env.action_spec.n
    qnet = QValueActor(
        MLP(in_features=obs_dim, out_features=n_act, num_cells=[128, 128]),
        spec=env.action_spec,
    )
    eg = EGreedyModule(spec=env.action_spec, eps_init=1.0, eps_end=0.05,
                       annealing_num_steps=total_frames // 2)
    policy = TensorDictSequential(qnet, eg)
    collector = Collector(env, policy, frames_per_batch=200, total_frames=total_frames,
                          init_random_frames=1000)
    trainer = DQNTrainer(
        value_network=qnet,
        collector=collector,
        total_frames=total_frames,
        buffer_size=50_000,
        batch_size=128,
        optim_steps_per_batch=8,
        logger=CSVLogger("dqn_cartpole"),
        progress_bar=True,
    )
    trainer.train()
    print("final training reward:", trainer._log_cache.get("r_training"))


if __name__ == "__main__":
    main()
