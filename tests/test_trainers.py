"""Trainer orchestration tests: hooks, algorithm trainers end-to-end,
checkpoint/resume."""
import os

import pytest
import torch

from rl_amd.checkpoint import Checkpoint, CheckpointRotation, GlobalRNGState
from rl_amd.collectors import Collector
from rl_amd.data import LazyTensorStorage, TensorDictReplayBuffer
from rl_amd.modules import (
    MLP,
    EGreedyModule,
    NormalParamExtractor,
    ProbabilisticActor,
    QValueActor,
    TanhNormal,
    ValueOperator,
)
from rl_amd.record import CSVLogger
from rl_amd.tensordict import TensorDict, TensorDictModule, TensorDictSequential
from rl_amd.testing import ContinuousActionVecMockEnv, DiscreteActionVecMockEnv
from rl_amd.trainers import (
    DQNTrainer,
    PPOTrainer,
    SACTrainer,
    Trainer,
)


def make_cont_actor_critic(obs_dim=7, act_dim=5):
    net = torch.nn.Sequential(
        MLP(in_features=obs_dim, out_features=2 * act_dim, num_cells=[32]),
        NormalParamExtractor(),
    )
    from rl_amd.data import Bounded

    actor = ProbabilisticActor(
        TensorDictModule(net, in_keys=["observation"], out_keys=["loc", "scale"]),
        in_keys=["loc", "scale"],
        distribution_class=TanhNormal,
        return_log_prob=True,
        spec=Bounded(-1.0, 1.0, shape=(act_dim,)),
    )
    critic = ValueOperator(
        MLP(in_features=obs_dim, out_features=1, num_cells=[32]),
        in_keys=["observation"],
    )
    return actor, critic


def make_q_actor(obs_dim=4, n_act=2):
    env = DiscreteActionVecMockEnv(batch_size=[4])
    return QValueActor(
        MLP(in_features=obs_dim, out_features=n_act, num_cells=[32]),
        spec=env.action_spec,
    )


class TestTrainers:
    def test_dqn_end_to_end(self, tmp_path):
        env = DiscreteActionVecMockEnv(batch_size=[4], max_steps=20)
        qnet = make_q_actor()
        eg = EGreedyModule(spec=env.action_spec, eps_init=0.5)
        policy = TensorDictSequential(qnet, eg)
        col = Collector(env, policy, frames_per_batch=64, total_frames=256)
        tr = DQNTrainer(
            value_network=qnet,
            collector=col,
            total_frames=256,
            batch_size=32,
            buffer_size=500,
            progress_bar=False,
            logger=CSVLogger("dqn", log_dir=str(tmp_path)),
        )
        tr.train()
        assert tr.collected_frames == 256
        assert tr._optim_count > 0
        assert "loss" in tr._log_cache

    def test_ppo_end_to_end(self):
        env = ContinuousActionVecMockEnv(batch_size=[4], max_steps=50)
        actor, critic = make_cont_actor_critic()
        col = Collector(env, actor, frames_per_batch=128, total_frames=256)
        tr = PPOTrainer(
            actor=actor,
            critic=critic,
            collector=col,
            total_frames=256,
            minibatch_size=64,
            num_epochs=2,
            progress_bar=False,
        )
        tr.train()
        assert tr.collected_frames == 256
        assert any(k.startswith("loss") for k in tr._log_cache)

    def test_sac_end_to_end(self):
        env = ContinuousActionVecMockEnv(batch_size=[2], max_steps=50)
        actor, _ = make_cont_actor_critic()
        qnet = ValueOperator(
            MLP(in_features=7 + 5, out_features=1, num_cells=[32]),
            in_keys=["observation", "action"],
        )
        col = Collector(env, actor, frames_per_batch=32, total_frames=64)
        tr = SACTrainer(
            actor=actor,
            qvalue=qnet,
            collector=col,
            total_frames=64,
            batch_size=16,
            buffer_size=500,
            optim_steps_per_batch=2,
            progress_bar=False,
        )
        tr.train()
        assert tr.collected_frames == 64

    def test_trainer_checkpoint(self, tmp_path):
        env = DiscreteActionVecMockEnv(batch_size=[4], max_steps=20)
        qnet = make_q_actor()
        col = Collector(env, qnet, frames_per_batch=64, total_frames=128)
        save_file = str(tmp_path / "trainer_ckpt")
        tr = DQNTrainer(
            value_network=qnet,
            collector=col,
            total_frames=128,
            batch_size=32,
            buffer_size=500,
            progress_bar=False,
            save_trainer_file=save_file,
        )
        tr.train()
        tr.save_trainer(force_save=True)
        assert os.path.exists(os.path.join(save_file, "manifest.json"))
        # resume into a fresh trainer
        env2 = DiscreteActionVecMockEnv(batch_size=[4], max_steps=20)
        qnet2 = make_q_actor()
        col2 = Collector(env2, qnet2, frames_per_batch=64, total_frames=128)
        tr2 = DQNTrainer(
            value_network=qnet2,
            collector=col2,
            total_frames=128,
            batch_size=32,
            buffer_size=500,
            progress_bar=False,
            save_trainer_file=save_file,
        )
        tr2.load_from_file(save_file)
        assert tr2.collected_frames == 128


class TestCheckpoint:
    def test_manifest_format(self, tmp_path):
        import json

        ckpt = Checkpoint()
        ckpt.register(torch.nn.Linear(3, 2), "model")
        ckpt.register(GlobalRNGState(), "rng")
        path = str(tmp_path / "ck")
        ckpt.save(path)
        with open(os.path.join(path, "manifest.json")) as f:
            manifest = json.load(f)
        assert manifest["format"] == "torchrl.checkpoint"
        assert manifest["version"] == 1
        assert "model" in manifest["components"]

    def test_roundtrip_module(self, tmp_path):
        m1 = torch.nn.Linear(3, 2)
        ckpt = Checkpoint().register(m1, "model")
        path = str(tmp_path / "ck")
        ckpt.save(path)
        m2 = torch.nn.Linear(3, 2)
        Checkpoint().register(m2, "model").load(path)
        assert torch.allclose(m1.weight, m2.weight)

    def test_zip_archive(self, tmp_path):
        m1 = torch.nn.Linear(3, 2)
        ckpt = Checkpoint().register(m1, "model")
        path = str(tmp_path / "ck.zip")
        ckpt.save(path)
        assert os.path.exists(path)
        m2 = torch.nn.Linear(3, 2)
        Checkpoint().register(m2, "model").load(path)
        assert torch.allclose(m1.weight, m2.weight)

    def test_rng_state_roundtrip(self, tmp_path):
        rng = GlobalRNGState()
        ckpt = Checkpoint().register(rng, "rng")
        path = str(tmp_path / "ck")
        torch.manual_seed(0)
        ckpt.save(path)
        a = torch.randn(3)
        Checkpoint().register(GlobalRNGState(), "rng").load(path)
        b = torch.randn(3)
        assert torch.allclose(a, b)

    def test_replay_buffer_in_checkpoint(self, tmp_path):
        rb = TensorDictReplayBuffer(storage=LazyTensorStorage(50), batch_size=4)
        rb.extend(TensorDict({"x": torch.randn(10, 2)}, batch_size=[10]))
        ckpt = Checkpoint().register(rb, "buffer")
        path = str(tmp_path / "ck")
        ckpt.save(path)
        rb2 = TensorDictReplayBuffer(storage=LazyTensorStorage(50), batch_size=4)
        Checkpoint().register(rb2, "buffer").load(path)
        assert len(rb2) == 10

    def test_rotation(self, tmp_path):
        m = torch.nn.Linear(2, 2)
        ckpt = Checkpoint().register(m, "model")
        rot = CheckpointRotation(ckpt, str(tmp_path / "rots"), keep_last=2)
        for i, metric in enumerate([1.0, 3.0, 2.0, 0.5]):
            rot.step(i, metric=metric)
        dirs = sorted(os.listdir(str(tmp_path / "rots")))
        assert "ckpt_best" in dirs
        assert len([d for d in dirs if d != "ckpt_best"]) == 2

    def test_rotation_best_metric(self, tmp_path):
        m = torch.nn.Linear(2, 2)
        ckpt = Checkpoint().register(m, "model")
        rot = CheckpointRotation(ckpt, str(tmp_path / "rots"), keep_last=1)
        with torch.no_grad():
            m.weight.fill_(1.0)
        rot.step(0, metric=5.0)
        with torch.no_grad():
            m.weight.fill_(2.0)
        rot.step(1, metric=1.0)
        # best should still hold weight=1
        m2 = torch.nn.Linear(2, 2)
        Checkpoint().register(m2, "model").load(rot.best_path)
        assert (m2.weight == 1.0).all()


class TestLoggers:
    def test_csv_logger(self, tmp_path):
        log = CSVLogger("exp1", log_dir=str(tmp_path))
        log.log_scalar("reward", 1.5, step=10)
        log.log_scalar("reward", 2.5, step=20)
        log.log_hparams({"lr": 0.001})
        path = os.path.join(log.experiment_dir, "scalars", "reward.csv")
        with open(path) as f:
            lines = f.read().strip().split("\n")
        assert len(lines) == 2
        assert lines[0] == "10,1.5"


class TestMoreAlgorithmTrainers:
    def _cont_env(self):
        from rl_amd.testing import ContinuousActionVecMockEnv

        return ContinuousActionVecMockEnv(batch_size=[2], max_steps=10)

    def _actor_critic_q(self):
        from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal, ValueOperator
        from rl_amd.tensordict import TensorDictModule

        net = torch.nn.Sequential(
            MLP(in_features=7, out_features=2 * 5, num_cells=[16]),
            NormalParamExtractor(),
        )
        from rl_amd.data import Bounded

        actor = ProbabilisticActor(
            TensorDictModule(net, in_keys=["observation"], out_keys=["loc", "scale"]),
            in_keys=["loc", "scale"],
            distribution_class=TanhNormal,
            return_log_prob=True,
            spec=Bounded(-1.0, 1.0, shape=(5,)),
        )
        qvalue = ValueOperator(
            MLP(in_features=12, out_features=1, num_cells=[16]),
            in_keys=["observation", "action"],
        )
        value = ValueOperator(MLP(in_features=7, out_features=1, num_cells=[16]), in_keys=["observation"])
        return actor, qvalue, value

    def test_reinforce_trainer(self):
        from rl_amd.collectors import Collector
        from rl_amd.trainers import ReinforceTrainer

        actor, _, value = self._actor_critic_q()
        col = Collector(self._cont_env(), actor, frames_per_batch=20, total_frames=40)
        t = ReinforceTrainer(actor=actor, critic=value, collector=col, total_frames=40)
        t.train()
        t.shutdown()

    def test_iql_trainer(self):
        from rl_amd.collectors import Collector
        from rl_amd.trainers import IQLTrainer

        actor, qvalue, value = self._actor_critic_q()
        col = Collector(self._cont_env(), actor, frames_per_batch=20, total_frames=40)
        t = IQLTrainer(actor=actor, qvalue=qvalue, value=value, collector=col,
                       total_frames=40, batch_size=16, optim_steps_per_batch=1)
        t.train()
        t.shutdown()

    def test_cql_trainer(self):
        from rl_amd.collectors import Collector
        from rl_amd.trainers import CQLTrainer

        actor, qvalue, _ = self._actor_critic_q()
        col = Collector(self._cont_env(), actor, frames_per_batch=20, total_frames=40)
        t = CQLTrainer(actor=actor, qvalue=qvalue, collector=col, total_frames=40,
                       batch_size=16, optim_steps_per_batch=1)
        t.train()
        t.shutdown()

    def test_offline_to_online_trainer(self):
        from rl_amd.collectors import Collector
        from rl_amd.data import LazyTensorStorage, TensorDictReplayBuffer
        from rl_amd.objectives import SACLoss
        from rl_amd.tensordict import TensorDict
        from rl_amd.trainers import OfflineToOnlineTrainer

        actor, qvalue, _ = self._actor_critic_q()
        loss = SACLoss(actor, qvalue, num_qvalue_nets=2)
        loss.make_value_estimator()
        offline = TensorDictReplayBuffer(storage=LazyTensorStorage(100), batch_size=16)
        n = 40
        offline.extend(TensorDict(
            {
                "observation": torch.randn(n, 7),
                "action": torch.rand(n, 5) * 2 - 1,
                "sample_log_prob": torch.randn(n),
                "next": {
                    "observation": torch.randn(n, 7),
                    "reward": torch.randn(n, 1),
                    "done": torch.zeros(n, 1, dtype=torch.bool),
                    "terminated": torch.zeros(n, 1, dtype=torch.bool),
                },
            },
            batch_size=[n],
        ))
        col = Collector(self._cont_env(), actor, frames_per_batch=20, total_frames=20)
        t = OfflineToOnlineTrainer(
            loss_module=loss, collector=col, total_frames=20,
            offline_buffer=offline, offline_steps=2, batch_size=16,
            optim_steps_per_batch=1,
        )
        t.pretrain()
        t.train()
        t.shutdown()


class TestGraphedPPOTrainer:
    def test_ppo_trainer_graphed_mode_cpu(self):
        """graphed=True runs the GraphedPPO loop (eager on CPU) and
        counts frames."""
        from rl_amd.collectors import Collector
        from rl_amd.trainers import PPOTrainer

        actor, critic = make_cont_actor_critic()
        env = ContinuousActionVecMockEnv(batch_size=[2], max_steps=50)
        col = Collector(env, actor, frames_per_batch=32, total_frames=-1)
        tr = PPOTrainer(
            actor=actor,
            critic=critic,
            collector=col,
            total_frames=64,
            minibatch_size=16,
            num_epochs=1,
            graphed=True,
            progress_bar=False,
        )
        tr.train()
        assert tr.collected_frames == 64
        assert any(p.grad is not None for p in actor.parameters())
        tr.shutdown()


class TestAsyncCollection:
    @pytest.mark.timeout(120)
    def test_sac_trainer_async_collection(self):
        """Collect/train overlap (reference trainers.py:1409): the
        collector thread streams into the buffer while the learner
        samples concurrently."""
        from rl_amd.trainers import SACTrainer

        env = ContinuousActionVecMockEnv(batch_size=[2], max_steps=50)
        actor, _ = make_cont_actor_critic()
        qnet = ValueOperator(
            MLP(in_features=7 + 5, out_features=1, num_cells=[32]),
            in_keys=["observation", "action"],
        )
        col = Collector(env, actor, frames_per_batch=16, total_frames=96)
        tr = SACTrainer(
            actor=actor,
            qvalue=qnet,
            collector=col,
            total_frames=96,
            batch_size=8,
            buffer_size=500,
            optim_steps_per_batch=1,
            progress_bar=False,
            async_collection=True,
        )
        tr.train()
        assert tr.collected_frames >= 96
        assert tr._optim_count > 0


@pytest.mark.gpu
@pytest.mark.timeout(300)
def test_graphed_ppo_multi_epoch_gpu():
    """GraphedPPO with epochs=2 on the GPU fast path: the epoch loop
    re-shuffles (Feistel), recomputes the batched advantage stats and
    eps draws, and replays cleanly under full-step capture."""
    import torch

    from rl_amd.collectors import Collector
    from rl_amd.envs.custom.synthetic import HalfCheetahVec
    from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal, ValueOperator
    from rl_amd.objectives import ClipPPOLoss
    from rl_amd.objectives.value.advantages import GAE
    from rl_amd.ops import (
        convert_linears_to_splitk,
        enable_splitk_bf16_cache,
        fuse_mlp3,
        refresh_splitk_caches,
    )
    from rl_amd.tensordict import TensorDictModule
    from rl_amd.trainers import GraphedPPO

    torch.manual_seed(0)
    dev = "cuda"
    env = HalfCheetahVec(batch_size=[256], device=dev)
    actor = ProbabilisticActor(
        TensorDictModule(
            torch.nn.Sequential(
                MLP(in_features=17, out_features=12, num_cells=[64, 64], device=dev),
                NormalParamExtractor(),
            ),
            in_keys=["observation"], out_keys=["loc", "scale"],
        ),
        in_keys=["loc", "scale"], distribution_class=TanhNormal,
        return_log_prob=True,
    )
    critic = ValueOperator(
        MLP(in_features=17, out_features=1, num_cells=[64, 64], device=dev),
        in_keys=["observation"],
    )
    convert_linears_to_splitk(actor)
    convert_linears_to_splitk(critic)
    enable_splitk_bf16_cache(actor)
    enable_splitk_bf16_cache(critic)
    actor.module[0].module[0] = fuse_mlp3(actor.module[0].module[0])
    critic.module = fuse_mlp3(critic.module)
    loss = ClipPPOLoss(actor, critic, normalize_advantage=True, critic_coeff=0.5)
    gae = GAE(gamma=0.99, lmbda=0.95, value_network=critic, vectorized=True)
    optim = torch.optim.Adam(
        list(actor.parameters()) + list(critic.parameters()), lr=3e-4,
        capturable=True, fused=True,
    )
    col = Collector(env, actor, frames_per_batch=256 * 8, total_frames=-1)
    runner = GraphedPPO(
        col, gae, loss, optim, minibatches=2, epochs=2,
        post_optim_hook=lambda: refresh_splitk_caches(actor, critic),
    )
    p0 = [p.detach().clone() for p in actor.parameters()]
    for _ in range(3):
        runner.step()
    torch.cuda.synchronize()
    moved = any(not torch.equal(a, b.detach())
                for a, b in zip(p0, actor.parameters()))
    assert moved
    for p in actor.parameters():
        assert torch.isfinite(p).all()
    col.shutdown()


def test_total_loss_prefers_kernel_presums():
    """GraphedPPO._total_loss: _loss_total wins; _loss_actor skips the
    component keys; plain sums otherwise."""
    import torch

    from rl_amd.tensordict import TensorDict
    from rl_amd.trainers.graphed import GraphedPPO

    t = lambda v: torch.tensor(float(v))
    self = type("S", (), {"_total_loss": GraphedPPO._total_loss})()
    out = TensorDict({"loss_objective": t(1), "loss_entropy": t(2),
                      "loss_critic": t(3)}, batch_size=[])
    assert float(GraphedPPO._total_loss(self, out)) == 6.0
    out.set("_loss_actor", t(3))  # pre-summed obj+ent
    assert float(GraphedPPO._total_loss(self, out)) == 6.0
    out.set("_loss_total", t(6))
    assert float(GraphedPPO._total_loss(self, out)) == 6.0
    out.set("loss_extra", t(4))  # extra loss keys still accumulate
    assert float(GraphedPPO._total_loss(self, out)) == 10.0
