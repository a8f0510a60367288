"""Collector tests: single-process hot loop, RB integration, multiprocess."""
import pytest
import torch

from rl_amd.collectors import (
    AsyncCollector,
    Collector,
    MultiAsyncCollector,
    MultiSyncCollector,
    split_trajectories,
)
from rl_amd.data import LazyTensorStorage, TensorDictReplayBuffer
from rl_amd.envs.transforms import InitTracker, StepCounter, TransformedEnv
from rl_amd.modules import MLP, ProbabilisticActor, TanhNormal, NormalParamExtractor
from rl_amd.tensordict import TensorDict, TensorDictModule
from rl_amd.testing import ContinuousActionVecMockEnv, CountingEnv


def make_env():
    return ContinuousActionVecMockEnv(batch_size=[2], max_steps=10)


def make_policy(obs_dim=7, act_dim=5):
    net = torch.nn.Sequential(
        MLP(in_features=obs_dim, out_features=2 * act_dim, num_cells=[16]),
        NormalParamExtractor(),
    )
    mod = TensorDictModule(net, in_keys=["observation"], out_keys=["loc", "scale"])
    return ProbabilisticActor(
        mod, in_keys=["loc", "scale"], distribution_class=TanhNormal, return_log_prob=True
    )


class TestCollector:
    def test_frames_exact(self):
        col = Collector(make_env(), frames_per_batch=40, total_frames=120)
        frames = sum(b.numel() for b in col)
        assert frames == 120
        col.shutdown()

    def test_batch_shape(self):
        col = Collector(
            ContinuousActionVecMockEnv(batch_size=[4]), frames_per_batch=64, total_frames=64
        )
        batch = next(iter(col))
        assert tuple(batch.batch_size) == (4, 16)
        assert ("next", "observation") in batch.keys(True, True)
        col.shutdown()

    def test_unbatched_env(self):
        col = Collector(CountingEnv(max_steps=4), frames_per_batch=12, total_frames=12)
        batch = next(iter(col))
        assert tuple(batch.batch_size) == (12,)
        col.shutdown()

    def test_policy(self):
        col = Collector(
            make_env(), make_policy(), frames_per_batch=20, total_frames=20
        )
        batch = next(iter(col))
        assert "sample_log_prob" in batch
        assert batch["action"].abs().max() <= 1.0
        col.shutdown()

    def test_traj_ids_advance(self):
        col = Collector(
            CountingEnv(max_steps=3, batch_size=[2]), frames_per_batch=24, total_frames=24
        )
        batch = next(iter(col))
        ids = batch.get(("collector", "traj_ids"))
        # 12 steps per env, 3 steps per traj → several distinct ids
        assert ids.max() > 1
        col.shutdown()

    def test_reset_at_each_iter(self):
        env = CountingEnv(max_steps=100, batch_size=[2])
        col = Collector(
            env, frames_per_batch=8, total_frames=16, reset_at_each_iter=True
        )
        batches = list(col)
        first_obs = batches[1].get("observation")[:, 0]
        assert (first_obs == 0).all()
        col.shutdown()

    def test_replay_buffer_mode(self):
        rb = TensorDictReplayBuffer(storage=LazyTensorStorage(500), batch_size=16)
        col = Collector(
            make_env(), frames_per_batch=40, total_frames=80, replay_buffer=rb
        )
        outs = [b for b in col]
        assert all(b is None for b in outs)
        assert len(rb) == 80
        col.shutdown()

    def test_init_random_frames(self):
        col = Collector(
            make_env(),
            make_policy(),
            frames_per_batch=20,
            total_frames=40,
            init_random_frames=20,
        )
        b1 = next(iter(col))
        assert "sample_log_prob" not in b1 or b1.get("sample_log_prob", None) is None
        col.shutdown()

    def test_state_dict_roundtrip(self):
        pol = make_policy()
        col = Collector(make_env(), pol, frames_per_batch=20, total_frames=40)
        next(iter(col))
        sd = col.state_dict()
        assert sd["frames"] == 20
        col2 = Collector(make_env(), make_policy(), frames_per_batch=20, total_frames=40)
        col2.load_state_dict(sd)
        assert col2._frames == 20
        col.shutdown()
        col2.shutdown()

    def test_update_policy_weights(self):
        pol = make_policy()
        col = Collector(make_env(), make_policy(), frames_per_batch=20, total_frames=20)
        col.update_policy_weights_(pol)
        for p1, p2 in zip(col.policy.parameters(), pol.parameters()):
            assert torch.allclose(p1, p2)
        col.shutdown()

    def test_transformed_env(self):
        env = TransformedEnv(make_env(), StepCounter())
        col = Collector(env, frames_per_batch=20, total_frames=20)
        batch = next(iter(col))
        assert ("next", "step_count") in batch.keys(True, True)
        col.shutdown()

    def test_start_background(self):
        rb = TensorDictReplayBuffer(storage=LazyTensorStorage(500), batch_size=8)
        col = Collector(
            make_env(), frames_per_batch=20, total_frames=100, replay_buffer=rb
        )
        col.start()
        import time

        t0 = time.time()
        while len(rb) < 20 and time.time() - t0 < 10:
            time.sleep(0.05)
        assert len(rb) >= 20
        col.async_shutdown()
        col.shutdown()


class TestSplitTrajs:
    def test_split_shapes(self):
        col = Collector(
            CountingEnv(max_steps=3, batch_size=[2]),
            frames_per_batch=20,
            total_frames=20,
            split_trajs=True,
        )
        batch = next(iter(col))
        assert "mask" in batch
        assert batch.batch_size[0] >= 2  # several trajectories split out
        # masked-out steps are zero
        col.shutdown()


class TestMultiCollectors:
    @pytest.mark.parametrize("cls", [MultiSyncCollector, MultiAsyncCollector])
    def test_frames(self, cls):
        col = cls([make_env] * 2, frames_per_batch=40, total_frames=80)
        frames = 0
        for b in col:
            frames += b.numel()
        assert frames == 80
        col.shutdown()

    def test_sync_batch_shape(self):
        col = MultiSyncCollector([make_env] * 2, frames_per_batch=40, total_frames=40)
        b = next(iter(col))
        # stack of 2 workers × [2 envs, 10 steps]
        assert tuple(b.batch_size) == (2, 2, 10)
        col.shutdown()

    def test_sync_cat_results(self):
        col = MultiSyncCollector(
            [make_env] * 2, frames_per_batch=40, total_frames=40, cat_results=0
        )
        b = next(iter(col))
        assert tuple(b.batch_size) == (4, 10)
        col.shutdown()

    def test_with_policy(self):
        col = MultiSyncCollector(
            [make_env] * 2, make_policy(), frames_per_batch=40, total_frames=40
        )
        b = next(iter(col))
        assert "sample_log_prob" in b.keys(True, True)
        col.shutdown()

    def test_update_weights(self):
        pol = make_policy()
        col = MultiSyncCollector(
            [make_env] * 2, pol, frames_per_batch=40, total_frames=40
        )
        col.update_policy_weights_()
        b = next(iter(col))
        assert b is not None
        col.shutdown()

    def test_async_collector_single(self):
        col = AsyncCollector(make_env, frames_per_batch=20, total_frames=40)
        frames = sum(b.numel() for b in col)
        assert frames == 40
        col.shutdown()

    def test_seed(self):
        col = MultiSyncCollector([make_env] * 2, frames_per_batch=40, total_frames=40)
        col.set_seed(42)
        col.shutdown()


class TestGraphedRollout:
    def test_cpu_fallback_collects(self):
        from rl_amd.collectors import GraphedRollout
        from rl_amd.envs.custom.synthetic import HalfCheetahVec
        from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal
        from rl_amd.tensordict import TensorDictModule

        env = HalfCheetahVec(batch_size=[8])
        net = torch.nn.Sequential(
            MLP(in_features=17, out_features=12, num_cells=[16]), NormalParamExtractor()
        )
        actor = ProbabilisticActor(
            TensorDictModule(net, in_keys=["observation"], out_keys=["loc", "scale"]),
            in_keys=["loc", "scale"],
            distribution_class=TanhNormal,
            return_log_prob=True,
        )
        gr = GraphedRollout(env, actor, horizon=5).initialize()
        out = gr.collect()
        assert tuple(out.batch_size) == (8, 5)
        assert torch.isfinite(out.get(("next", "reward"))).all()
        # consecutive collects continue the stream (obs changes)
        o1 = out.get("observation").clone()
        out2 = gr.collect()
        assert not torch.allclose(o1, out2.get("observation"))

    @pytest.mark.gpu
    def test_gpu_capture_matches_semantics(self):
        from rl_amd.collectors import GraphedRollout
        from rl_amd.envs.custom.synthetic import HalfCheetahVec
        from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal
        from rl_amd.tensordict import TensorDictModule

        env = HalfCheetahVec(batch_size=[64], device="cuda")
        net = torch.nn.Sequential(
            MLP(in_features=17, out_features=12, num_cells=[16], device="cuda"),
            NormalParamExtractor(),
        )
        actor = ProbabilisticActor(
            TensorDictModule(net, in_keys=["observation"], out_keys=["loc", "scale"]),
            in_keys=["loc", "scale"],
            distribution_class=TanhNormal,
            return_log_prob=True,
        )
        gr = GraphedRollout(env, actor, horizon=8).initialize()
        assert gr.captured, "hipGraph capture must succeed on this env/policy"
        out = gr.collect()
        assert torch.isfinite(out.get("sample_log_prob")).all()
        obs1 = out.get("observation")[:, 0].clone()
        gr.collect()
        obs2 = out.get("observation")[:, 0]
        assert not torch.allclose(obs1, obs2)
        # env dynamics hold inside the capture: next_obs = f(obs, action)
        o = out.get("observation")[:, 0]
        a = out.get("action")[:, 0].clamp(-1, 1)
        expected = torch.tanh(o @ env.A + a @ env.B)
        assert torch.allclose(expected, out.get(("next", "observation"))[:, 0], atol=1e-4)


def _make_slow_env():
    import time as _time

    from rl_amd.testing import ContinuousActionVecMockEnv

    class SlowEnv(ContinuousActionVecMockEnv):
        def _step(self, td):
            _time.sleep(0.05)
            return super()._step(td)

    return SlowEnv(batch_size=[2], max_steps=10)


def _make_fast_env():
    from rl_amd.testing import ContinuousActionVecMockEnv

    return ContinuousActionVecMockEnv(batch_size=[2], max_steps=10)


class TestPreemption:
    @pytest.mark.timeout(180)
    def test_preemptive_threshold_pads_stragglers(self):
        """One slow worker + one fast: with preemptive_threshold=0.5 the
        fast worker's completion interrupts the slow one; its partial
        batch is zero-padded so the stack still shapes up (reference
        behavior: _multi_sync preemption)."""
        col = MultiSyncCollector(
            [_make_fast_env, _make_slow_env],
            frames_per_batch=80,
            total_frames=160,
            preemptive_threshold=0.5,
        )
        try:
            n = 0
            for batch in col:
                # [workers=2, envs=2, T<=20]
                assert batch.batch_size[0] == 2
                assert batch.batch_size[-1] <= 20
                n += 1
            assert n == 2
        finally:
            col.shutdown()


class TestGraphedPPONumerics:
    """GraphedPPO's orchestration must be EXACTLY the textbook loop —
    this is the math under bench.py's measured number."""

    def _build(self, seed):
        from rl_amd.collectors import Collector
        from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal, ValueOperator
        from rl_amd.objectives import ClipPPOLoss
        from rl_amd.objectives.value.advantages import GAE
        from rl_amd.tensordict import TensorDictModule
        from rl_amd.testing import ContinuousActionVecMockEnv

        torch.manual_seed(seed)
        env = ContinuousActionVecMockEnv(batch_size=[4], max_steps=100)
        env.set_seed(seed)
        net = torch.nn.Sequential(
            MLP(in_features=7, out_features=2 * 5, num_cells=[16]),
            NormalParamExtractor(),
        )
        actor = ProbabilisticActor(
            TensorDictModule(net, in_keys=["observation"], out_keys=["loc", "scale"]),
            in_keys=["loc", "scale"],
            distribution_class=TanhNormal,
            return_log_prob=True,
        )
        critic = ValueOperator(
            MLP(in_features=7, out_features=1, num_cells=[16]), in_keys=["observation"]
        )
        col = Collector(env, actor, frames_per_batch=16, total_frames=-1)
        loss = ClipPPOLoss(actor, critic, normalize_advantage=True)
        gae = GAE(gamma=0.99, lmbda=0.95, value_network=critic, vectorized=True)
        optim = torch.optim.Adam(
            list(actor.parameters()) + list(critic.parameters()), lr=1e-3
        )
        return col, actor, critic, loss, gae, optim

    def test_step_matches_manual_loop(self):
        from rl_amd.trainers import GraphedPPO

        col1, actor1, critic1, loss1, gae1, optim1 = self._build(7)
        runner = GraphedPPO(col1, gae1, loss1, optim1, minibatches=2, epochs=1,
                            capture=False)
        runner.step()
        runner.step()

        col2, actor2, critic2, loss2, gae2, optim2 = self._build(7)
        for _ in range(2):
            batch = col2.rollout()
            with torch.no_grad():
                gae2(batch)
            flat = batch.reshape(-1)
            n = flat.batch_size[0]
            mb = n // 2
            perm = torch.randperm(n)
            shuffled = flat[perm]
            for i in range(2):
                sub = shuffled[i * mb : (i + 1) * mb]
                out = loss2(sub)
                total = sum(
                    v for k, v in out.items()
                    if isinstance(k, str) and k.startswith("loss_")
                )
                optim2.zero_grad(set_to_none=False)
                total.backward()
                torch.nn.utils.clip_grad_norm_(
                    list(actor2.parameters()) + list(critic2.parameters()), 1.0
                )
                optim2.step()

        for p1, p2 in zip(actor1.parameters(), actor2.parameters()):
            assert torch.equal(p1, p2)
        for p1, p2 in zip(critic1.parameters(), critic2.parameters()):
            assert torch.equal(p1, p2)


def _gpu_cheetah_env():
    from rl_amd.envs.custom.synthetic import HalfCheetahVec

    return HalfCheetahVec(batch_size=[8], device="cuda")


@pytest.mark.gpu
@pytest.mark.timeout(300)
def test_multisync_collector_gpu_envs():
    """MultiSync workers with GPU-resident envs on one device (HIP IPC
    shares the result buffers across processes)."""
    from rl_amd.collectors import MultiSyncCollector

    col = MultiSyncCollector(
        [_gpu_cheetah_env] * 2, frames_per_batch=64, total_frames=128
    )
    batches = list(col)
    assert sum(b.numel() for b in batches) == 128
    assert batches[0].get("observation").shape[-1] == 17
    col.shutdown()
