"""Env layer tests: EnvBase contract, step_mdp, transforms."""
import pytest
import torch

from rl_amd.envs import check_env_specs, step_mdp, terminated_or_truncated
from rl_amd.envs.transforms import (
    CatFrames,
    CatTensors,
    ClipTransform,
    Compose,
    DoubleToFloat,
    ExcludeTransform,
    FiniteTensorDictCheck,
    FlattenObservation,
    InitTracker,
    ObservationNorm,
    RenameTransform,
    RewardClipping,
    RewardScaling,
    RewardSum,
    SignTransform,
    StepCounter,
    TransformedEnv,
    UnsqueezeTransform,
    VecNorm,
)
from rl_amd.tensordict import TensorDict
from rl_amd.testing import (
    ContinuousActionVecMockEnv,
    CountingEnv,
    DiscreteActionVecMockEnv,
    NestedCountingEnv,
)


def _ones_policy(td):
    td.set("action", torch.ones(*td.batch_size, 1, dtype=torch.bool))
    return td


class TestEnvBase:
    @pytest.mark.parametrize("batch_size", [(), (4,)])
    def test_counting_env_specs(self, batch_size):
        env = CountingEnv(batch_size=batch_size)
        check_env_specs(env, seed=0)

    def test_counting_env_determinism(self):
        env = CountingEnv(max_steps=5, batch_size=[2])
        td = env.reset()
        for t in range(3):
            td.set("action", torch.ones(2, 1, dtype=torch.bool))
            td = env.step(td)
            assert (td.get(("next", "observation")) == t + 1).all()
            td = step_mdp(td)

    def test_step_and_maybe_reset(self):
        env = CountingEnv(max_steps=3, batch_size=[2])
        td = env.reset()
        for _ in range(3):
            td.set("action", torch.ones(2, 1, dtype=torch.bool))
            td, td_next = env.step_and_maybe_reset(td)
            td = td_next
        # after exactly max_steps the env must have reset
        assert (td["observation"] == 0).all()

    def test_rollout_break_on_done(self):
        env = CountingEnv(max_steps=3)

        def policy(td):
            td.set("action", torch.ones(1, dtype=torch.bool))
            return td

        r = env.rollout(100, policy=policy)
        assert r.batch_size[0] == 3

    def test_rollout_no_break(self):
        env = CountingEnv(max_steps=3)
        r = env.rollout(10, break_when_any_done=False)
        assert r.batch_size[0] == 10
        # done was hit several times; obs resets to 0 after each done
        obs = r.get("observation").flatten()
        assert obs.max() <= 3

    def test_rollout_policy(self):
        env = ContinuousActionVecMockEnv(batch_size=[2])

        def policy(td):
            td.set("action", torch.zeros(2, env.action_dim))
            return td

        r = env.rollout(5, policy=policy)
        assert (r.get("action") == 0).all()

    def test_nested_env(self):
        env = NestedCountingEnv(batch_size=[2])
        r = env.rollout(4, break_when_any_done=False)
        assert ("data", "states") in r.keys(True, True)

    def test_step_mdp_carries_other_keys(self):
        env = CountingEnv(batch_size=[2])
        td = env.reset()
        td.set("hidden", torch.randn(2, 8))
        td.set("action", torch.ones(2, 1, dtype=torch.bool))
        td = env.step(td)
        nxt = step_mdp(td)
        assert "hidden" in nxt
        assert "action" not in nxt
        assert "reward" not in nxt

    def test_terminated_or_truncated(self):
        td = TensorDict(
            {
                "terminated": torch.tensor([[True], [False]]),
                "truncated": torch.tensor([[False], [False]]),
                "done": torch.tensor([[True], [False]]),
            },
            batch_size=[2],
        )
        assert terminated_or_truncated(td, key="_reset")
        assert td.get("_reset")[0].item() is True

    def test_fake_tensordict(self):
        env = ContinuousActionVecMockEnv(batch_size=[3])
        fake = env.fake_tensordict()
        assert fake.get("observation").shape == (3, env.obs_dim)
        assert ("next", "reward") in fake.keys(True, True)


class TestTransforms:
    def test_step_counter_truncates(self):
        env = TransformedEnv(CountingEnv(max_steps=100), StepCounter(max_steps=5))
        r = env.rollout(100)
        assert r.batch_size[0] == 5
        assert r.get(("next", "truncated"))[-1].item() is True

    def test_init_tracker(self):
        env = TransformedEnv(CountingEnv(max_steps=3), InitTracker())
        r = env.rollout(10, policy=_ones_policy, break_when_any_done=False)
        is_init = r.get("is_init").flatten()
        assert is_init[0].item() is True
        # reset happens after step 3 → step index 3 is_init again
        assert is_init[3].item() is True
        assert is_init[1].item() is False

    def test_reward_sum(self):
        env = TransformedEnv(CountingEnv(max_steps=10), RewardSum())
        r = env.rollout(5)
        er = r.get(("next", "episode_reward")).flatten()
        assert torch.allclose(er, torch.arange(1.0, 6.0))

    def test_observation_norm(self):
        env = TransformedEnv(
            CountingEnv(max_steps=10), ObservationNorm(loc=1.0, scale=2.0, in_keys=["observation"])
        )
        td = env.reset()
        assert td["observation"].item() == pytest.approx(-0.5)

    def test_observation_norm_init_stats(self):
        env = TransformedEnv(
            ContinuousActionVecMockEnv(batch_size=[4]),
            ObservationNorm(in_keys=["observation"]),
        )
        env.transform[0].init_stats(num_iter=32)
        assert env.transform[0].initialized
        td = env.reset()
        assert td["observation"].shape == (4, 7)

    def test_reward_scaling_clipping_sign(self):
        env = TransformedEnv(
            CountingEnv(max_steps=10),
            Compose(RewardScaling(loc=0.0, scale=3.0), RewardClipping(-2, 2), SignTransform()),
        )
        r = env.rollout(3)
        assert (r.get(("next", "reward")) == 1.0).all()

    def test_cat_tensors(self):
        env = TransformedEnv(
            NestedCountingEnv(), CatTensors(in_keys=[("data", "states")], out_key="obs_vec")
        )
        td = env.reset()
        assert "obs_vec" in td

    def test_rename(self):
        env = TransformedEnv(
            CountingEnv(), RenameTransform(in_keys=["observation"], out_keys=["obs2"])
        )
        td = env.reset()
        assert "obs2" in td and "observation" not in td
        assert "obs2" in env.observation_spec

    def test_exclude(self):
        env = TransformedEnv(
            TransformedEnv(CountingEnv(), StepCounter()), ExcludeTransform("step_count")
        )
        td = env.reset()
        assert "step_count" not in td

    def test_double_to_float(self):
        class F64Env(CountingEnv):
            def _reset(self, tensordict=None, **kwargs):
                td = super()._reset(tensordict, **kwargs)
                td.set("observation", td.get("observation").double())
                return td

            def _step(self, tensordict):
                td = super()._step(tensordict)
                td.set("observation", td.get("observation").double())
                return td

        env = TransformedEnv(F64Env(), DoubleToFloat())
        td = env.reset()
        assert td["observation"].dtype == torch.float32

    def test_cat_frames(self):
        env = TransformedEnv(
            CountingEnv(max_steps=100),
            CatFrames(N=3, dim=-1, in_keys=["observation"]),
        )
        td = env.reset()
        assert td["observation"].shape == (3,)
        r = env.rollout(4, policy=_ones_policy)
        # after 4 always-increment steps the 3-frame stack holds [2,3,4]
        last = r.get(("next", "observation"))[-1]
        assert last.tolist() == [2.0, 3.0, 4.0]

    def test_finite_check(self):
        env = TransformedEnv(CountingEnv(), FiniteTensorDictCheck())
        env.rollout(2)

    def test_clip(self):
        env = TransformedEnv(
            CountingEnv(max_steps=10), ClipTransform(in_keys=["observation"], low=0, high=2)
        )
        r = env.rollout(5)
        assert r.get(("next", "observation")).max() <= 2

    def test_vecnorm_stats_converge(self):
        env = TransformedEnv(
            ContinuousActionVecMockEnv(batch_size=[8]), VecNorm(in_keys=["observation"], decay=1.0)
        )
        r = env.rollout(20, break_when_any_done=False)
        obs = r.get(("next", "observation"))
        # normalized obs should have roughly unit scale
        assert obs.abs().mean() < 5.0

    def test_compose_order_and_spec(self):
        env = TransformedEnv(
            ContinuousActionVecMockEnv(batch_size=[2]),
            Compose(
                UnsqueezeTransform(dim=-1, in_keys=["observation"]),
                FlattenObservation(first_dim=-2, last_dim=-1, in_keys=["observation"]),
            ),
        )
        td = env.reset()
        assert td["observation"].shape == (2, 7)
        check_env_specs(env)

    def test_transformed_env_append(self):
        env = TransformedEnv(CountingEnv())
        env.append_transform(StepCounter())
        td = env.reset()
        assert "step_count" in td


class TestToyVLAEnv:
    def test_echo_mode_schema(self):
        from rl_amd.envs import ToyVLAEnv, check_env_specs

        env = ToyVLAEnv(batch_size=[2], seed=0)
        check_env_specs(env)
        td = env.reset()
        assert td["observation", "image"].shape == (2, 3, 16, 16)
        assert td["observation", "image"].dtype == torch.uint8
        td.set("action", 0.5 * torch.ones(2, 4))
        td = env.step(td)
        # the state echoes the executed action
        assert (td["next", "observation", "state"][:, :4] == 0.5).all()
        # effort penalty
        assert (td["next", "reward"] < 0).all()
        assert not td["next", "done"].any()

    def test_tracking_oracle_succeeds(self):
        from rl_amd.envs import ToyVLAEnv

        env = ToyVLAEnv(action_dim=2, state_dim=4, success_steps=2, seed=0)
        td = env.reset()
        for _ in range(2):
            td.set("action", td["observation", "state"][..., 2:4])
            td = env.step(td)["next"].exclude("reward")
        assert bool(td["success"].item()) and bool(td["terminated"].item())

    def test_tracking_random_rarely_succeeds(self):
        from rl_amd.envs import ToyVLAEnv

        env = ToyVLAEnv(batch_size=[8], action_dim=4, state_dim=8, success_steps=3, seed=1)
        r = env.rollout(10, break_when_any_done=False)
        assert not r["next", "success"].all()

    def test_group_repeats(self):
        from rl_amd.envs import ToyVLAEnv

        env = ToyVLAEnv(action_dim=2, state_dim=4, success_steps=1, group_repeats=2, seed=0)
        targets, gids = [], []
        for _ in range(4):
            td = env.reset()
            targets.append(td["observation", "state"][..., 2:4].clone())
            gids.append(int(td["group_id"].item()))
        assert torch.equal(targets[0], targets[1])
        assert not torch.equal(targets[1], targets[2])
        assert gids == [0, 0, 1, 1]

    def test_from_pixels_render(self):
        from rl_amd.envs import ToyVLAEnv

        env = ToyVLAEnv(batch_size=[2], success_steps=3, state_dim=8, from_pixels=True, seed=0)
        td = env.reset()
        assert td["pixels"].shape == (2, 64, 64, 3)
        # target marker drawn in green channel
        assert (td["pixels"][..., 1] > 0).any()


class TestChessEnv:
    def test_gated_import_error(self):
        """chess is not installed in this image: the env must raise a
        clear ImportError (reference gating pattern)."""
        import importlib.util

        from rl_amd.envs import ChessEnv

        if importlib.util.find_spec("chess") is None:
            with pytest.raises(ImportError, match="chess"):
                ChessEnv()
        else:
            env = ChessEnv()
            td = env.reset()
            assert td["action_mask"].sum() == 20  # legal openings
            td.set("action", torch.tensor([0]))
            td = env.step(td)
            assert not td["next", "done"].item()


class TestGatedLibWrappers:
    @pytest.mark.parametrize(
        "name",
        [
            "EnvPoolEnv", "ProcgenEnv", "SafetyGymnasiumEnv", "RoboHiveEnv",
            "HabitatEnv", "JumanjiEnv", "IsaacLabEnv", "IsaacGymEnv",
            "MjLabEnv", "MujocoPlaygroundEnv", "GenesisEnv", "SMACv2Env",
            "MeltingpotEnv", "OpenSpielEnv", "UnityMLAgentsEnv", "LiberoEnv",
            "OpenMLEnv",
        ],
    )
    def test_raises_clear_import_error(self, name):
        import rl_amd.envs.libs as libs

        cls = getattr(libs, name)
        with pytest.raises((ImportError, NotImplementedError)) as exc:
            cls(None)
        if isinstance(exc.value, ImportError):
            assert "not installed" in str(exc.value)


def test_step_mdp_excludes_nested_reward_and_action():
    """Nested (agent-grouped) reward/action keys must not leak into the
    next root (regression: leaked ('agents','reward') made collector
    batches heterogeneous → silent lazy-stack degradation)."""
    from rl_amd.testing import MultiAgentCountingEnv

    env = MultiAgentCountingEnv(n_agents=3, batch_size=[2])
    td = env.reset()
    td = env.rand_action(td)
    td, next_root = env.step_and_maybe_reset(td)
    assert ("agents", "reward") not in next_root
    assert ("agents", "action") not in next_root
    assert ("agents", "observation") in next_root
    # collector batches stay DENSE
    from rl_amd.collectors import Collector
    from rl_amd.tensordict import LazyStackedTensorDict

    col = Collector(env, None, frames_per_batch=8, total_frames=8)
    batch = next(iter(col))
    assert not isinstance(batch, LazyStackedTensorDict)
    assert not isinstance(batch.get("agents"), LazyStackedTensorDict)
