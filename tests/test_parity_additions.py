"""Unit tests for the reference-parity additions: MARL grouping, MCTS
scores, cross-group critic, sample units, trajectory queries, robot
macros, FinancialRegimeEnv, LLMHashingEnv and root utilities."""
import pytest
import torch

from rl_amd.tensordict import TensorDict


class TestMarlGrouping:
    def test_group_map_types(self):
        from rl_amd.envs import MarlGroupMapType, check_marl_grouping

        agents = ["a", "b", "c"]
        one = MarlGroupMapType.ALL_IN_ONE_GROUP.get_group_map(agents)
        assert one == {"agents": agents}
        per = MarlGroupMapType.ONE_GROUP_PER_AGENT.get_group_map(agents)
        assert per == {"a": ["a"], "b": ["b"], "c": ["c"]}
        check_marl_grouping(one, agents)
        check_marl_grouping(per, agents)

    @pytest.mark.parametrize(
        "bad",
        [
            {},  # empty map
            {"g": []},  # empty group
            {"g": ["a"], "h": ["a", "b"]},  # duplicate agent
            {"g": ["a", "z"]},  # unknown agent
            {"g": ["a"]},  # missing agent b
        ],
    )
    def test_invalid_groupings_raise(self, bad):
        from rl_amd.envs import check_marl_grouping

        with pytest.raises(ValueError):
            check_marl_grouping(bad, ["a", "b"])


class TestMCTSScores:
    def _td(self):
        return TensorDict(
            {
                "win_count": torch.tensor([[3.0, 1.0, 0.0]]),
                "visits": torch.tensor([[4.0, 2.0, 0.0]]),
                "prior_prob": torch.tensor([[0.5, 0.3, 0.2]]),
            },
            batch_size=[1],
        )

    def test_puct_matches_formula(self):
        from rl_amd.modules import MCTSScores

        td = MCTSScores.PUCT(c=2.0)(self._td())
        n_total = 6.0
        expect0 = 3.0 / 4.0 + 2.0 * 0.5 * n_total ** 0.5 / (1 + 4.0)
        assert torch.isclose(td.get("score")[0, 0], torch.tensor(expect0))

    def test_ucb1_unvisited_is_inf(self):
        from rl_amd.modules import MCTSScores

        td = MCTSScores.UCB1()(self._td())
        assert torch.isinf(td.get("score")[0, 2])

    def test_ucb1_tuned_narrower_than_ucb1_for_low_variance(self):
        from rl_amd.modules import MCTSScores

        td = self._td()
        td.set("reward_variance", torch.zeros(1, 3))
        tuned = MCTSScores.UCB1_TUNED()(td.clone()).get("score")[0, 0]
        plain = MCTSScores.UCB1()(self._td()).get("score")[0, 0]
        assert tuned < plain

    def test_exp3_is_distribution(self):
        from rl_amd.modules import MCTSScores

        td = MCTSScores.EXP3(gamma=0.1)(self._td())
        p = td.get("score")
        assert torch.isclose(p.sum(), torch.tensor(1.0))
        assert (p >= 0.1 / 3 - 1e-6).all()


class TestCrossGroupCritic:
    def test_shapes_and_grad(self):
        from rl_amd.modules import CrossCriticGroupSpec, CrossGroupCritic

        specs = [
            CrossCriticGroupSpec(obs_dim=5, n_agents=3, obs_key=("g1", "obs"), value_key=("g1", "v")),
            CrossCriticGroupSpec(obs_dim=7, n_agents=2, obs_key=("g2", "obs"), value_key=("g2", "v")),
        ]
        critic = CrossGroupCritic(specs, embed_dim=8, hidden_dim=16)
        td = TensorDict(
            {"g1": {"obs": torch.randn(4, 3, 5)}, "g2": {"obs": torch.randn(4, 2, 7)}},
            batch_size=[4],
        )
        out = critic(td)
        assert out.get(("g1", "v")).shape == (4, 3, 1)
        assert out.get(("g2", "v")).shape == (4, 2, 1)
        (out.get(("g1", "v")).sum() + out.get(("g2", "v")).sum()).backward()
        assert all(p.grad is not None for p in critic.parameters())


class TestSampleUnits:
    def test_transition_identity(self):
        from rl_amd.data import Transition

        idx = torch.tensor([3, 7])
        out, info = Transition().expand(idx)
        assert torch.equal(out, idx) and info == {}

    def test_sequence_window(self):
        from rl_amd.data import Sequence

        unit = Sequence(3, burn_in=2, bootstrap=1)
        idx, info = unit.expand(torch.tensor([10]), storage=list(range(100)))
        assert idx.tolist() == [[8, 9, 10, 11, 12, 13]]
        assert info["burn_in"] == 2

    def test_sequence_clamps_to_storage(self):
        from rl_amd.data import Sequence

        idx, _ = Sequence(3, burn_in=2).expand(torch.tensor([0]), storage=list(range(50)))
        assert idx.min() >= 0


class TestTrajectoryQueries:
    def _data(self):
        done = torch.zeros(10, 1, dtype=torch.bool)
        done[3] = True
        done[7] = True
        return TensorDict(
            {"reward": torch.arange(10.0).reshape(10, 1), "next": {"done": done}},
            batch_size=[10],
        )

    def test_iter_trajectories_by_done(self):
        from rl_amd.data import iter_trajectories

        lens = [len(t) for t in iter_trajectories(self._data())]
        assert lens == [4, 4, 2]

    def test_traj_predicate_filter(self):
        from rl_amd.data import filter_trajectories, traj

        out = filter_trajectories(self._data(), traj.length > 2)
        assert out.batch_size[0] == 8
        out2 = filter_trajectories(self._data(), traj.total_reward > 20)
        assert out2.batch_size[0] == 4  # only rewards 4+5+6+7=22

    def test_predicate_composition(self):
        from rl_amd.data import filter_trajectories, traj

        pred = (traj.length > 2) & ~(traj.total_reward > 20)
        out = filter_trajectories(self._data(), pred)
        assert out.batch_size[0] == 4  # first traj: 0+1+2+3=6

    def test_find_start_stop_traj(self):
        from rl_amd.data import find_start_stop_traj

        end = torch.zeros(10, dtype=torch.bool)
        end[3] = True
        starts, stops, lens = find_start_stop_traj(end=end, at_capacity=False)
        assert starts.tolist() == [0, 4] and stops.tolist() == [3, 9]
        assert lens.tolist() == [4, 6]


class TestDataExtras:
    def test_tensor_map(self):
        from rl_amd.data import TensorMap

        tm = TensorMap()
        k1, k2 = torch.tensor([1, 2, 3]), torch.tensor([1, 2, 4])
        tm[k1] = "a"
        assert k1 in tm and k2 not in tm and tm[k1] == "a"

    def test_vocab_tail_tokenizer(self):
        from rl_amd.data import VocabTailActionTokenizer

        tok = VocabTailActionTokenizer(50257, n_bins=128)
        a = torch.tensor([[0.25, -0.7]])
        enc = tok.encode(a)
        assert enc.min() >= 50257 - 128 and enc.max() < 50257
        assert (tok.decode(enc) - a).abs().max() <= 1 / 128 + 1e-6

    def test_validate_vla_schema(self):
        from rl_amd.data import validate_vla_tensordict

        good = TensorDict(
            {"observation": {"image": torch.zeros(2, 3, 8, 8, dtype=torch.uint8)}},
            batch_size=[2],
        )
        validate_vla_tensordict(good)
        bad = TensorDict(
            {"observation": {"image": torch.zeros(2, 3, 8, 8)}}, batch_size=[2]
        )
        with pytest.raises(ValueError):
            validate_vla_tensordict(bad)

    def test_prefill_replay_buffer(self):
        from rl_amd.data import ReplayBuffer, prefill_replay_buffer
        from rl_amd.data.replay_buffers import LazyTensorStorage

        rb = ReplayBuffer(storage=LazyTensorStorage(100))
        data = TensorDict({"x": torch.randn(20, 3)}, batch_size=[20])
        n = prefill_replay_buffer(rb, data, num_transitions=10)
        assert n == 10 and len(rb) == 10


class TestRobotMacros:
    def test_reach_joints_expansion(self):
        from rl_amd.envs import RobotMacroAction, URScriptPrimitiveTransform

        tf = URScriptPrimitiveTransform(gripper_dim=False)
        start = torch.zeros(6)
        macro = RobotMacroAction.reach_joints(torch.ones(6), steps=4)
        seq = tf.expand_macro(macro, start)
        assert seq.shape == (4, 6)
        assert torch.allclose(seq[-1], torch.ones(6))
        assert torch.allclose(seq[0], torch.full((6,), 0.25))

    def test_gripper_channel(self):
        from rl_amd.envs import RobotMacroAction, URScriptPrimitiveTransform

        tf = URScriptPrimitiveTransform(gripper_dim=True)
        start = torch.zeros(7)
        seq = tf.expand_macro(RobotMacroAction.open_gripper(steps=2), start)
        assert seq.shape == (2, 7)
        assert (seq[:, -1] == 1.0).all()  # gripper channel driven open
        assert torch.allclose(seq[:, :-1], torch.zeros(2, 6))  # arm holds

    def test_satellite_slew_normalized(self):
        from rl_amd.envs import SatelliteAttitudeTransform, SatelliteMacroAction

        tf = SatelliteAttitudeTransform(gripper_dim=False)
        start = torch.tensor([1.0, 0.0, 0.0, 0.0])
        target = torch.tensor([0.0, 1.0, 0.0, 0.0])
        seq = tf.expand_macro(SatelliteMacroAction.slew_to(target, steps=5), start)
        assert seq.shape == (5, 4)
        assert torch.allclose(seq.norm(dim=-1), torch.ones(5), atol=1e-5)


class TestNewEnvs:
    def test_financial_regime_rollout(self):
        from rl_amd.envs import FinancialRegimeEnv

        env = FinancialRegimeEnv(batch_size=(8,), window_size=20, episode_len=16)
        r = env.rollout(20)
        assert r.batch_size == torch.Size([8, 16])  # episode_len terminates
        assert r.get(("next", "price_history")).shape == (8, 16, 20)

    def test_financial_buy_costs(self):
        from rl_amd.envs import FinancialRegimeEnv

        torch.manual_seed(0)
        env = FinancialRegimeEnv(batch_size=(4,), volatility=0.0, drift=0.0,
                                 transaction_cost=0.01)
        env.reset()
        td = TensorDict({"action": torch.ones(4, dtype=torch.long)}, batch_size=[4])
        out = env.step(td)
        # zero vol/drift: reward is exactly the transaction cost
        assert (out.get(("next", "reward")) < 0).all()
        assert out.get(("next", "current_holdings")).all()

    def test_llm_hashing_env_distinguishes_chains(self):
        from rl_amd.envs import LLMHashingEnv

        env = LLMHashingEnv(64)
        td = env.reset()
        a = td.clone()
        a.set("action", torch.tensor(3))
        b = td.clone()
        b.set("action", torch.tensor(4))
        ha = env.step(a).get(("next", "hash"))
        env.reset()
        hb = env.step(b).get(("next", "hash"))
        assert ha.item() != hb.item()

    def test_gated_envs_raise_import_error(self):
        from rl_amd.envs import AntEnv, BraxEnv, PettingZooEnv, VmasEnv

        for cls in (AntEnv, BraxEnv, PettingZooEnv, VmasEnv):
            with pytest.raises(ImportError):
                cls("x")


class TestModulesExtras:
    def test_vmap_module(self):
        from rl_amd.modules import VmapModule
        from rl_amd.tensordict.nn import TensorDictModule

        mod = TensorDictModule(torch.nn.Linear(3, 2), in_keys=["x"], out_keys=["y"])
        vm = VmapModule(mod, vmap_dim=0)
        td = TensorDict({"x": torch.randn(5, 4, 3)}, batch_size=[5, 4])
        out = vm(td)
        assert out.get("y").shape == (5, 4, 2)

    def test_one_hot_ordinal(self):
        from rl_amd.modules import OneHotOrdinal

        d = OneHotOrdinal(torch.randn(6, 10))
        s = d.sample()
        assert s.shape == (6, 10) and (s.sum(-1) == 1).all()

    def test_distributions_maps(self):
        from rl_amd.modules import distributions_maps

        assert distributions_maps("tanh_normal").__name__ == "TanhNormal"
        with pytest.raises(NotImplementedError):
            distributions_maps("nope")

    def test_recurrent_precision_roundtrip(self):
        from rl_amd.modules import (
            RecurrentMatmulPrecision,
            get_recurrent_matmul_precision,
            set_recurrent_matmul_precision,
        )

        prev = get_recurrent_matmul_precision()
        set_recurrent_matmul_precision("low")
        assert get_recurrent_matmul_precision() is RecurrentMatmulPrecision.LOW
        set_recurrent_matmul_precision(prev)

    def test_get_env_transforms_from_module(self):
        from rl_amd.modules import LSTMModule, get_env_transforms_from_module

        lstm = LSTMModule(input_size=4, hidden_size=8, in_key="observation", out_key="emb")
        tf = get_env_transforms_from_module(lstm)
        assert type(tf).__name__ == "Compose"


class TestObjectivesExtras:
    def test_group_optimizers_steps_all(self):
        from rl_amd.objectives import group_optimizers

        p1 = torch.nn.Parameter(torch.ones(3))
        p2 = torch.nn.Parameter(torch.ones(2))
        opt = group_optimizers(torch.optim.SGD([p1], lr=1.0), torch.optim.SGD([p2], lr=1.0))
        (p1.sum() + p2.sum()).backward()
        opt.step()
        assert torch.allclose(p1.detach(), torch.zeros(3))
        assert torch.allclose(p2.detach(), torch.zeros(2))

    def test_group_optimizers_type_mismatch(self):
        from rl_amd.objectives import group_optimizers

        p = torch.nn.Parameter(torch.ones(1))
        with pytest.raises(ValueError):
            group_optimizers(torch.optim.SGD([p], lr=1.0), torch.optim.Adam([p]))

    def test_categorical_kl_balanced_nonnegative(self):
        from rl_amd.objectives import categorical_kl_balanced

        kl = categorical_kl_balanced(torch.randn(4, 8, 16), torch.randn(4, 8, 16))
        assert kl.item() >= 0


class TestRootUtils:
    def test_mask_batch(self):
        from rl_amd.trainers import mask_batch

        td = TensorDict(
            {"x": torch.arange(4.0), "collector": {"mask": torch.tensor([1, 0, 1, 0], dtype=torch.bool)}},
            batch_size=[4],
        )
        out = mask_batch(td)
        assert out.batch_size[0] == 2

    def test_transport_backend_scoped(self):
        import rl_amd

        with rl_amd.transport_backend("gloo"):
            pass
        with pytest.raises(ValueError):
            rl_amd.transport_backend("bogus").__enter__()

    def test_compile_with_warmup_eager_path(self):
        import rl_amd

        calls = []

        @rl_amd.compile_with_warmup(warmup=100)
        def f(x):
            calls.append(1)
            return x + 1

        assert f(torch.tensor(1.0)) == 2.0 and len(calls) == 1

    def test_cuda_memory_stats_cpu_safe(self):
        import rl_amd

        assert isinstance(rl_amd.cuda_memory_stats(), dict)
        rl_amd.reset_cuda_peak_stats()

    def test_merge_ray_runtime_env(self):
        import rl_amd

        merged = rl_amd.merge_ray_runtime_env(
            {"env_vars": {"A": "1"}, "pip": ["x"]}, {"env_vars": {"B": "2"}}
        )
        assert merged["env_vars"] == {"A": "1", "B": "2"} and merged["pip"] == ["x"]
