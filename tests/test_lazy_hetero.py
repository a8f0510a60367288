"""Lazy-stacked TensorDicts/specs + heterogeneous/multi-key/dynamic
mock envs (reference LazyStackedTensorDict and
torchrl/testing/mocking_classes.py:1787,1992,2307)."""
import pytest
import torch

from rl_amd.data.tensor_specs import (
    Composite,
    LazyStackedComposite,
    LazyStackedSpec,
    Stacked,
    StackedComposite,
    Unbounded,
)
from rl_amd.envs.utils import check_env_specs
from rl_amd.tensordict import (
    LazyStackedTensorDict,
    TensorDict,
    lazy_stack,
    stack,
)
from rl_amd.testing import (
    EnvWithDynamicSpec,
    HeterogeneousCountingEnv,
    MultiKeyCountingEnv,
)


class TestLazyStackedTensorDict:
    def _hetero_pair(self):
        a = TensorDict({"obs": torch.randn(3, 4), "x": torch.randn(3, 2)}, batch_size=[3])
        b = TensorDict({"obs": torch.randn(3, 7), "x": torch.randn(3, 2)}, batch_size=[3])
        return a, b

    def test_stack_falls_back_to_lazy_on_hetero(self):
        a, b = self._hetero_pair()
        s = stack([a, b], 0)
        assert isinstance(s, LazyStackedTensorDict)
        assert s.batch_size == torch.Size([2, 3])

    def test_homo_stays_eager(self):
        a, _ = self._hetero_pair()
        s = stack([a, a.clone()], 0)
        assert not isinstance(s, LazyStackedTensorDict)

    def test_get_homo_key_dense(self):
        a, b = self._hetero_pair()
        s = lazy_stack([a, b], 0)
        assert s.get("x").shape == (2, 3, 2)

    def test_get_hetero_key_raises_with_hint(self):
        a, b = self._hetero_pair()
        s = lazy_stack([a, b], 0)
        with pytest.raises(RuntimeError, match="heterogeneous"):
            s.get("obs")

    def test_get_nestedtensor(self):
        a, b = self._hetero_pair()
        s = lazy_stack([a, b], 0)
        nt = s.get_nestedtensor("obs")
        assert nt.size(0) == 2

    def test_index_and_clone(self):
        a, b = self._hetero_pair()
        s = lazy_stack([a, b], 0)
        assert s[0] is a
        c = s.clone()
        assert isinstance(c, LazyStackedTensorDict)
        assert torch.equal(c[1].get("obs"), b.get("obs"))
        sub = s[0:1]
        assert isinstance(sub, LazyStackedTensorDict) and len(sub.tensordicts) == 1

    def test_set_distributes(self):
        a, b = self._hetero_pair()
        s = lazy_stack([a, b], 0)
        s.set("y", torch.ones(2, 3, 5))
        assert a.get("y").shape == (3, 5)

    def test_nested_inside_dense_td(self):
        a, b = self._hetero_pair()
        parent = TensorDict({}, batch_size=[])
        parent.set("group", lazy_stack([a[0], b[0]], 0))
        got = parent.get(("group", "x"))
        assert got.shape == (2, 2)

    def test_select_exclude(self):
        a, b = self._hetero_pair()
        s = lazy_stack([a, b], 0)
        assert list(s.select("x").keys()) == ["x"]
        assert "obs" not in s.exclude("obs")


class TestLazyStackedSpecs:
    def test_hetero_composite_spec(self):
        a = Composite({"obs": Unbounded(shape=(3,))})
        b = Composite({"obs": Unbounded(shape=(5,))})
        st = Stacked(a, b)
        assert isinstance(st, LazyStackedComposite)
        td = st.rand()
        assert isinstance(td, LazyStackedTensorDict)
        assert st.is_in(td)
        leaf = st["obs"]
        assert isinstance(leaf, LazyStackedSpec)
        assert tuple(leaf.shape) == (2, -1)

    def test_homo_stays_dense(self):
        a = Composite({"obs": Unbounded(shape=(3,))})
        st = StackedComposite(a, a.clone())
        assert isinstance(st, Composite)
        assert st["obs"].shape == (2, 3)

    def test_leaf_spec_roundtrip(self):
        st = Stacked(Unbounded(shape=(2,)), Unbounded(shape=(4,)))
        assert isinstance(st, LazyStackedSpec)
        vals = st.rand()
        assert [tuple(v.shape) for v in vals] == [(2,), (4,)]
        assert st.is_in(vals)


class TestNewMockEnvs:
    def test_multikey_check_env_specs(self):
        env = MultiKeyCountingEnv(batch_size=[2])
        check_env_specs(env)

    def test_multikey_rollout_keys(self):
        env = MultiKeyCountingEnv(batch_size=[2])
        r = env.rollout(3)
        assert ("nested_1", "action") in r
        assert ("nested_2", "azione") in r
        assert ("nested_1", "gift") in r.get("next")
        assert r.get(("next", "nested_2", "observation")).shape == (2, 3, 4)

    def test_heterogeneous_env_rollout(self):
        env = HeterogeneousCountingEnv(n_agents=3)
        r = env.rollout(3, return_contiguous=False)
        assert isinstance(r, LazyStackedTensorDict)
        ag = r[1].get("agents")
        assert [tuple(t.get("observation").shape) for t in ag.tensordicts] == [
            (1,), (2,), (3,),
        ]
        # counting semantics: obs at step t equals t
        assert float(ag.tensordicts[0].get("observation")[0]) == 1.0

    def test_heterogeneous_env_spec_contract(self):
        env = HeterogeneousCountingEnv(n_agents=3)
        td = env.reset()
        spec = env.observation_spec["agents"]
        assert isinstance(spec, LazyStackedComposite)
        assert spec.is_in(td.get("agents"))

    def test_dynamic_spec_env_rollout_is_lazy(self):
        env = EnvWithDynamicSpec(max_steps=4)
        r = env.rollout(3, return_contiguous=False)
        assert isinstance(r, LazyStackedTensorDict)
        assert [tuple(r[i].get("observation").shape) for i in range(3)] == [
            (1, 2), (2, 2), (3, 2),
        ]
        with pytest.raises(RuntimeError, match="heterogeneous"):
            r.get("observation")


class TestMoreMockEnvs:
    def test_stateless_counting(self):
        from rl_amd.testing import StatelessCountingEnv

        env = StatelessCountingEnv(batch_size=[3])
        check_env_specs(env)

    def test_batch_locked(self):
        from rl_amd.testing import MockBatchedLockedEnv

        env = MockBatchedLockedEnv(batch_size=[2])
        check_env_specs(env)
        bad = env.reset()[0:1]
        with pytest.raises(RuntimeError, match="batch-locked"):
            env._step(bad)

    def test_conv_mocks(self):
        from rl_amd.testing import (
            ContinuousActionConvMockEnv,
            DiscreteActionConvMockEnv,
        )

        check_env_specs(DiscreteActionConvMockEnv(batch_size=[2]))
        check_env_specs(ContinuousActionConvMockEnv(batch_size=[2]))

    def test_multiagent_counting_rollout(self):
        from rl_amd.testing import MultiAgentCountingEnv

        env = MultiAgentCountingEnv(n_agents=3, batch_size=[2])
        r = env.rollout(3)
        assert r.get(("agents", "observation")).shape == (2, 3, 3, 3)
        assert r.get(("next", "agents", "reward")).shape == (2, 3, 3, 1)

    def test_env_with_metadata(self):
        from rl_amd.testing import EnvWithMetadata

        env = EnvWithMetadata(batch_size=[2])
        td = env.reset()
        assert td.get_non_tensor("info_str") == "reset"
        r = env.rollout(2)
        assert r is not None

    def test_counting_policy_drives_counting_env(self):
        from rl_amd.collectors import Collector
        from rl_amd.testing import CountingEnv, CountingPolicy

        env = CountingEnv(batch_size=[2], max_steps=100)
        pol = CountingPolicy(env.action_spec)
        col = Collector(env, pol, frames_per_batch=10, total_frames=10)
        batch = next(iter(col))
        # counting semantics assertable exactly
        assert torch.equal(
            batch.get("observation")[0].reshape(-1),
            torch.arange(0, 5, dtype=torch.float32),
        )


class TestLazyDenseParity:
    """On HOMOGENEOUS data a forced lazy stack must behave exactly like
    the dense stack for the common operations."""

    def _pair(self):
        torch.manual_seed(0)
        tds = [
            TensorDict(
                {"a": torch.randn(4, 3), "n": {"b": torch.randn(4, 2)}},
                batch_size=[4],
            )
            for _ in range(5)
        ]
        return lazy_stack(tds, 0), stack([td.clone() for td in tds], 0)

    def test_get_and_keys(self):
        lz, dn = self._pair()
        assert lz.batch_size == dn.batch_size
        assert set(map(str, lz.keys(True, True))) == set(map(str, dn.keys(True, True)))
        assert torch.equal(lz.get("a"), dn.get("a"))
        assert torch.equal(lz.get(("n", "b")), dn.get(("n", "b")))

    def test_indexing(self):
        lz, dn = self._pair()
        assert torch.equal(lz[2].get("a"), dn[2].get("a"))
        assert torch.equal(lz[1:4].get("a"), dn[1:4].get("a"))
        assert torch.equal(lz[2, 1:3].get("a"), dn[2, 1:3].get("a"))

    def test_set_roundtrip(self):
        lz, dn = self._pair()
        v = torch.randn(5, 4, 7)
        lz.set("c", v)
        dn.set("c", v)
        assert torch.equal(lz.get("c"), dn.get("c"))

    def test_to_tensordict_apply_select(self):
        lz, dn = self._pair()
        assert torch.equal(lz.to_tensordict().get("a"), dn.get("a"))
        assert torch.equal(lz.apply(lambda t: t * 2).get("a"), dn.get("a") * 2)
        assert torch.equal(lz.select("a").get("a"), dn.get("a"))

    def test_update(self):
        lz, dn = self._pair()
        upd = TensorDict({"a": torch.ones(5, 4, 3)}, batch_size=[5, 4])
        lz.update(upd)
        dn.update(upd)
        assert torch.equal(lz.get("a"), dn.get("a"))
