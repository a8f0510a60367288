"""Batch-3 transform tests (reference test model: pytorch/rl
test/test_transforms.py per-transform classes)."""
import math

import pytest
import torch

from rl_amd.data import LazyTensorStorage, TensorDictReplayBuffer
from rl_amd.data.vla import UniformActionTokenizer
from rl_amd.envs.transforms import (
    TransformedEnv,
    ActionChunkTransform,
    ActionScaling,
    ActionTokenizerTransform,
    ExpandAs,
    FlattenTensorDict,
    LineariseRewards,
    MeanActionSelector,
    ModuleTransform,
    MultiStepTransform,
    NextObservationDelta,
    NextStateReconstructor,
    PolicyAgeFilter,
    SuccessReward,
    TerminateTransform,
    Timer,
    gSDENoise,
)
from rl_amd.tensordict import TensorDict, TensorDictModule
from rl_amd.testing import ContinuousActionVecMockEnv, CountingEnv


def make_env():
    return ContinuousActionVecMockEnv(batch_size=[2], max_steps=10)


class TestTerminate:
    def test_predicate_ends_rollout(self):
        env = TransformedEnv(
            CountingEnv(max_steps=100, batch_size=[2]),
            TerminateTransform(lambda td: td["observation"] >= 3),
        )
        r = env.rollout(10)
        # random bool actions: the first env to count to 3 ends the rollout
        assert r.batch_size[-1] <= 10
        assert r[..., -1]["next", "done"].any()
        assert (r["observation"] < 3).all()

    def test_write_done_false(self):
        env = TransformedEnv(
            CountingEnv(max_steps=100, batch_size=[2]),
            TerminateTransform(lambda td: td["observation"] >= 3, write_done=False),
        )
        td = env.reset()
        for _ in range(3):
            td.set("action", torch.ones(2, 1, dtype=torch.bool))
            td = env.step(td)
            td = td["next"].exclude("reward")
        assert td["terminated"].all()


class TestGSDENoise:
    def test_primes_eps(self):
        env = TransformedEnv(make_env(), gSDENoise(state_dim=7, action_dim=7))
        td = env.reset()
        assert td["_eps_gSDE"].shape == (2, 7, 7)
        # random init (not all zeros)
        assert td["_eps_gSDE"].abs().sum() > 0


class TestActionScaling:
    def test_inverse_maps_to_bounds(self):
        env = TransformedEnv(make_env(), ActionScaling())
        spec = env.full_action_spec["action"]
        assert (spec.low == -1).all() and (spec.high == 1).all()
        td = env.reset()
        td.set("action", torch.ones(2, 5))
        td = env.step(td)  # must not error; action scaled into base bounds
        t = env.transform[0]
        a = torch.zeros(2, 5)
        assert torch.allclose(t._inv_apply_transform(a), t._loc.expand(2, 5))

    def test_forward_normalizes(self):
        t = ActionScaling(loc=torch.tensor(1.0), scale=torch.tensor(2.0))
        a = torch.tensor([3.0])
        assert torch.allclose(t._apply_transform(a), torch.tensor([1.0]))
        assert torch.allclose(t._inv_apply_transform(torch.tensor([1.0])), torch.tensor([3.0]))


class TestActionChunk:
    def test_chunks_and_pad(self):
        T, A, H = 5, 2, 3
        action = torch.arange(T).float().unsqueeze(-1).expand(T, A).clone()
        td = TensorDict({"action": action}, batch_size=[T])
        ActionChunkTransform(chunk_size=H)(td)
        chunk = td[("vla_action", "chunk")]
        assert chunk.shape == (T, H, A)
        assert torch.equal(chunk[0, :, 0], torch.tensor([0.0, 1.0, 2.0]))
        # last step: all future positions padded with the final action
        assert torch.equal(chunk[T - 1, :, 0], torch.tensor([4.0, 4.0, 4.0]))
        pad = td["action_is_pad"]
        assert not pad[0].any()
        assert pad[T - 1, 1:].all() and not pad[T - 1, 0]

    def test_done_boundary(self):
        T, H = 4, 3
        td = TensorDict(
            {
                "action": torch.arange(T).float().unsqueeze(-1),
                "next": {"done": torch.tensor([[False], [True], [False], [False]])},
            },
            batch_size=[T],
        )
        ActionChunkTransform(chunk_size=H)(td)
        pad = td["action_is_pad"]
        # step 0's window crosses the done at t=1 → positions 2.. padded
        assert not pad[0, 0] and not pad[0, 1] and pad[0, 2]
        # step 2 (new traj) unaffected until end-of-window
        assert not pad[2, 0] and not pad[2, 1]


class TestActionTokenizer:
    def test_encode_decode_roundtrip(self):
        tok = UniformActionTokenizer(256, -1.0, 1.0)
        t = ActionTokenizerTransform(tok)
        a = torch.rand(4, 3) * 2 - 1
        td = TensorDict({"action": a.clone()}, batch_size=[4])
        t(td)
        assert td["action_tokens"].dtype == torch.long
        t._inv_call(td)
        assert (td["action"] - a).abs().max() < 1.0 / 128

    def test_env_spec_rewrite(self):
        tok = UniformActionTokenizer(16, -1.0, 1.0)
        env = TransformedEnv(make_env(), ActionTokenizerTransform(tok))
        spec = env.full_action_spec["action"]
        from rl_amd.data.tensor_specs import Categorical

        assert isinstance(spec, Categorical)
        td = env.reset()
        td.set("action", torch.randint(0, 16, (2, 5)))
        env.step(td)  # decoded internally


class TestRewardTransforms:
    def test_linearise(self):
        t = LineariseRewards(in_keys=["reward"], weights=[1.0, 2.0])
        td = TensorDict({"reward": torch.tensor([[1.0, 3.0]])}, batch_size=[1])
        t._call(td)
        assert td["reward"].tolist() == [[7.0]]

    def test_success_reward(self):
        t = SuccessReward(scale=5.0)
        td = TensorDict({"success": torch.tensor([[True], [False]])}, batch_size=[2])
        t._call(td)
        assert td["reward"].squeeze(-1).tolist() == [5.0, 0.0]


class TestKeysShapes:
    def test_flatten_tensordict_rb(self):
        rb = TensorDictReplayBuffer(storage=LazyTensorStorage(100), batch_size=4)
        rb.append_transform(FlattenTensorDict())
        td = TensorDict({"x": torch.zeros(2, 5, 1)}, batch_size=[2, 5])
        rb.extend(td)
        assert len(rb) == 10

    def test_expand_as(self):
        t = ExpandAs(in_key="done", ref_key="reward")
        td = TensorDict(
            {"done": torch.tensor([True, False]), "reward": torch.zeros(2, 3)},
            batch_size=[2],
        )
        t._call(td)
        assert td["done"].shape == (2, 3)


class TestNextObservationDelta:
    def test_roundtrip_via_rb(self):
        rb = TensorDictReplayBuffer(storage=LazyTensorStorage(100), batch_size=8)
        rb.append_transform(NextObservationDelta(in_keys=["observation"]))
        obs = torch.randn(8, 4)
        nxt = obs + torch.randn(8, 4) * 0.1
        td = TensorDict({"observation": obs, "next": {"observation": nxt}}, batch_size=[8])
        rb.extend(td)
        s = rb.sample()
        assert ("next", "observation") in s.keys(True)
        # reconstruction within fp16 round-trip
        # (sample order is random: check against the value consistent with root)
        err = (s["next", "observation"] - s["observation"]).abs().max()
        assert err < 1.0  # deltas were small
        assert s["next", "observation"].dtype == torch.float32


class TestMultiStepTransform:
    def test_stream_insensitive(self):
        # one big extend == two half extends (after flush)
        def run(chunks):
            t = MultiStepTransform(n_steps=2, gamma=0.9)
            outs = [t._inv_call(c.clone()) for c in chunks]
            return [o for o in outs if o.batch_size[-1] > 0]

        T = 8
        reward = torch.arange(1.0, T + 1).reshape(1, T, 1)
        done = torch.zeros(1, T, 1, dtype=torch.bool)
        td = TensorDict(
            {"obs": torch.arange(T).float().reshape(1, T, 1), "next": {"reward": reward, "done": done, "obs": torch.arange(1, T + 1).float().reshape(1, T, 1)}},
            batch_size=[1, T],
        )
        whole = run([td])
        halves = run([td[:, :4], td[:, 4:]])
        from rl_amd.tensordict import cat as td_cat

        w = td_cat(whole, dim=1)
        h = td_cat(halves, dim=1)
        n = min(w.batch_size[1], h.batch_size[1])
        assert torch.allclose(w["next", "reward"][:, :n], h["next", "reward"][:, :n])

    def test_nstep_reward_value(self):
        T = 4
        reward = torch.ones(1, T, 1)
        done = torch.zeros(1, T, 1, dtype=torch.bool)
        td = TensorDict(
            {"next": {"reward": reward, "done": done, "obs": torch.zeros(1, T, 1)}},
            batch_size=[1, T],
        )
        t = MultiStepTransform(n_steps=2, gamma=0.5)
        out = t._inv_call(td)
        # emitted steps have full 2-step lookahead: r + 0.5 r = 1.5
        assert torch.allclose(out["next", "reward"], torch.full_like(out["next", "reward"], 1.5))


class TestNextStateReconstructor:
    def test_shift_and_nan(self):
        obs = torch.arange(6).float().unsqueeze(-1)
        traj = torch.tensor([0, 0, 0, 1, 1, 1])
        td = TensorDict(
            {"obs": obs, "collector": {"traj_ids": traj}}, batch_size=[6]
        )
        NextStateReconstructor(in_keys=["obs"])(td)
        nxt = td["next", "obs"].squeeze(-1)
        assert nxt[0] == 1 and nxt[1] == 2
        assert math.isnan(nxt[2].item())  # traj boundary
        assert nxt[3] == 4
        assert math.isnan(nxt[5].item())  # batch end


class TestPolicyAgeFilter:
    def test_filters_stale(self):
        version = {"v": 10}
        t = PolicyAgeFilter(lambda: version["v"], max_policy_lag=2)
        td = TensorDict(
            {"x": torch.arange(4).float(), "policy_version": torch.tensor([10, 9, 7, 3])},
            batch_size=[4],
        )
        out = t.forward(td)
        assert out.batch_size[0] == 2  # versions 10, 9 kept (lag ≤ 2); 7, 3 dropped

    def test_rb_extend_path(self):
        rb = TensorDictReplayBuffer(storage=LazyTensorStorage(100), batch_size=2)
        rb.append_transform(PolicyAgeFilter(5, max_policy_lag=1))
        td = TensorDict(
            {"x": torch.zeros(3), "policy_version": torch.tensor([5, 4, 1])},
            batch_size=[3],
        )
        rb.extend(td)
        assert len(rb) == 2


class TestModuleTimerMean:
    def test_module_transform(self):
        mod = TensorDictModule(
            lambda x: x * 2, in_keys=["observation"], out_keys=["observation"]
        )
        env = TransformedEnv(CountingEnv(max_steps=5, batch_size=[2]), ModuleTransform(mod))
        td = env.reset()
        assert (td["observation"] == 0).all()
        td.set("action", torch.ones(2, 1, dtype=torch.bool))
        td = env.step(td)
        assert (td["next", "observation"] == 2).all()  # count 1 doubled

    def test_timer_writes_keys(self):
        env = TransformedEnv(CountingEnv(max_steps=5, batch_size=[2]), Timer())
        td = env.reset()
        td.set("action", torch.ones(2, 1, dtype=torch.bool))
        td = env.step(td)
        assert td["next", "time_step"].shape == (2,)
        assert (td["next", "time_step"] >= 0).all()

    def test_mean_action_selector(self):
        t = MeanActionSelector()
        td = TensorDict({"observation": torch.randn(2, 3)}, batch_size=[2])
        obs = td["observation"].clone()
        t._call(td)
        assert torch.equal(td["observation", "mean"], obs)
        assert (td["observation", "var"] == 0).all()
        act = TensorDict(
            {"action": TensorDict({"mean": torch.ones(2, 3), "var": torch.zeros(2, 3, 3)}, batch_size=[2])},
            batch_size=[2],
        )
        t._inv_call(act)
        assert torch.equal(act["action"], torch.ones(2, 3))


class TestBatch4:
    def test_macro_primitive_expand(self):
        from rl_amd.envs.transforms import MacroPrimitiveTransform, TargetMacroAction
        from rl_amd.tensordict import TensorDict

        t = MacroPrimitiveTransform()
        td = TensorDict({}, batch_size=[])
        td.set_non_tensor("macro_action", TargetMacroAction.move(torch.ones(3), steps=4))
        t._inv_call(td)
        seq = td["action"]
        assert seq.shape == (4, 3)
        assert torch.allclose(seq[-1], torch.ones(3))
        assert torch.allclose(seq[0], torch.full((3,), 0.25))  # linear ramp from 0
        # WAIT holds the last action
        from rl_amd.envs.transforms import MacroAction

        td2 = TensorDict({}, batch_size=[])
        td2.set_non_tensor("macro_action", MacroAction.wait(steps=2))
        t._inv_call(td2)
        assert torch.allclose(td2["action"], torch.ones(2, 3))

    def test_running_mean_std(self):
        from rl_amd.envs.transforms import RunningMeanStd

        torch.manual_seed(0)
        rms = RunningMeanStd(shape=(3,))
        data = torch.randn(1000, 3) * 2 + 5
        for chunk in data.split(100):
            rms.update(chunk)
        assert torch.allclose(rms.mean, data.mean(0), atol=0.05)
        normed = rms.normalize(data)
        assert normed.mean().abs() < 0.05 and (normed.std(0) - 1).abs().max() < 0.1

    def test_gated_encoders(self):
        import importlib.util

        from rl_amd.envs.transforms import R3MTransform, RayModuleTransform

        with pytest.raises((ImportError, NotImplementedError)):
            R3MTransform("resnet18")
        if importlib.util.find_spec("ray") is None:
            with pytest.raises(ImportError, match="ray"):
                RayModuleTransform(lambda: None)

    def test_aliases(self):
        from rl_amd.envs import transforms as T

        assert T.Hash is T.HashTransform
        assert T.Tokenizer is T.TokenizerTransform
        assert T.RandomTruncationTransform is T.RandomTruncation
