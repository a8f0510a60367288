"""Replay stack tests: storages × samplers × writers, segment trees, PER."""
import numpy as np
import pytest
import torch

from rl_amd.data import (
    LazyMemmapStorage,
    LazyTensorStorage,
    ListStorage,
    MinSegmentTree,
    PrioritizedSampler,
    RandomSampler,
    ReplayBuffer,
    SamplerWithoutReplacement,
    SliceSampler,
    SliceSamplerWithoutReplacement,
    SumSegmentTree,
    TensorDictMaxValueWriter,
    TensorDictPrioritizedReplayBuffer,
    TensorDictReplayBuffer,
    TensorStorage,
)
from rl_amd.tensordict import TensorDict, stack


class TestSegmentTree:
    def test_sum_tree_matches_numpy(self):
        torch.manual_seed(0)
        st = SumSegmentTree(1000)
        ref = np.zeros(1000)
        for _ in range(20):
            idx = torch.randint(0, 1000, (64,))
            val = torch.rand(64).double()
            st.update(idx, val)
            for i, v in zip(idx.tolist(), val.tolist()):
                ref[i] = v
        assert st.query(0, 1000).item() == pytest.approx(ref.sum(), rel=1e-9)
        assert st.query(10, 500).item() == pytest.approx(ref[10:500].sum(), rel=1e-6)

    def test_sum_tree_duplicate_last_writer_wins(self):
        st = SumSegmentTree(8)
        st.update(torch.tensor([3, 3, 3]), torch.tensor([1.0, 2.0, 5.0]))
        assert st[3].item() == 5.0
        assert st.query(0, 8).item() == 5.0

    def test_scan_lower_bound_exact(self):
        st = SumSegmentTree(64)
        vals = torch.rand(64).double()
        st.update(torch.arange(64), vals)
        cs = np.cumsum(vals.numpy())
        mass = torch.rand(128).double() * cs[-1]
        found = st.scan_lower_bound(mass)
        expected = np.searchsorted(cs, mass.numpy(), side="right")
        assert (found.numpy() == expected).all()

    def test_min_tree(self):
        mt = MinSegmentTree(32)
        mt.update(torch.arange(10), torch.arange(10).double() + 3)
        assert mt.query(0, 10).item() == 3.0
        assert mt.query(5, 10).item() == 8.0

    @pytest.mark.gpu
    def test_sum_tree_gpu(self):
        st = SumSegmentTree(1024, device="cuda")
        idx = torch.randint(0, 1024, (256,), device="cuda")
        val = torch.rand(256, device="cuda").double()
        st.update(idx, val)
        mass = torch.rand(512, device="cuda").double() * st.query(0, 1024)
        found = st.scan_lower_bound(mass)
        assert found.device.type == "cuda"
        assert (found >= 0).all() and (found < 1024).all()


@pytest.mark.parametrize(
    "storage_cls", [ListStorage, LazyTensorStorage, LazyMemmapStorage]
)
class TestStorages:
    def test_extend_sample(self, storage_cls, tmp_path):
        kwargs = {"scratch_dir": str(tmp_path)} if storage_cls is LazyMemmapStorage else {}
        rb = TensorDictReplayBuffer(storage=storage_cls(100, **kwargs), batch_size=8)
        for i in range(5):
            rb.extend(
                TensorDict({"obs": torch.full((10, 3), float(i))}, batch_size=[10])
            )
        assert len(rb) == 50
        s = rb.sample()
        assert s["obs"].shape == (8, 3)
        assert "index" in s

    def test_overwrite_circular(self, storage_cls, tmp_path):
        kwargs = {"scratch_dir": str(tmp_path)} if storage_cls is LazyMemmapStorage else {}
        rb = TensorDictReplayBuffer(storage=storage_cls(10, **kwargs), batch_size=4)
        rb.extend(TensorDict({"x": torch.arange(15).float().unsqueeze(-1)}, batch_size=[15]))
        assert len(rb) == 10
        vals = rb.storage.get(torch.arange(10))
        if isinstance(vals, TensorDict):
            x = vals["x"].flatten()
        else:
            x = torch.cat([v["x"] for v in vals])
        # slots 0..4 rewritten with items 10..14
        assert x[0].item() == 10.0


class TestSamplers:
    def test_without_replacement_epoch(self):
        rb = ReplayBuffer(
            storage=LazyTensorStorage(20),
            sampler=SamplerWithoutReplacement(),
            batch_size=5,
        )
        rb.extend(TensorDict({"x": torch.arange(20).float().unsqueeze(-1)}, batch_size=[20]))
        seen = torch.cat([rb.sample()["x"].flatten() for _ in range(4)])
        assert sorted(seen.tolist()) == list(range(20))

    def test_prioritized_bias(self):
        torch.manual_seed(0)
        sampler = PrioritizedSampler(64, alpha=1.0, beta=1.0)
        storage = LazyTensorStorage(64)
        rb = ReplayBuffer(storage=storage, sampler=sampler, batch_size=256)
        rb.extend(TensorDict({"x": torch.arange(64).float()}, batch_size=[64]))
        # put all priority mass on index 7
        prios = torch.full((64,), 1e-6)
        prios[7] = 100.0
        rb.update_priority(torch.arange(64), prios)
        s = rb.sample()
        frac = (s["x"] == 7).float().mean().item()
        assert frac > 0.95

    def test_prioritized_weights(self):
        sampler = PrioritizedSampler(8, alpha=1.0, beta=1.0)
        storage = LazyTensorStorage(8)
        rb = ReplayBuffer(storage=storage, sampler=sampler, batch_size=16)
        rb.extend(TensorDict({"x": torch.arange(4).float()}, batch_size=[4]))
        rb.update_priority(torch.arange(4), torch.tensor([1.0, 2.0, 3.0, 4.0]))
        s, info = rb.sample(return_info=True)
        w = info["_weight"]
        assert w.min() > 0
        # weight of the highest-priority item is the smallest
        assert w[s["x"] == 3].max() <= w[s["x"] == 0].min() + 1e-5

    def test_per_sampling_distribution(self):
        torch.manual_seed(0)
        sampler = PrioritizedSampler(16, alpha=1.0, beta=0.0)
        storage = LazyTensorStorage(16)
        rb = ReplayBuffer(storage=storage, sampler=sampler, batch_size=4096)
        rb.extend(TensorDict({"x": torch.arange(4).float()}, batch_size=[4]))
        rb.update_priority(torch.arange(4), torch.tensor([1.0, 1.0, 2.0, 4.0]))
        s = rb.sample()
        counts = torch.bincount(s["x"].long(), minlength=4).float()
        probs = counts / counts.sum()
        assert probs[3].item() == pytest.approx(0.5, abs=0.05)
        assert probs[2].item() == pytest.approx(0.25, abs=0.05)

    def test_slice_sampler_within_traj(self):
        steps = []
        for traj in range(6):
            T = 8 + traj
            for t in range(T):
                steps.append(
                    TensorDict(
                        {
                            "obs": torch.tensor([float(traj)]),
                            "collector": {"traj_ids": torch.tensor(traj)},
                        },
                        batch_size=[],
                    )
                )
        data = stack(steps, 0)
        rb = ReplayBuffer(
            storage=LazyTensorStorage(200), sampler=SliceSampler(slice_len=5), batch_size=20
        )
        rb.extend(data)
        s, info = rb.sample(return_info=True)
        obs = s["obs"].reshape(info["num_slices"], info["slice_len"])
        assert (obs == obs[:, :1]).all()

    def test_slice_without_replacement(self):
        steps = []
        for traj in range(4):
            for t in range(10):
                steps.append(
                    TensorDict(
                        {
                            "obs": torch.tensor([float(traj)]),
                            "collector": {"traj_ids": torch.tensor(traj)},
                        },
                        batch_size=[],
                    )
                )
        data = stack(steps, 0)
        rb = ReplayBuffer(
            storage=LazyTensorStorage(100),
            sampler=SliceSamplerWithoutReplacement(slice_len=5),
            batch_size=10,
        )
        rb.extend(data)
        trajs = []
        for _ in range(2):
            s, info = rb.sample(return_info=True)
            obs = s["obs"].reshape(info["num_slices"], info["slice_len"])
            trajs.extend(obs[:, 0].tolist())
        assert len(set(trajs)) == 4


class TestWriters:
    def test_max_value_writer(self):
        rb = ReplayBuffer(
            storage=LazyTensorStorage(3),
            writer=TensorDictMaxValueWriter(rank_key="score"),
            batch_size=3,
        )
        for score in [1.0, 5.0, 3.0, 0.5, 9.0]:
            rb.add(TensorDict({"score": torch.tensor([score])}, batch_size=[]))
        kept = rb.storage.get(torch.arange(3))["score"].flatten()
        assert sorted(kept.tolist()) == [3.0, 5.0, 9.0]


class TestCheckpointing:
    def test_rb_state_dict_roundtrip(self):
        rb = TensorDictReplayBuffer(storage=LazyTensorStorage(20), batch_size=4)
        rb.extend(TensorDict({"x": torch.randn(10, 2)}, batch_size=[10]))
        sd = rb.state_dict()
        rb2 = TensorDictReplayBuffer(storage=LazyTensorStorage(20), batch_size=4)
        rb2.load_state_dict(sd)
        assert len(rb2) == 10
        assert torch.allclose(
            rb2.storage.get(torch.arange(10))["x"], rb.storage.get(torch.arange(10))["x"]
        )

    def test_rb_dumps_loads(self, tmp_path):
        rb = TensorDictPrioritizedReplayBuffer(
            storage=LazyTensorStorage(20), batch_size=4
        )
        rb.extend(TensorDict({"x": torch.randn(10, 2)}, batch_size=[10]))
        rb.update_priority(torch.arange(10), torch.rand(10) + 0.1)
        rb.dumps(str(tmp_path / "rb"))
        rb2 = TensorDictPrioritizedReplayBuffer(
            storage=LazyTensorStorage(20), batch_size=4
        )
        rb2.loads(str(tmp_path / "rb"))
        assert len(rb2) == 10
        s = rb2.sample()
        assert s["x"].shape == (4, 2)


class TestCheckpointers:
    def _ted_buffer(self, n=20):
        from rl_amd.data import LazyTensorStorage, TensorDictReplayBuffer
        from rl_amd.tensordict import TensorDict

        torch.manual_seed(0)
        rb = TensorDictReplayBuffer(storage=LazyTensorStorage(50), batch_size=4)
        obs = torch.randn(n, 3)
        nxt_obs = torch.randn(n, 3)
        done = torch.zeros(n, 1, dtype=torch.bool)
        done[n // 2] = True
        done[n - 1] = True
        # TED consistency: next_obs[i] == obs[i+1] inside trajectories
        nxt_obs[:-1] = obs[1:]
        rb.extend(TensorDict(
            {"obs": obs, "next": {"obs": nxt_obs, "done": done}}, batch_size=[n]
        ))
        return rb, obs, nxt_obs

    def test_tensor_storage_checkpointer(self, tmp_path):
        from rl_amd.data.replay_buffers import TensorStorageCheckpointer

        rb, obs, nxt = self._ted_buffer()
        cp = TensorStorageCheckpointer()
        cp.dumps(rb._storage, str(tmp_path))
        rb2, _, _ = self._ted_buffer(5)
        cp.loads(rb2._storage, str(tmp_path))
        assert len(rb2._storage) == 20
        assert torch.equal(rb2._storage._storage["obs"][:20], obs)

    def test_flat_checkpointer_roundtrip(self, tmp_path):
        from rl_amd.data.replay_buffers import FlatStorageCheckpointer

        rb, obs, nxt = self._ted_buffer()
        cp = FlatStorageCheckpointer()
        cp.dumps(rb._storage, str(tmp_path))
        # the dedupe must shrink what's stored for duplicated keys
        sd = torch.load(str(tmp_path) + "/flat_storage.pt", weights_only=False)
        assert ("next", "obs") not in sd["compact"].keys(True, True)
        rb2, _, _ = self._ted_buffer(5)
        cp.loads(rb2._storage, str(tmp_path))
        rebuilt = rb2._storage._storage
        assert torch.equal(rebuilt["obs"][:20], obs)
        assert torch.allclose(rebuilt["next", "obs"][:20], nxt)

    def test_h5_gated(self):
        import importlib.util

        from rl_amd.data.replay_buffers import H5StorageCheckpointer

        if importlib.util.find_spec("h5py") is None:
            with pytest.raises(ImportError, match="h5py"):
                H5StorageCheckpointer()

    def test_scheduler_list(self):
        from rl_amd.data.replay_buffers import LambdaScheduler, SchedulerList

        class Obj:
            a = 1.0
            b = 2.0

        o = Obj()
        sl = SchedulerList([
            LambdaScheduler(o, "a", lambda t: 1.0 / (1 + t)),
            LambdaScheduler(o, "b", lambda t: 2.0 + t),
        ])
        sl.step()
        assert o.a == 0.5 and o.b == 3.0
