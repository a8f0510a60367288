"""Offline-dataset pipeline tests against local fixtures (VERDICT r1
item 8; reference torchrl/data/datasets/): converters source→memmap,
cache resolution, streaming storage."""
import gzip
import os

import numpy as np
import pytest
import torch

from rl_amd.data import (
    AtariDQNExperienceReplay,
    D4RLExperienceReplay,
    MinariExperienceReplay,
    OpenXExperienceReplay,
    StreamingEpisodeStorage,
    convert_atari_shards,
    convert_d4rl_hdf5,
    convert_minari_hdf5,
)


def _write_d4rl_npz(path, n=50, obs_dim=4, act_dim=2):
    rng = np.random.default_rng(0)
    np.savez(
        path,
        observations=rng.standard_normal((n, obs_dim)).astype(np.float32),
        actions=rng.standard_normal((n, act_dim)).astype(np.float32),
        rewards=rng.standard_normal(n).astype(np.float32),
        terminals=(rng.random(n) < 0.1),
        timeouts=(rng.random(n) < 0.05),
    )


class TestD4RL:
    def test_convert_and_load(self, tmp_path):
        src = str(tmp_path / "halfcheetah-medium-v2.npz")
        _write_d4rl_npz(src)
        rb = D4RLExperienceReplay(
            "halfcheetah-medium-v2", root=str(tmp_path), batch_size=16
        )
        assert len(rb) == 50
        batch = rb.sample()
        assert batch.get("observation").shape == (16, 4)
        assert ("next", "done") in batch
        # memmap cache now exists → second construction skips conversion
        assert (tmp_path / "halfcheetah-medium-v2").is_dir()
        os.remove(src)
        rb2 = D4RLExperienceReplay(
            "halfcheetah-medium-v2", root=str(tmp_path), batch_size=8
        )
        assert len(rb2) == 50

    def test_missing_without_download_raises(self, tmp_path):
        with pytest.raises(FileNotFoundError):
            D4RLExperienceReplay("nope-v0", root=str(tmp_path))

    def test_immutable(self, tmp_path):
        src = str(tmp_path / "d-v2.npz")
        _write_d4rl_npz(src)
        rb = D4RLExperienceReplay("d-v2", root=str(tmp_path), batch_size=4)
        with pytest.raises(RuntimeError):
            rb.extend(rb.sample())


class TestMinari:
    def test_convert_episodes(self, tmp_path):
        srcdir = tmp_path / "door-human-v1-src"
        srcdir.mkdir()
        rng = np.random.default_rng(1)
        arrays = {}
        total = 0
        for ep in range(3):
            T = 5 + ep
            arrays[f"episode_{ep}/observations"] = rng.standard_normal(
                (T + 1, 3)
            ).astype(np.float32)
            arrays[f"episode_{ep}/actions"] = rng.standard_normal((T, 2)).astype(
                np.float32
            )
            arrays[f"episode_{ep}/rewards"] = rng.standard_normal(T).astype(np.float32)
            term = np.zeros(T, bool)
            term[-1] = ep % 2 == 0
            trunc = np.zeros(T, bool)
            trunc[-1] = not term[-1]
            arrays[f"episode_{ep}/terminations"] = term
            arrays[f"episode_{ep}/truncations"] = trunc
            total += T
        np.savez(str(srcdir / "main_data.npz"), **arrays)
        rb = MinariExperienceReplay("door-human-v1", root=str(tmp_path), batch_size=8)
        assert len(rb) == total
        b = rb.sample()
        assert b.get("action").shape == (8, 2)
        # episode boundaries marked done
        data = rb._storage[0 : len(rb)]
        dones = data.get(("next", "done")).reshape(-1)
        assert int(dones.sum()) == 3

    def test_next_obs_within_episode(self, tmp_path):
        srcdir = tmp_path / "m-v1-src"
        srcdir.mkdir()
        obs = np.arange(6, dtype=np.float32).reshape(6, 1)  # T=5
        np.savez(
            str(srcdir / "main_data.npz"),
            **{
                "episode_0/observations": obs,
                "episode_0/actions": np.zeros((5, 1), np.float32),
                "episode_0/rewards": np.zeros(5, np.float32),
                "episode_0/terminations": np.array([0, 0, 0, 0, 1], bool),
                "episode_0/truncations": np.zeros(5, bool),
            },
        )
        rb = MinariExperienceReplay("m-v1", root=str(tmp_path), batch_size=5)
        data = rb._storage[0:5]
        assert torch.equal(
            data.get(("next", "observation")).reshape(-1),
            torch.arange(1, 6, dtype=torch.float32),
        )


class TestAtari:
    def test_convert_shards(self, tmp_path):
        shards = tmp_path / "Pong-v5-shards"
        shards.mkdir()
        rng = np.random.default_rng(2)
        for ck in range(2):
            for field, arr in {
                "observation": rng.integers(0, 255, (10, 4, 4), dtype=np.uint8),
                "action": rng.integers(0, 6, (10,), dtype=np.int64),
                "reward": rng.standard_normal(10).astype(np.float32),
                "terminal": (rng.random(10) < 0.1),
            }.items():
                with gzip.open(
                    str(shards / f"$store$_{field}_ckpt.{ck}.npy.gz"), "wb"
                ) as f:
                    np.save(f, arr)
        rb = AtariDQNExperienceReplay("Pong-v5", root=str(tmp_path), batch_size=4)
        assert len(rb) == 20
        b = rb.sample()
        assert b.get("observation").shape == (4, 4, 4)


class TestStreaming:
    def _write_episodes(self, d, n_eps=4):
        rng = np.random.default_rng(3)
        lengths = []
        for ep in range(n_eps):
            T = 3 + ep
            np.savez(
                str(d / f"episode_{ep:05d}.npz"),
                observations=rng.standard_normal((T + 1, 2)).astype(np.float32),
                actions=rng.standard_normal((T, 1)).astype(np.float32),
                rewards=np.full(T, float(ep), np.float32),
                terminals=np.zeros(T, bool),
            )
            lengths.append(T)
        return lengths

    def test_streaming_storage_indexing(self, tmp_path):
        lengths = self._write_episodes(tmp_path)
        st = StreamingEpisodeStorage(str(tmp_path), cache_episodes=2)
        assert len(st) == sum(lengths)
        # transition 3 is episode 1 step 0 (episode 0 has 3 steps)
        row = st.get(3)
        assert float(row.get(("next", "reward"))) == 1.0
        batch = st.get(torch.tensor([0, 3, 7]))
        assert batch.batch_size[0] == 3
        assert [float(r) for r in batch.get(("next", "reward")).reshape(-1)] == [
            0.0, 1.0, 2.0,
        ]

    def test_openx_replay(self, tmp_path):
        d = tmp_path / "openx-bridge"
        d.mkdir()
        self._write_episodes(d)
        rb = OpenXExperienceReplay("openx-bridge", root=str(tmp_path), batch_size=6)
        b = rb.sample()
        assert b.get("observation").shape == (6, 2)
        with pytest.raises(RuntimeError):
            rb.extend(b)
