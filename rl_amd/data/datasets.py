"""Offline-dataset experience replays: acquisition → conversion → memmap.

Reference: pytorch/rl torchrl/data/datasets/ (BaseDatasetExperienceReplay
common.py:21, D4RL d4rl.py:30, Minari minari_data.py:75, AtariDQN
atari_dqn.py:36 with custom storage :742, OpenX openx.py:36 with
streaming storage :548).

Pipeline (matching the reference's): ``download=True`` fetches the
published source files (D4RL's public HTTP bucket; Minari via the
``minari`` hub client), a CONVERTER turns them into the rl_amd
TensorDict memmap layout under ``root/<dataset_id>`` once, and later
constructions load straight from that cache.  The converters are plain
functions (``convert_d4rl_hdf5``, ``convert_minari_hdf5``,
``convert_atari_shards``) so they are testable against local fixtures —
which is exactly how tests/test_datasets.py exercises them (this build
has no network egress; the download step then raises the underlying
connection error like the reference does offline).

OpenX-scale corpora do not fit memory: ``StreamingEpisodeStorage``
indexes per-episode files and materializes transitions on demand with
an LRU episode cache (reference openx.py:548 streaming storage).
"""
from __future__ import annotations

import gzip
import json
import os
from collections import OrderedDict
from typing import Callable, Dict, List, Optional, Sequence

import numpy as np
import torch

from ..tensordict import TensorDict, TensorDictBase
from .replay_buffers.buffers import TensorDictReplayBuffer
from .replay_buffers.samplers import Sampler
from .replay_buffers.storages import LazyMemmapStorage, LazyTensorStorage, Storage
from .replay_buffers.writers import ImmutableDatasetWriter

__all__ = [
    "BaseDatasetExperienceReplay",
    "LocalHDF5ExperienceReplay",
    "LocalMemmapExperienceReplay",
    "D4RLExperienceReplay",
    "MinariExperienceReplay",
    "AtariDQNExperienceReplay",
    "OpenXExperienceReplay",
    "StreamingEpisodeStorage",
    "convert_d4rl_hdf5",
    "convert_minari_hdf5",
    "convert_atari_shards",
    "download_file",
]

# The published D4RL file server (reference d4rl_infos.py URL scheme).
D4RL_BASE_URL = "http://rail.eecs.berkeley.edu/datasets/offline_rl"


def _default_root() -> str:
    return os.environ.get(
        "RL_AMD_DATA", os.path.join(os.path.expanduser("~"), ".cache", "rl_amd", "datasets")
    )


def download_file(url: str, dest: str, chunk_bytes: int = 1 << 20) -> str:
    """Stream ``url`` to ``dest`` (atomic rename; resumable by re-run)."""
    import requests

    os.makedirs(os.path.dirname(dest) or ".", exist_ok=True)
    tmp = dest + ".part"
    with requests.get(url, stream=True, timeout=60) as r:
        r.raise_for_status()
        with open(tmp, "wb") as f:
            for chunk in r.iter_content(chunk_size=chunk_bytes):
                f.write(chunk)
    os.replace(tmp, dest)
    return dest


def _transitions_from_arrays(
    observations: np.ndarray,
    actions: np.ndarray,
    rewards: np.ndarray,
    terminals: np.ndarray,
    timeouts: Optional[np.ndarray] = None,
    next_observations: Optional[np.ndarray] = None,
    device=None,
) -> TensorDictBase:
    """(N,·) arrays → [N] transition TensorDict (shared by all loaders)."""
    N = observations.shape[0]
    obs = torch.as_tensor(observations)
    act = torch.as_tensor(actions)
    rew = torch.as_tensor(rewards).reshape(N, 1).float()
    term = torch.as_tensor(terminals).reshape(N, 1).bool()
    trunc = (
        torch.as_tensor(timeouts).reshape(N, 1).bool()
        if timeouts is not None
        else torch.zeros(N, 1, dtype=torch.bool)
    )
    if next_observations is not None:
        next_obs = torch.as_tensor(next_observations)
    else:
        next_obs = torch.cat([obs[1:], obs[-1:]], 0)
    return TensorDict(
        {
            "observation": obs,
            "action": act,
            "next": {
                "observation": next_obs,
                "reward": rew,
                "terminated": term,
                "truncated": trunc,
                "done": term | trunc,
            },
        },
        batch_size=[N],
        device=device,
    )


# --------------------------------------------------------------------- #
# Converters (source format → TensorDict memmap directory)
# --------------------------------------------------------------------- #
def _open_source(path: str):
    """Open a dataset source file: ``.npz`` via numpy (always
    available), ``.hdf5`` via h5py (import-gated) — both expose
    ``f[key]`` / ``key in f``."""
    if path.endswith(".npz"):
        return np.load(path)
    try:
        import h5py
    except ImportError as e:
        raise ImportError(
            f"{path}: reading HDF5 requires h5py; convert to .npz or "
            "install h5py"
        ) from e
    return h5py.File(path, "r")


def convert_d4rl_hdf5(src_path: str, out_dir: str) -> str:
    """D4RL flat layout (observations/actions/rewards/terminals
    [/timeouts][/next_observations]) → memmap cache (reference d4rl.py
    download→_process path).  Accepts .hdf5 or .npz sources."""
    f = _open_source(src_path)
    try:
        td = _transitions_from_arrays(
            f["observations"][:],
            f["actions"][:],
            f["rewards"][:],
            f["terminals"][:],
            f["timeouts"][:] if "timeouts" in f else None,
            f["next_observations"][:] if "next_observations" in f else None,
        )
    finally:
        if hasattr(f, "close"):
            f.close()
    td.memmap_(out_dir)
    return out_dir


def convert_minari_hdf5(src_path: str, out_dir: str) -> str:
    """Minari ``main_data.hdf5`` (episode_N groups with per-episode
    observations [T+1,·], actions/rewards/terminations/truncations [T,·])
    → flat transition memmap (reference minari_data.py:75 _process).
    Accepts .hdf5 (groups) or .npz (keys ``episode_N/<field>``)."""
    f = _open_source(src_path)
    try:
        if src_path.endswith(".npz"):
            names = sorted(
                {k.split("/")[0] for k in f.files if k.startswith("episode_")},
                key=lambda k: int(k.split("_")[1]),
            )
            groups = [
                {field: f[f"{name}/{field}"]
                 for field in ("observations", "actions", "rewards",
                               "terminations", "truncations")}
                for name in names
            ]
        else:
            names = sorted(
                (k for k in f.keys() if k.startswith("episode_")),
                key=lambda k: int(k.split("_")[1]),
            )
            groups = [
                {field: np.asarray(f[name][field])
                 for field in ("observations", "actions", "rewards",
                               "terminations", "truncations")}
                for name in names
            ]
        parts: List[TensorDictBase] = []
        for g in groups:
            obs = g["observations"]
            act = g["actions"]
            T = act.shape[0]
            parts.append(
                _transitions_from_arrays(
                    obs[:T], act, g["rewards"], g["terminations"],
                    g["truncations"], next_observations=obs[1 : T + 1],
                )
            )
    finally:
        if hasattr(f, "close"):
            f.close()
    from ..tensordict import cat as td_cat

    td = td_cat(parts, 0)
    td.memmap_(out_dir)
    return out_dir


def convert_atari_shards(shard_dir: str, out_dir: str, max_transitions: Optional[int] = None) -> str:
    """Atari-DQN replay shards (``$store$_<field>_ckpt.N.npy.gz`` numpy
    archives: observation/action/reward/terminal) → memmap cache
    (reference atari_dqn.py:36 download/_process path)."""

    def _load(field: str) -> np.ndarray:
        matches = sorted(
            f for f in os.listdir(shard_dir) if field in f and f.endswith(".npy.gz")
        )
        arrays = []
        for fname in matches:
            with gzip.open(os.path.join(shard_dir, fname), "rb") as f:
                arrays.append(np.load(f, allow_pickle=False))
        if not arrays:
            raise FileNotFoundError(f"no {field!r} shards under {shard_dir}")
        return np.concatenate(arrays, 0)

    obs = _load("observation")
    act = _load("action")
    rew = _load("reward")
    term = _load("terminal")
    if max_transitions:
        obs, act, rew, term = (
            x[:max_transitions] for x in (obs, act, rew, term)
        )
    td = _transitions_from_arrays(obs, act, rew, term)
    td.memmap_(out_dir)
    return out_dir


# --------------------------------------------------------------------- #
# Streaming storage (OpenX-scale corpora)
# --------------------------------------------------------------------- #
class StreamingEpisodeStorage(Storage):
    """Index-addressed storage over per-episode ``.npz`` files,
    materialized on demand (reference openx.py:548 streaming storage).

    Each episode file holds arrays ``observations [T(+1),·]``,
    ``actions [T,·]``, ``rewards [T]``, ``terminals [T]``.  The flat
    transition index maps to (episode, step) through cumulative episode
    lengths; parsed episodes live in an LRU cache so slice-style access
    patterns (SliceSampler) stay cheap without the corpus ever being
    resident."""

    def __init__(self, episode_dir: str, cache_episodes: int = 16):
        super().__init__(max_size=0)
        files = sorted(
            os.path.join(episode_dir, f)
            for f in os.listdir(episode_dir)
            if f.endswith(".npz")
        )
        if not files:
            raise FileNotFoundError(f"no .npz episodes under {episode_dir}")
        self.files = files
        self._lengths = []
        for path in files:
            with np.load(path) as z:
                self._lengths.append(int(z["actions"].shape[0]))
        self._cum = np.cumsum([0] + self._lengths)
        self._cache: "OrderedDict[int, TensorDictBase]" = OrderedDict()
        self.cache_episodes = cache_episodes
        self.max_size = int(self._cum[-1])

    def __len__(self) -> int:
        return self.max_size

    def _episode(self, ep: int) -> TensorDictBase:
        td = self._cache.get(ep)
        if td is None:
            with np.load(self.files[ep]) as z:
                obs = z["observations"]
                act = z["actions"]
                T = act.shape[0]
                nxt = (
                    obs[1 : T + 1]
                    if obs.shape[0] > T
                    else np.concatenate([obs[1:], obs[-1:]], 0)
                )
                td = _transitions_from_arrays(
                    obs[:T], act, z["rewards"], z["terminals"],
                    next_observations=nxt,
                )
            self._cache[ep] = td
            if len(self._cache) > self.cache_episodes:
                self._cache.popitem(last=False)
        else:
            self._cache.move_to_end(ep)
        return td

    def get(self, index):
        if isinstance(index, int):
            index = torch.tensor([index])
            single = True
        else:
            single = False
            index = torch.as_tensor(index).reshape(-1)
        eps = np.searchsorted(self._cum, index.numpy(), side="right") - 1
        out = []
        for i, ep in zip(index.tolist(), eps.tolist()):
            out.append(self._episode(ep)[i - int(self._cum[ep])])
        from ..tensordict import stack as td_stack

        batch = td_stack(out, 0)
        return batch[0] if single else batch

    def set(self, index, data):
        raise RuntimeError("StreamingEpisodeStorage is read-only")

    def state_dict(self):
        return {"files": self.files}

    def load_state_dict(self, sd):
        pass


# --------------------------------------------------------------------- #
# Dataset replay buffers
# --------------------------------------------------------------------- #
class BaseDatasetExperienceReplay(TensorDictReplayBuffer):
    """Dataset-backed immutable replay buffer (reference common.py:21)."""

    def __init__(self, *, storage, sampler: Optional[Sampler] = None, batch_size: Optional[int] = None, transform=None, **kwargs):
        super().__init__(
            storage=storage,
            sampler=sampler,
            writer=ImmutableDatasetWriter(),
            batch_size=batch_size,
            transform=transform,
            **kwargs,
        )

    _transitions_from_arrays = staticmethod(_transitions_from_arrays)

    @classmethod
    def _fill(cls, storage, td: TensorDictBase):
        storage.set(torch.arange(td.batch_size[0]), td)
        return storage


class LocalHDF5ExperienceReplay(BaseDatasetExperienceReplay):
    """Load a D4RL-layout HDF5 file (observations/actions/rewards/
    terminals[/timeouts][/next_observations] datasets)."""

    def __init__(self, path: str, *, batch_size: int = 256, device=None, sampler=None, in_memory: bool = True, **kwargs):
        f = _open_source(path)
        try:
            td = _transitions_from_arrays(
                f["observations"][:],
                f["actions"][:],
                f["rewards"][:],
                f["terminals"][:],
                f["timeouts"][:] if "timeouts" in f else None,
                f["next_observations"][:] if "next_observations" in f else None,
                device=device,
            )
        finally:
            if hasattr(f, "close"):
                f.close()
        N = td.batch_size[0]
        storage = (
            LazyTensorStorage(N, device=device) if in_memory else LazyMemmapStorage(N)
        )
        self._fill(storage, td)
        super().__init__(storage=storage, sampler=sampler, batch_size=batch_size, **kwargs)


class LocalMemmapExperienceReplay(BaseDatasetExperienceReplay):
    """Load a TensorDict memmap directory as an immutable dataset."""

    def __init__(self, path: str, *, batch_size: int = 256, device=None, sampler=None, **kwargs):
        td = TensorDict.load_memmap(path)
        N = td.batch_size[0]
        storage = LazyTensorStorage(N, device=device)
        self._fill(storage, td.to(device) if device else td)
        super().__init__(storage=storage, sampler=sampler, batch_size=batch_size, **kwargs)


class D4RLExperienceReplay(LocalMemmapExperienceReplay):
    """D4RL datasets (reference d4rl.py:30).

    Resolution order under ``root`` (default ``$RL_AMD_DATA``):
    memmap cache ``<dataset_id>/`` → source ``<dataset_id>.hdf5``
    (converted once) → ``download=True`` fetches from the public D4RL
    file server, converts, caches."""

    def __init__(self, dataset_id: str, *, root: Optional[str] = None, batch_size: int = 256, download: bool = False, **kwargs):
        root = root or _default_root()
        self.dataset_id = dataset_id
        cache = os.path.join(root, dataset_id)
        hdf5 = os.path.join(root, f"{dataset_id}.hdf5")
        npz = os.path.join(root, f"{dataset_id}.npz")
        if not os.path.isdir(cache):
            src = npz if os.path.exists(npz) else hdf5
            if not os.path.exists(src):
                if not download:
                    raise FileNotFoundError(
                        f"neither memmap cache {cache!r} nor source {hdf5!r} "
                        "exist; pass download=True or stage the files"
                    )
                download_file(f"{D4RL_BASE_URL}/{dataset_id}.hdf5", hdf5)
                src = hdf5
            convert_d4rl_hdf5(src, cache)
        super().__init__(cache, batch_size=batch_size, **kwargs)


class MinariExperienceReplay(LocalMemmapExperienceReplay):
    """Minari datasets (reference minari_data.py:75).

    Resolution: memmap cache → ``main_data.hdf5`` under
    ``root/<dataset_id>-src/`` (converted once) → ``download=True``
    via the ``minari`` hub client when installed."""

    def __init__(self, dataset_id: str, *, root: Optional[str] = None, batch_size: int = 256, download: bool = False, **kwargs):
        root = root or _default_root()
        self.dataset_id = dataset_id
        cache = os.path.join(root, dataset_id)
        src = os.path.join(root, f"{dataset_id}-src", "main_data.hdf5")
        npz = os.path.join(root, f"{dataset_id}-src", "main_data.npz")
        if os.path.exists(npz):
            src = npz
        if not os.path.isdir(cache):
            if not os.path.exists(src):
                if not download:
                    raise FileNotFoundError(
                        f"neither memmap cache {cache!r} nor source {src!r} "
                        "exist; pass download=True or stage the files"
                    )
                try:
                    import minari
                except ImportError as e:
                    raise ImportError(
                        "download=True for Minari requires the `minari` "
                        "package (the hub client)"
                    ) from e
                os.environ.setdefault("MINARI_DATASETS_PATH", os.path.join(root, f"{dataset_id}-src"))
                minari.download_dataset(dataset_id)
                found = None
                for dirpath, _dirs, files in os.walk(os.path.join(root, f"{dataset_id}-src")):
                    if "main_data.hdf5" in files:
                        found = os.path.join(dirpath, "main_data.hdf5")
                        break
                if found is None:
                    raise FileNotFoundError("minari download produced no main_data.hdf5")
                src = found
            convert_minari_hdf5(src, cache)
        super().__init__(cache, batch_size=batch_size, **kwargs)


class AtariDQNExperienceReplay(LocalMemmapExperienceReplay):
    """Atari-DQN replay corpus (reference atari_dqn.py:36): memmap
    cache, or converted once from pre-staged ``.npy.gz`` shard files
    under ``root/<dataset_id>-shards/`` (the corpus is served from a
    requester-pays GCS bucket — stage shards with gsutil)."""

    def __init__(self, dataset_id: str, *, root: Optional[str] = None, batch_size: int = 256, max_transitions: Optional[int] = None, **kwargs):
        root = root or _default_root()
        self.dataset_id = dataset_id
        cache = os.path.join(root, dataset_id)
        shards = os.path.join(root, f"{dataset_id}-shards")
        if not os.path.isdir(cache):
            if not os.path.isdir(shards):
                raise FileNotFoundError(
                    f"neither memmap cache {cache!r} nor shard dir {shards!r} exist"
                )
            convert_atari_shards(shards, cache, max_transitions=max_transitions)
        super().__init__(cache, batch_size=batch_size, **kwargs)


class OpenXExperienceReplay(BaseDatasetExperienceReplay):
    """Open-X Embodiment (reference openx.py:36): streams per-episode
    files through :class:`StreamingEpisodeStorage` — the corpus never
    becomes resident (reference streaming storage openx.py:548)."""

    def __init__(self, dataset_id: str, *, root: Optional[str] = None, batch_size: int = 256, sampler=None, cache_episodes: int = 16, **kwargs):
        root = root or _default_root()
        self.dataset_id = dataset_id
        path = os.path.join(root, dataset_id)
        if not os.path.isdir(path):
            raise FileNotFoundError(
                f"{path} missing; stage per-episode .npz files there "
                "(observations/actions/rewards/terminals arrays)"
            )
        storage = StreamingEpisodeStorage(path, cache_episodes=cache_episodes)
        super().__init__(storage=storage, sampler=sampler, batch_size=batch_size, **kwargs)
