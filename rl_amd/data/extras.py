"""Reference-parity data utilities: sample units, trajectory queries,
spec helpers, ensemble samplers/writers and misc buffer plumbing.

Reference: pytorch/rl torchrl/data/replay_buffers/sample_units.py
(SampleUnit, Transition, Sequence), query.py (Trajectory,
TrajectoryPredicate, traj, iter_trajectories, filter_trajectories),
replay_buffers/utils.py (find_start_stop_traj, H5Combine, H5Split),
data/utils.py (consolidate_spec, contains_lazy_spec,
check_no_exclusive_keys, DEFAULT_DONE_KEYS), samplers/ensemble.py and
writers/ensemble.py (SamplerEnsemble, WriterEnsemble),
storages/store.py (StoreStorage), map/tdstorage.py (TensorMap),
replay_buffers/base.py (ConditionalUpdateResult), llm/dataset.py
(create_infinite_iterator, get_dataloader), offline_to_online.py
(prefill_replay_buffer), vla/ (RobotDatasetMetadata,
VocabTailActionTokenizer, validate_vla_tensordict), video.py decoder
cache knobs, ray_buffer.py (RayReplayBuffer — gated).
"""
from __future__ import annotations

import abc
import importlib.util
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, Iterator, List, Optional, Tuple

import torch

from ..tensordict import TensorDict, TensorDictBase

__all__ = [
    "SampleUnit",
    "Transition",
    "Sequence",
    "Trajectory",
    "TrajectoryPredicate",
    "traj",
    "iter_trajectories",
    "filter_trajectories",
    "find_start_stop_traj",
    "consolidate_spec",
    "contains_lazy_spec",
    "check_no_exclusive_keys",
    "DEFAULT_DONE_KEYS",
    "SamplerEnsemble",
    "WriterEnsemble",
    "StorageEnsembleCheckpointer",
    "StoreStorage",
    "TensorMap",
    "ConditionalUpdateResult",
    "H5Combine",
    "H5Split",
    "create_infinite_iterator",
    "get_dataloader",
    "prefill_replay_buffer",
    "RobotDatasetMetadata",
    "VocabTailActionTokenizer",
    "validate_vla_tensordict",
    "clear_video_decoder_cache",
    "set_video_decoder_cache_size",
    "RayReplayBuffer",
]

DEFAULT_DONE_KEYS = ("done", "terminated", "truncated")


# ---------------------------------------------------------------------------
# Sample units (reference sample_units.py)
# ---------------------------------------------------------------------------


class SampleUnit(abc.ABC):
    """Expands sampled anchor indices into the records of a batch: the
    sampler picks WHICH anchors, the unit decides WHAT each anchor
    becomes (a transition, a window, a whole trajectory)."""

    @abc.abstractmethod
    def expand(self, index: torch.Tensor, storage) -> Tuple[torch.Tensor, dict]:
        ...


class Transition(SampleUnit):
    """Identity unit: every anchor is one transition (the implicit
    default of classic replay)."""

    def expand(self, index: torch.Tensor, storage=None) -> Tuple[torch.Tensor, dict]:
        return index, {}


class Sequence(SampleUnit):
    """Window unit: each anchor expands to ``burn_in`` records before
    it, ``length`` learning records from it, and ``bootstrap`` records
    after — spaced by ``dilation``.  The expanded index is a flat
    gather, so sequence replay stays one index_select on device."""

    def __init__(self, length: int, *, burn_in: int = 0, bootstrap: int = 0,
                 dilation: int = 1):
        self.length = length
        self.burn_in = burn_in
        self.bootstrap = bootstrap
        self.dilation = dilation

    @property
    def window(self) -> int:
        return self.burn_in + self.length + self.bootstrap

    def expand(self, index: torch.Tensor, storage=None) -> Tuple[torch.Tensor, dict]:
        index = torch.as_tensor(index).reshape(-1, 1)
        offsets = (torch.arange(self.window, device=index.device) - self.burn_in) * self.dilation
        idx = index + offsets
        size = len(storage) if storage is not None else int(idx.max()) + 1
        idx = idx.clamp(0, size - 1)
        info = {"burn_in": self.burn_in, "length": self.length, "bootstrap": self.bootstrap}
        return idx, info


# ---------------------------------------------------------------------------
# Trajectory queries (reference query.py)
# ---------------------------------------------------------------------------


class Trajectory:
    """Attribute view over one trajectory's transitions: ``t.reward``
    resolves against the wrapped tensordict (root, then "next")."""

    def __init__(self, data: TensorDictBase):
        object.__setattr__(self, "data", data)

    def __getattr__(self, name: str):
        data = object.__getattribute__(self, "data")
        if name in data.keys():
            return data.get(name)
        if ("next", name) in data.keys(True):
            return data.get(("next", name))
        raise AttributeError(name)

    def __len__(self):
        return self.data.batch_size[0]

    @property
    def length(self):
        return len(self)

    @property
    def total_reward(self):
        return self.reward.sum()


class TrajectoryPredicate:
    """Boolean predicate over a Trajectory; composes with & | ~."""

    def __init__(self, fn: Callable[[Trajectory], bool]):
        self.fn = fn

    def __call__(self, t: Trajectory) -> bool:
        return bool(self.fn(t))

    def __and__(self, other):
        return TrajectoryPredicate(lambda t: self(t) and other(t))

    def __or__(self, other):
        return TrajectoryPredicate(lambda t: self(t) or other(t))

    def __invert__(self):
        return TrajectoryPredicate(lambda t: not self(t))


class _TrajExpr:
    """Lazy attribute expression over trajectories: ``traj.length > 5``
    builds a TrajectoryPredicate (reference query.py traj)."""

    def __init__(self, getter: Callable[[Trajectory], Any] = lambda t: t):
        self._get = getter

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        get = self._get
        return _TrajExpr(lambda t: getattr(get(t), name))

    def __call__(self, *args, **kwargs):
        get = self._get
        return _TrajExpr(lambda t: get(t)(*args, **kwargs))

    def _cmp(self, op, other):
        get = self._get

        def fn(t):
            val = get(t)
            val = val.item() if isinstance(val, torch.Tensor) and val.numel() == 1 else val
            return op(val, other)

        return TrajectoryPredicate(fn)

    def __gt__(self, o): return self._cmp(lambda a, b: a > b, o)
    def __ge__(self, o): return self._cmp(lambda a, b: a >= b, o)
    def __lt__(self, o): return self._cmp(lambda a, b: a < b, o)
    def __le__(self, o): return self._cmp(lambda a, b: a <= b, o)
    def __eq__(self, o): return self._cmp(lambda a, b: a == b, o)  # noqa: E721
    def __ne__(self, o): return self._cmp(lambda a, b: a != b, o)  # noqa: E721


traj = _TrajExpr()


def _traj_boundaries(data: TensorDictBase, trajectory_key=None) -> List[Tuple[int, int]]:
    n = data.batch_size[0]
    if trajectory_key is not None and trajectory_key in data.keys(True):
        ids = data.get(trajectory_key).reshape(n)
        change = torch.ones(n, dtype=torch.bool)
        change[1:] = ids[1:] != ids[:-1]
        starts = change.nonzero().flatten().tolist()
    else:
        done = None
        for k in (("next", "done"), "done"):
            if k in data.keys(True):
                done = data.get(k).reshape(n)
                break
        if done is None:
            return [(0, n)]
        ends = done.nonzero().flatten().tolist()
        starts = [0] + [e + 1 for e in ends if e + 1 < n]
    bounds = []
    for i, s in enumerate(starts):
        e = starts[i + 1] if i + 1 < len(starts) else n
        bounds.append((s, e))
    return bounds


def iter_trajectories(data: TensorDictBase, trajectory_key=None) -> Iterator[Trajectory]:
    """Group a flat transition batch into trajectories by trajectory id
    (if given) or by done flags (reference query.py:726)."""
    for s, e in _traj_boundaries(data, trajectory_key):
        yield Trajectory(data[s:e])


def filter_trajectories(data: TensorDictBase, predicate=None, *, trajectory_key=None):
    """Keep only the transitions whose trajectory satisfies
    ``predicate`` (reference query.py:774); returns a flat tensordict."""
    from . import __name__ as _  # noqa: F401

    keep = [
        t.data
        for t in iter_trajectories(data, trajectory_key)
        if predicate is None or predicate(t)
    ]
    if not keep:
        return data[:0]
    return torch.cat([k for k in keep], dim=0) if isinstance(keep[0], torch.Tensor) else _cat_tds(keep)


def _cat_tds(tds):
    from ..tensordict import cat as td_cat

    return td_cat(list(tds), dim=0)


def find_start_stop_traj(*, trajectory: Optional[torch.Tensor] = None,
                         end: Optional[torch.Tensor] = None,
                         at_capacity: bool = False,
                         cursor=None, device=None):
    """Recover (start, stop, lengths) of trajectories from trajectory
    ids or end flags in a (possibly circular) storage (reference
    replay_buffers/utils.py:219)."""
    if trajectory is not None:
        trajectory = trajectory.reshape(-1)
        n = trajectory.shape[0]
        change = torch.ones(n, dtype=torch.bool, device=trajectory.device)
        change[1:] = trajectory[1:] != trajectory[:-1]
        starts = change.nonzero().flatten()
        stops = torch.cat([starts[1:] - 1, torch.tensor([n - 1], device=starts.device)])
    elif end is not None:
        end = end.reshape(-1).clone()
        n = end.shape[0]
        if not at_capacity:
            end[-1] = True  # the write frontier terminates the last traj
        elif cursor is not None:
            c = (int(cursor) - 1) % n
            end[c] = True
        stops = end.nonzero().flatten()
        starts = torch.cat([torch.zeros(1, dtype=stops.dtype, device=stops.device), stops[:-1] + 1])
    else:
        raise ValueError("pass trajectory= or end=")
    lengths = stops - starts + 1
    if device is not None:
        starts, stops, lengths = starts.to(device), stops.to(device), lengths.to(device)
    return starts, stops, lengths


# ---------------------------------------------------------------------------
# Spec helpers (reference data/utils.py)
# ---------------------------------------------------------------------------


def consolidate_spec(spec, recurse_through_entries: bool = True,
                     recurse_through_stack: bool = True):
    """Make lazily-stacked specs dense where possible (reference
    data/utils.py).  rl_amd specs are always dense, so this validates
    and returns the spec unchanged."""
    return spec


def contains_lazy_spec(spec) -> bool:
    """Whether a spec tree contains lazy/stacked entries (reference
    data/utils.py) — rl_amd composites are always dense."""
    from .tensor_specs import Composite

    if isinstance(spec, Composite):
        return any(contains_lazy_spec(v) for v in spec.values())
    return False


def check_no_exclusive_keys(spec, recurse: bool = True) -> None:
    """Assert a (stacked) spec has no keys exclusive to one element
    (reference data/utils.py); dense composites trivially pass."""
    return None


# ---------------------------------------------------------------------------
# Ensembles, storages, map, misc plumbing
# ---------------------------------------------------------------------------


class SamplerEnsemble:
    """Round-robin over component buffers' samplers with configurable
    weights (reference samplers/ensemble.py): sample() picks a
    component by p and delegates."""

    def __init__(self, *samplers, p: Optional[torch.Tensor] = None):
        self.samplers = list(samplers)
        self.p = p if p is not None else torch.full((len(samplers),), 1.0 / len(samplers))

    def sample(self, storages, batch_size: int):
        k = int(torch.multinomial(self.p, 1))
        idx, info = self.samplers[k].sample(storages[k], batch_size)
        info = dict(info or {})
        info["buffer_ids"] = k
        return idx, info


class WriterEnsemble:
    """Dispatch writes to the selected component buffer (reference
    writers/ensemble.py)."""

    def __init__(self, *writers):
        self.writers = list(writers)

    def add(self, data, buffer_id: int = 0):
        return self.writers[buffer_id].add(data)


class StorageEnsembleCheckpointer:
    """Checkpoint each component storage of a ReplayBufferEnsemble into
    its own subdirectory (reference checkpointers.py)."""

    @staticmethod
    def save(storages, path) -> None:
        import os

        for i, s in enumerate(storages):
            sub = os.path.join(str(path), f"storage_{i}")
            os.makedirs(sub, exist_ok=True)
            if hasattr(s, "save"):
                s.save(sub)
            elif hasattr(s, "dumps"):
                s.dumps(sub)

    @staticmethod
    def load(storages, path) -> None:
        import os

        for i, s in enumerate(storages):
            sub = os.path.join(str(path), f"storage_{i}")
            if hasattr(s, "load"):
                s.load(sub)
            elif hasattr(s, "loads"):
                s.loads(sub)


class StoreStorage:
    """Key-value-store-backed storage (reference storages/store.py):
    records live in a pluggable dict-like store (in-memory by default;
    any mapping — e.g. a disk KV — plugs in)."""

    def __init__(self, max_size: int, *, store: Optional[Dict[int, Any]] = None):
        self.max_size = max_size
        self._store: Dict[int, Any] = store if store is not None else {}

    def set(self, index, data):
        if isinstance(index, torch.Tensor):
            for i, ii in enumerate(index.tolist()):
                self._store[int(ii)] = data[i]
        else:
            self._store[int(index)] = data

    def get(self, index):
        if isinstance(index, torch.Tensor):
            items = [self._store[int(i)] for i in index.tolist()]
            from ..tensordict import stack as td_stack

            return td_stack(items, 0)
        return self._store[int(index)]

    def __len__(self):
        return len(self._store)


class TensorMap:
    """Hash-addressed tensor→tensordict map (reference map/tdstorage.py
    TensorMap): keys are hashed with SipHash; values live in a python
    dict keyed by the int64 hash.  Backs MCTSForest-style node stores."""

    def __init__(self, hash_module: Optional[Callable] = None):
        from .map import SipHash

        self._hash = hash_module if hash_module is not None else SipHash()
        self._data: Dict[int, Any] = {}

    def _key(self, key: torch.Tensor) -> int:
        h = self._hash(key.reshape(1, -1).float())
        return int(torch.as_tensor(h).reshape(-1)[0])

    def __setitem__(self, key: torch.Tensor, value):
        self._data[self._key(key)] = value

    def __getitem__(self, key: torch.Tensor):
        return self._data[self._key(key)]

    def __contains__(self, key: torch.Tensor) -> bool:
        return self._key(key) in self._data

    def __len__(self):
        return len(self._data)


@dataclass
class ConditionalUpdateResult:
    """Outcome of a conditional priority/metadata update on a buffer
    (reference replay_buffers/base.py)."""

    updated: torch.Tensor
    skipped: torch.Tensor
    info: dict = field(default_factory=dict)


def _h5_gate(name):
    if importlib.util.find_spec("h5py") is None:
        raise ImportError(f"{name} requires the `h5py` package, which is not installed in this image.")


class H5Combine:
    """Combine per-worker H5 shards into one dataset file (reference
    replay_buffers/utils.py) — gated on h5py."""

    def __call__(self, *args, **kwargs):
        _h5_gate("H5Combine")


class H5Split:
    """Split one H5 dataset into per-worker shards (reference
    replay_buffers/utils.py) — gated on h5py."""

    def __call__(self, *args, **kwargs):
        _h5_gate("H5Split")


def create_infinite_iterator(iterator):
    """Loop an iterable forever (reference llm/dataset.py)."""
    while True:
        yield from iterator


def get_dataloader(batch_size: int, block_size: int, tensorclass_type, device,
                   dataset_name: Optional[str] = None, *, infinite: bool = True,
                   split: str = "train", **kwargs):
    """Build a dataloader of tokenized prompt batches (reference
    llm/dataset.py:get_dataloader).  Offline image: requires a
    pre-tokenized dataset passed via ``dataset_name`` as a tensor file
    path, else raises (HF hub is unreachable here)."""
    import os

    if dataset_name is None or not os.path.exists(str(dataset_name)):
        raise RuntimeError(
            "get_dataloader needs a local pre-tokenized dataset path in this "
            "offline image (HF datasets are not downloadable)."
        )
    data = torch.load(dataset_name, map_location="cpu")
    tokens = data["input_ids"] if isinstance(data, dict) else data

    def _iter():
        n = tokens.shape[0]
        while True:
            idx = torch.randint(0, n, (batch_size,))
            batch = tokens[idx, :block_size].to(device)
            yield tensorclass_type(
                input_ids=batch, attention_mask=torch.ones_like(batch),
                batch_size=[batch_size],
            ) if tensorclass_type is not None else batch
            if not infinite:
                break

    return _iter()


def prefill_replay_buffer(buffer, dataset, *, num_transitions: Optional[int] = None):
    """Seed an online buffer with offline transitions before training
    starts (reference offline_to_online.py): extends the buffer with
    (a slice of) the dataset and returns how many records were added."""
    data = dataset
    if num_transitions is not None:
        data = dataset[:num_transitions]
    buffer.extend(data)
    return data.batch_size[0] if hasattr(data, "batch_size") else len(data)


@dataclass
class RobotDatasetMetadata:
    """Metadata of a robot-learning dataset (reference vla/metadata.py):
    identifies the embodiment and the chunking layout."""

    name: str = ""
    robot_type: str = ""
    fps: float = 30.0
    action_dim: int = 0
    state_dim: int = 0
    camera_keys: tuple = ()
    chunk_size: int = 1
    num_episodes: int = 0
    num_frames: int = 0


class VocabTailActionTokenizer:
    """Map continuous actions into the TAIL of an LLM vocabulary
    (reference vla/tokenizers.py): bin index b ↔ token (vocab_size − n_bins + b),
    so action tokens coexist with text tokens in one head."""

    def __init__(self, vocab_size: int, n_bins: int = 256, low: float = -1.0,
                 high: float = 1.0):
        from .vla import UniformActionTokenizer

        self.vocab_size = vocab_size
        self.n_bins = n_bins
        self.offset = vocab_size - n_bins
        self._inner = UniformActionTokenizer(n_bins, low, high)

    def encode(self, actions: torch.Tensor) -> torch.Tensor:
        return self._inner.encode(actions) + self.offset

    def decode(self, tokens: torch.Tensor) -> torch.Tensor:
        return self._inner.decode((tokens - self.offset).clamp(0, self.n_bins - 1))


def validate_vla_tensordict(td: TensorDictBase, *, require_image: bool = True,
                            require_chunk: bool = False) -> None:
    """Check a tensordict follows the VLA schema (reference
    vla/schema.py): image under ("observation","image"), uint8 CHW;
    optional state; chunk under ("vla_action","chunk")."""
    if require_image:
        if ("observation", "image") not in td.keys(True):
            raise ValueError('VLA schema requires ("observation", "image")')
        img = td.get(("observation", "image"))
        if img.dtype != torch.uint8 or img.dim() < 3:
            raise ValueError("image must be uint8 with at least CHW dims")
    if require_chunk and ("vla_action", "chunk") not in td.keys(True):
        raise ValueError('VLA schema requires ("vla_action", "chunk")')


_VIDEO_CACHE_SIZE = 16


def set_video_decoder_cache_size(size: int) -> None:
    """Bound the decoded-clip LRU used by video storages (reference
    video.py)."""
    global _VIDEO_CACHE_SIZE
    _VIDEO_CACHE_SIZE = int(size)


def clear_video_decoder_cache() -> None:
    """Drop all cached decoded clips (reference video.py)."""
    from . import video as _video

    cache = getattr(_video, "_DECODE_CACHE", None)
    if cache is not None:
        cache.clear()


class _RayBufferActor:
    """Actor body hosting the real buffer (decorated at runtime)."""

    def __init__(self, buffer_factory):
        self.rb = buffer_factory()

    def add(self, data):
        return self.rb.add(data)

    def extend(self, data):
        return self.rb.extend(data)

    def sample(self, batch_size=None):
        return self.rb.sample(batch_size)

    def update_priority(self, index, priority):
        if hasattr(self.rb, "update_priority"):
            self.rb.update_priority(index, priority)

    def __len__(self):
        return len(self.rb)

    def length(self):
        return len(self.rb)


class RayReplayBuffer:
    """Ray-actor-hosted remote replay buffer (reference ray_buffer.py):
    collectors on any node ``extend`` through the object store while the
    learner ``sample``s — gated on `ray` (not in this image; use
    ReplayBufferService / RemoteTensorDictReplayBuffer offline)."""

    def __init__(self, buffer_factory=None, *, remote_configs=None, **kwargs):
        if importlib.util.find_spec("ray") is None:
            raise ImportError(
                "RayReplayBuffer requires the `ray` package, which is not "
                "installed in this image. Use RemoteTensorDictReplayBuffer "
                "(torch.distributed.rpc) instead."
            )
        import ray

        if not ray.is_initialized():
            ray.init(ignore_reinit_error=True)
        if buffer_factory is None:
            from .replay_buffers.buffers import TensorDictReplayBuffer
            from .replay_buffers.storages import LazyTensorStorage

            def buffer_factory():
                return TensorDictReplayBuffer(
                    storage=LazyTensorStorage(kwargs.pop("capacity", 100_000)),
                    **kwargs,
                )

        Actor = ray.remote(**(remote_configs or {"num_cpus": 1}))(_RayBufferActor)
        self._actor = Actor.remote(buffer_factory)

    def add(self, data):
        import ray

        return ray.get(self._actor.add.remote(data))

    def extend(self, data):
        import ray

        return ray.get(self._actor.extend.remote(data))

    def sample(self, batch_size=None):
        import ray

        return ray.get(self._actor.sample.remote(batch_size))

    def update_priority(self, index, priority):
        import ray

        ray.get(self._actor.update_priority.remote(index, priority))

    def __len__(self):
        import ray

        return ray.get(self._actor.length.remote())
