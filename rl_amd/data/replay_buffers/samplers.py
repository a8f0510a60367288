"""Replay samplers: Random, WithoutReplacement, Prioritized, Slice.

Reference: pytorch/rl torchrl/data/replay_buffers/samplers/
(Sampler base.py:57, RandomSampler random.py:23,
SamplerWithoutReplacement random.py:422, PrioritizedSampler
prioritized.py:64 — sample:478 → scan_lower_bound:510, IS weights
:528-537, update_priority:550 —, SliceSampler slice.py:25).

The prioritized path runs on vectorized segment trees (segment_tree.py);
with GPU-resident storage the trees live in HBM and sample/update never
leave the device.
"""
from __future__ import annotations

from typing import Any, Dict, Optional, Sequence, Tuple, Union

import numpy as np
import torch

from .segment_tree import MinSegmentTree, SumSegmentTree

__all__ = [
    "Sampler",
    "RandomSampler",
    "SamplerWithoutReplacement",
    "PrioritizedSampler",
    "SliceSampler",
    "SliceSamplerWithoutReplacement",
    "PrioritizedSliceSampler",
]


class Sampler:
    """ABC (reference samplers/base.py:57)."""

    def sample(self, storage, batch_size: int) -> Tuple[torch.Tensor, dict]:
        raise NotImplementedError

    def add(self, index) -> None:
        pass

    def extend(self, index) -> None:
        pass

    def update_priority(self, index, priority, *, storage=None) -> None:
        pass

    def mark_update(self, index, *, storage=None) -> None:
        pass

    def state_dict(self) -> dict:
        return {}

    def load_state_dict(self, sd: dict) -> None:
        pass

    def _empty(self):
        pass

    @property
    def ran_out(self) -> bool:
        return False

    def dumps(self, path):
        pass

    def loads(self, path):
        pass


class RandomSampler(Sampler):
    """Uniform with replacement (reference random.py:23)."""

    def sample(self, storage, batch_size: int):
        n = len(storage)
        if n == 0:
            raise RuntimeError("cannot sample from an empty storage")
        device = getattr(storage, "device", None)
        index = torch.randint(0, n, (batch_size,), device=device)
        return index, {}


class SamplerWithoutReplacement(Sampler):
    """Random permutation epochs (reference random.py:422)."""

    def __init__(self, drop_last: bool = False, shuffle: bool = True):
        self.drop_last = drop_last
        self.shuffle = shuffle
        self._sample_list: Optional[torch.Tensor] = None
        self._ran_out = False
        self.len_storage = 0

    def _single_karwalk(self, n, device):
        if self.shuffle:
            return torch.randperm(n, device=device)
        return torch.arange(n, device=device)

    def sample(self, storage, batch_size: int):
        n = len(storage)
        device = getattr(storage, "device", None)
        if n == 0:
            raise RuntimeError("cannot sample from an empty storage")
        if (
            self._sample_list is None
            or self.len_storage != n
            or self._sample_list.numel() == 0
        ):
            self._sample_list = self._single_karwalk(n, device)
            self.len_storage = n
        if self._sample_list.numel() < batch_size:
            if self.drop_last:
                self._sample_list = self._single_karwalk(n, device)
            else:
                extra = self._single_karwalk(n, device)
                self._sample_list = torch.cat([self._sample_list, extra])
        index = self._sample_list[:batch_size]
        self._sample_list = self._sample_list[batch_size:]
        self._ran_out = self._sample_list.numel() == 0
        return index, {}

    @property
    def ran_out(self):
        return self._ran_out

    def _empty(self):
        self._sample_list = None
        self.len_storage = 0

    def state_dict(self):
        return {
            "_sample_list": self._sample_list,
            "len_storage": self.len_storage,
            "drop_last": self.drop_last,
        }

    def load_state_dict(self, sd):
        self._sample_list = sd["_sample_list"]
        self.len_storage = sd["len_storage"]
        self.drop_last = sd["drop_last"]


class PrioritizedSampler(Sampler):
    """Proportional prioritized experience replay
    (reference prioritized.py:64; Schaul et al. 2015).

    ``p_i = (|δ_i| + eps)^α``; sampling by inverse-CDF descent on a sum
    tree; importance weights ``w_i = (p_i / p_min)^{-β}`` normalized by the
    min-tree (reference :528-537).
    """

    def __init__(
        self,
        max_capacity: int,
        alpha: float = 0.7,
        beta: float = 0.5,
        eps: float = 1e-8,
        reduction: str = "max",
        max_priority_within_buffer: bool = False,
        device=None,
        dtype=torch.float64,
    ):
        if alpha < 0 or beta < 0:
            raise ValueError("alpha and beta must be non-negative")
        self.max_capacity = max_capacity
        self.alpha = alpha
        self.beta = beta
        self.eps = eps
        self.reduction = reduction
        self.device = device
        self.dtype = dtype
        self._init_trees()

    def _init_trees(self):
        self._sum_tree = SumSegmentTree(
            self.max_capacity, device=self.device, dtype=self.dtype
        )
        self._min_tree = MinSegmentTree(
            self.max_capacity, device=self.device, dtype=self.dtype
        )
        self._max_priority = 1.0
        self._device_tree = None

    def _maybe_promote(self, storage) -> None:
        """Move the trees into HBM (fused HIP kernels) once the storage is
        seen to live on a HIP device (reference selects CUDA trees the same
        way, prioritized.py:289-336)."""
        if self._device_tree is not None:
            return
        dev = getattr(storage, "device", None)
        if dev is None or torch.device(dev).type != "cuda":
            self._device_tree = False  # sentinel: stay on torch trees
            return
        try:
            from ...ops import DeviceSumTree, HAS_HIP_EXT

            if not HAS_HIP_EXT:
                self._device_tree = False
                return
            tree = DeviceSumTree(self.max_capacity, device=dev, with_min=True)
            vals = self._sum_tree.dump_values()
            nz = vals.nonzero().reshape(-1)
            if nz.numel():
                tree.update(nz.to(dev), vals[nz].to(dev))
            self._device_tree = tree
        except Exception:
            self._device_tree = False

    @property
    def default_priority(self) -> float:
        return (self._max_priority + self.eps) ** self.alpha

    def add(self, index):
        self.extend(index)

    def extend(self, index):
        index = torch.as_tensor(index, dtype=torch.long).reshape(-1)
        val = torch.full(
            (index.numel(),), self.default_priority, dtype=self.dtype
        )
        if self._device_tree:
            self._device_tree.update(index, val)
            return
        self._sum_tree.update(index, val)
        self._min_tree.update(index, val)

    def sample(self, storage, batch_size: int):
        n = len(storage)
        if n == 0:
            raise RuntimeError("cannot sample from an empty storage")
        self._maybe_promote(storage)
        if self._device_tree:
            tree = self._device_tree
            p_sum = tree.total()
            p_min = tree.min()
            mass = (
                torch.rand(batch_size, device=tree.device, dtype=torch.float64)
                * p_sum
            )
            index = tree.scan_lower_bound(mass).clamp_max(n - 1)
            p = tree.get(index)
            weight = (p / p_min).pow(-self.beta)
            return index, {"_weight": weight.to(torch.float32)}
        p_sum = self._sum_tree.query(0, n)
        p_min = self._min_tree.query(0, n)
        if p_sum <= 0:
            raise RuntimeError("empty priority sum")
        mass = (
            torch.rand(batch_size, device=self._sum_tree.device, dtype=self.dtype)
            * p_sum
        )
        index = self._sum_tree.scan_lower_bound(mass)
        index = index.clamp_max(n - 1)
        p = self._sum_tree[index]
        weight = (p / p_min).pow(-self.beta)
        dev = getattr(storage, "device", None)
        if dev is not None and index.device != torch.device(dev):
            index = index.to(dev)
            weight = weight.to(dev)
        return index, {"_weight": weight.to(torch.float32)}

    def update_priority(self, index, priority, *, storage=None):
        index = torch.as_tensor(index, dtype=torch.long).reshape(-1)
        priority = torch.as_tensor(priority, dtype=torch.float64).reshape(-1)
        if priority.numel() == 1 and index.numel() > 1:
            priority = priority.expand(index.numel())
        if self.reduction != "none" and priority.numel() != index.numel():
            # multi-dim priorities per sample: reduce
            priority = priority.reshape(index.numel(), -1)
            if self.reduction == "max":
                priority = priority.max(-1).values
            elif self.reduction == "mean":
                priority = priority.mean(-1)
            elif self.reduction == "min":
                priority = priority.min(-1).values
        self._max_priority = max(self._max_priority, float(priority.max()))
        p_alpha = (priority + self.eps).pow(self.alpha)
        if self._device_tree:
            self._device_tree.update(index, p_alpha)
            return
        self._sum_tree.update(index, p_alpha)
        self._min_tree.update(index, p_alpha)

    def mark_update(self, index, *, storage=None):
        self.update_priority(index, torch.full_like(
            torch.as_tensor(index, dtype=torch.float64), self._max_priority
        ))

    def state_dict(self):
        if self._device_tree:
            leaves = self._device_tree.sum_tree[
                self._device_tree.size : self._device_tree.size + self.max_capacity
            ].cpu()
            sum_values = min_values = leaves
        else:
            sum_values = self._sum_tree.dump_values()
            min_values = self._min_tree.dump_values()
        return {
            "alpha": self.alpha,
            "beta": self.beta,
            "eps": self.eps,
            "_max_priority": self._max_priority,
            "sum_values": sum_values,
            "min_values": min_values,
        }

    def load_state_dict(self, sd):
        self.alpha = sd["alpha"]
        self.beta = sd["beta"]
        self.eps = sd["eps"]
        self._max_priority = sd["_max_priority"]
        self._init_trees()
        self._sum_tree.load_values(sd["sum_values"])
        self._min_tree.load_values(sd["min_values"])


class SliceSampler(Sampler):
    """Sample fixed-length trajectory windows from a [B*T]-flat or
    trajectory-annotated storage (reference slice.py:25).

    Needs either ``traj_key`` leaves stored with the data (episode ids) or
    ``end_key`` done flags; scans once per sample call (vectorized).
    """

    def __init__(
        self,
        num_slices: Optional[int] = None,
        slice_len: Optional[int] = None,
        end_key=("next", "done"),
        traj_key=("collector", "traj_ids"),
        truncated_key=("next", "truncated"),
        strict_length: bool = True,
        compile_mode=None,
    ):
        if (num_slices is None) == (slice_len is None):
            raise ValueError("provide exactly one of num_slices / slice_len")
        self.num_slices = num_slices
        self.slice_len = slice_len
        self.end_key = end_key
        self.traj_key = traj_key
        self.truncated_key = truncated_key
        self.strict_length = strict_length

    def _get_traj_bounds(self, storage):
        """Returns (start, length) per trajectory from stored ids or dones."""
        n = len(storage)
        data = storage[0:n] if not hasattr(storage, "_storage") else storage._storage[0:n]
        traj = None
        try:
            traj = data.get(self.traj_key)
        except (KeyError, AttributeError):
            pass
        if traj is not None:
            traj = traj.reshape(-1)
            change = torch.ones_like(traj, dtype=torch.bool)
            change[1:] = traj[1:] != traj[:-1]
            starts = change.nonzero().reshape(-1)
        else:
            try:
                done = data.get(self.end_key).reshape(-1)
            except (KeyError, AttributeError):
                done = torch.zeros(n, dtype=torch.bool)
            starts = torch.cat(
                [
                    torch.zeros(1, dtype=torch.long, device=done.device),
                    done[:-1].nonzero().reshape(-1) + 1,
                ]
            )
        ends = torch.cat(
            [starts[1:], torch.tensor([n], device=starts.device)]
        )
        lengths = ends - starts
        return starts, lengths

    def sample(self, storage, batch_size: int):
        starts, lengths = self._get_traj_bounds(storage)
        if self.slice_len is not None:
            slice_len = self.slice_len
            num_slices = batch_size // slice_len
        else:
            num_slices = self.num_slices
            slice_len = batch_size // num_slices
        valid = lengths >= slice_len
        if not valid.any():
            if self.strict_length:
                raise RuntimeError(
                    f"no stored trajectory is at least {slice_len} steps long"
                )
            valid = lengths > 0
        v_starts = starts[valid]
        v_lengths = lengths[valid]
        pick = torch.randint(0, v_starts.numel(), (num_slices,), device=v_starts.device)
        traj_start = v_starts[pick]
        traj_len = v_lengths[pick]
        max_off = (traj_len - slice_len).clamp_min(0)
        off = (torch.rand(num_slices, device=v_starts.device) * (max_off + 1).float()).long()
        slice_starts = traj_start + off
        index = (
            slice_starts.unsqueeze(1)
            + torch.arange(slice_len, device=v_starts.device).unsqueeze(0)
        ).reshape(-1)
        # the last step of every slice is a truncation boundary (reference
        # SliceSampler sets ('next','truncated') there); the buffer applies
        # this mask to the gathered batch
        truncated = torch.zeros(index.numel(), dtype=torch.bool, device=index.device)
        truncated[slice_len - 1 :: slice_len] = True
        return index, {
            "slice_len": slice_len,
            "num_slices": num_slices,
            "truncated": truncated,
            "truncated_key": self.truncated_key,
        }


class SliceSamplerWithoutReplacement(SliceSampler):
    """Trajectory slices without replacement within an epoch
    (reference slice_without_replacement.py:18)."""

    def __init__(self, *args, drop_last: bool = False, **kwargs):
        super().__init__(*args, **kwargs)
        self.drop_last = drop_last
        self._used: Optional[set] = None

    def sample(self, storage, batch_size: int):
        starts, lengths = self._get_traj_bounds(storage)
        if self._used is None:
            self._used = set()
        if self.slice_len is not None:
            slice_len = self.slice_len
            num_slices = batch_size // slice_len
        else:
            num_slices = self.num_slices
            slice_len = batch_size // num_slices
        avail = [
            i
            for i in range(starts.numel())
            if i not in self._used and lengths[i] >= slice_len
        ]
        if len(avail) < num_slices:
            self._used = set()
            avail = [i for i in range(starts.numel()) if lengths[i] >= slice_len]
        pick = np.random.choice(len(avail), size=num_slices, replace=False)
        chosen = [avail[i] for i in pick]
        self._used.update(chosen)
        chosen_t = torch.as_tensor(chosen, dtype=torch.long, device=starts.device)
        traj_start = starts[chosen_t]
        traj_len = lengths[chosen_t]
        max_off = (traj_len - slice_len).clamp_min(0)
        off = (torch.rand(num_slices, device=starts.device) * (max_off + 1).float()).long()
        slice_starts = traj_start + off
        index = (
            slice_starts.unsqueeze(1)
            + torch.arange(slice_len, device=starts.device).unsqueeze(0)
        ).reshape(-1)
        truncated = torch.zeros(index.numel(), dtype=torch.bool, device=index.device)
        truncated[slice_len - 1 :: slice_len] = True
        return index, {
            "slice_len": slice_len,
            "num_slices": num_slices,
            "truncated": truncated,
            "truncated_key": self.truncated_key,
        }

    def _empty(self):
        self._used = None


class PrioritizedSliceSampler(SliceSampler):
    """Slice sampling where the slice START is drawn by priority
    (reference prioritized_slice.py:22)."""

    def __init__(self, max_capacity: int, *args, alpha: float = 0.7, beta: float = 0.5, eps: float = 1e-8, **kwargs):
        super().__init__(*args, **kwargs)
        self._prio = PrioritizedSampler(max_capacity, alpha=alpha, beta=beta, eps=eps)

    def extend(self, index):
        self._prio.extend(index)

    def add(self, index):
        self._prio.add(index)

    def update_priority(self, index, priority, *, storage=None):
        self._prio.update_priority(index, priority, storage=storage)

    def sample(self, storage, batch_size: int):
        starts, lengths = self._get_traj_bounds(storage)
        if self.slice_len is not None:
            slice_len = self.slice_len
            num_slices = batch_size // slice_len
        else:
            num_slices = self.num_slices
            slice_len = batch_size // num_slices
        seed_idx, info = self._prio.sample(storage, num_slices)
        # map each seed index into its trajectory, clamp window inside
        n = len(storage)
        # find trajectory of each seed via searchsorted
        traj_of = torch.searchsorted(starts.to(seed_idx.device), seed_idx, right=True) - 1
        t_start = starts.to(seed_idx.device)[traj_of]
        t_len = lengths.to(seed_idx.device)[traj_of]
        max_start = (t_start + t_len - slice_len).clamp_min(t_start)
        slice_starts = torch.minimum(seed_idx, max_start)
        index = (
            slice_starts.unsqueeze(1)
            + torch.arange(slice_len, device=seed_idx.device).unsqueeze(0)
        ).reshape(-1)
        w = info.get("_weight")
        if w is not None:
            info["_weight"] = w.repeat_interleave(slice_len)
        info.update({"slice_len": slice_len, "num_slices": num_slices})
        return index, info
