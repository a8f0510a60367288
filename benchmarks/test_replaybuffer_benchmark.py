"""Replay-buffer benchmarks — same metric names as the reference suite
(pytorch/rl benchmarks/test_replaybuffer_benchmark.py: extend/sample/
update-priority rates for uniform and prioritized buffers, slice
samplers, storage-write bandwidth)."""
import os
import sys
import time

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rl_amd.data import (
    LazyMemmapStorage,
    LazyTensorStorage,
    ReplayBuffer,
    SliceSampler,
    TensorDictPrioritizedReplayBuffer,
    TensorDictReplayBuffer,
)
from rl_amd.tensordict import TensorDict


def _data(n, obs=32, device=None):
    return TensorDict(
        {
            "observation": torch.randn(n, obs, device=device),
            "action": torch.randn(n, 8, device=device),
            "next": {
                "observation": torch.randn(n, obs, device=device),
                "reward": torch.randn(n, 1, device=device),
                "done": torch.rand(n, 1, device=device) < 0.02,
            },
        },
        batch_size=[n],
        device=device,
    )


def _rate(fn, n_iter, work_per_iter, sync=False):
    fn()  # warmup
    if sync and torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n_iter):
        fn()
    if sync and torch.cuda.is_available():
        torch.cuda.synchronize()
    return n_iter * work_per_iter / (time.perf_counter() - t0)


@pytest.mark.parametrize("storage_cls", [LazyTensorStorage, LazyMemmapStorage])
def test_rb_extend_speed(storage_cls):
    rb = TensorDictReplayBuffer(storage=storage_cls(200_000), batch_size=256)
    chunk = _data(1024)
    rate = _rate(lambda: rb.extend(chunk), 50, 1024)
    print(f"\nrb_extend_{storage_cls.__name__}: {rate:,.0f} frames/s")


@pytest.mark.parametrize("storage_cls", [LazyTensorStorage, LazyMemmapStorage])
def test_rb_sample_speed(storage_cls):
    rb = TensorDictReplayBuffer(storage=storage_cls(100_000), batch_size=256)
    rb.extend(_data(50_000))
    rate = _rate(lambda: rb.sample(), 100, 256)
    print(f"\nrb_sample_{storage_cls.__name__}: {rate:,.0f} samples/s")


def test_per_sample_update_speed():
    rb = TensorDictPrioritizedReplayBuffer(
        storage=LazyTensorStorage(100_000), batch_size=256
    )
    rb.extend(_data(50_000))

    def step():
        s = rb.sample()
        s.set("td_error", torch.rand(256))
        rb.update_tensordict_priority(s)

    rate = _rate(step, 100, 256)
    print(f"\nrb_per_sample_update: {rate:,.0f} samples/s")


def test_slice_sampler_speed():
    n = 50_000
    traj = torch.arange(n) // 250
    data = _data(n)
    data.set(("collector", "traj_ids"), traj)
    rb = ReplayBuffer(
        storage=LazyTensorStorage(n), sampler=SliceSampler(slice_len=32), batch_size=256
    )
    rb.extend(data)
    rate = _rate(lambda: rb.sample(), 100, 256)
    print(f"\nrb_slice_sample: {rate:,.0f} samples/s")


@pytest.mark.gpu
def test_per_hbm_speed_gpu():
    """PER with storage + trees resident in HBM (device kernels)."""
    rb = TensorDictPrioritizedReplayBuffer(
        storage=LazyTensorStorage(1_000_000, device="cuda"), batch_size=256
    )
    rb.extend(_data(500_000, device="cuda"))

    def step():
        s = rb.sample()
        s.set("td_error", torch.rand(256, device="cuda"))
        rb.update_tensordict_priority(s)

    rate = _rate(step, 200, 256, sync=True)
    print(f"\nrb_per_hbm_sample_update_gpu: {rate:,.0f} samples/s")


if __name__ == "__main__":
    test_rb_extend_speed(LazyTensorStorage)
    test_rb_sample_speed(LazyTensorStorage)
    test_per_sample_update_speed()
    test_slice_sampler_speed()
