"""Collector throughput benchmarks — same metric names as the reference
suite (pytorch/rl benchmarks/test_collectors_benchmark.py: Collector /
MultiSyncCollector / MultiAsyncCollector frames-per-second, payload
sensitivity via a zero-work env with a 64 KB observation :37).

Run as pytest (each test prints its fps) or standalone:
``python benchmarks/test_collectors_benchmark.py``.
"""
import os
import sys
import time

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rl_amd.collectors import AsyncCollector, Collector, MultiAsyncCollector, MultiSyncCollector
from rl_amd.data.tensor_specs import Composite, Unbounded, Bounded
from rl_amd.envs.common import EnvBase
from rl_amd.tensordict import TensorDict
from rl_amd.testing import ContinuousActionVecMockEnv


class _PayloadEnv(EnvBase):
    """Zero-work env with a configurable observation payload
    (reference _PayloadEnv:37)."""

    _supports_masked_reset = True

    def __init__(self, payload_bytes: int = 65536, batch_size=(), device=None):
        super().__init__(device=device, batch_size=batch_size)
        n = payload_bytes // 4
        bs = self.batch_size
        self.observation_spec = Composite(
            {"observation": Unbounded(shape=(*bs, n), device=self.device)},
            shape=bs,
            device=self.device,
        )
        self.action_spec = Bounded(low=-1, high=1, shape=(*bs, 1), device=self.device)
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)
        self._obs = None

    def _reset(self, tensordict=None, **kwargs):
        bs = self.batch_size
        if self._obs is None:
            self._obs = torch.zeros(
                (*bs, self.observation_spec["observation"].shape[-1]), device=self.device
            )
        return TensorDict(
            {
                "observation": self._obs,
                "done": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
                "terminated": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
            },
            batch_size=bs,
            device=self.device,
        )

    def _step(self, tensordict):
        bs = self.batch_size
        return TensorDict(
            {
                "observation": self._obs,
                "reward": torch.ones((*bs, 1), device=self.device),
                "done": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
                "terminated": torch.zeros((*bs, 1), dtype=torch.bool, device=self.device),
            },
            batch_size=bs,
            device=self.device,
        )

    def _set_seed(self, seed):
        return seed


def _measure(collector, total_frames: int) -> float:
    t0 = time.perf_counter()
    n = 0
    for batch in collector:
        n += batch.numel() if batch is not None else 0
    dt = time.perf_counter() - t0
    collector.shutdown()
    return n / dt


def _env():
    return ContinuousActionVecMockEnv(batch_size=[8], max_steps=200)


def test_single_collector_speed(benchmark=None):
    col = Collector(_env(), frames_per_batch=800, total_frames=8000)
    fps = _measure(col, 8000)
    print(f"\ncollector_fps_single: {fps:,.0f}")
    assert fps > 0


def test_payload_collector_speed():
    env = _PayloadEnv(payload_bytes=65536, batch_size=[8])
    col = Collector(env, frames_per_batch=800, total_frames=8000)
    fps = _measure(col, 8000)
    print(f"\ncollector_fps_payload64k: {fps:,.0f}")
    assert fps > 0


@pytest.mark.slow
def test_multisync_collector_speed():
    col = MultiSyncCollector([_env] * 4, frames_per_batch=3200, total_frames=16000)
    fps = _measure(col, 16000)
    print(f"\ncollector_fps_multisync4: {fps:,.0f}")
    assert fps > 0


@pytest.mark.slow
def test_multiasync_collector_speed():
    col = MultiAsyncCollector([_env] * 4, frames_per_batch=3200, total_frames=16000)
    fps = _measure(col, 16000)
    print(f"\ncollector_fps_multiasync4: {fps:,.0f}")
    assert fps > 0


@pytest.mark.gpu
def test_single_collector_speed_gpu():
    env = ContinuousActionVecMockEnv(batch_size=[1024], max_steps=200)
    col = Collector(env.to("cuda"), frames_per_batch=102400, total_frames=512000)
    fps = _measure(col, 512000)
    print(f"\ncollector_fps_single_gpu1024: {fps:,.0f}")
    assert fps > 0


if __name__ == "__main__":
    test_single_collector_speed()
    test_payload_collector_speed()
    test_multisync_collector_speed()
    test_multiasync_collector_speed()
