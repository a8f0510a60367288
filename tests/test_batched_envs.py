"""SerialEnv / ParallelEnv tests."""
import pytest
import torch

from rl_amd.collectors import Collector
from rl_amd.envs import check_env_specs, step_mdp
from rl_amd.envs.batched_envs import ParallelEnv, SerialEnv
from rl_amd.tensordict import TensorDict
from rl_amd.testing import ContinuousActionVecMockEnv, CountingEnv


def make_counting():
    return CountingEnv(max_steps=5)


def make_cont():
    return ContinuousActionVecMockEnv(max_steps=10)


class TestSerialEnv:
    def test_shapes(self):
        env = SerialEnv(3, make_counting)
        td = env.reset()
        assert td.batch_size == torch.Size([3])
        td.set("action", torch.ones(3, 1, dtype=torch.bool))
        td = env.step(td)
        assert td.get(("next", "observation")).shape == (3, 1)
        env.close()

    def test_rollout(self):
        env = SerialEnv(3, make_cont)
        r = env.rollout(5, break_when_any_done=False)
        assert r.batch_size == torch.Size([3, 5])
        env.close()

    def test_specs(self):
        env = SerialEnv(2, make_cont)
        check_env_specs(env)
        env.close()

    def test_partial_reset(self):
        env = SerialEnv(2, make_counting)
        td = env.reset()
        for _ in range(5):
            td.set("action", torch.ones(2, 1, dtype=torch.bool))
            td, td_next = env.step_and_maybe_reset(td)
            td = td_next
        # both envs hit max_steps=5 → auto-reset to 0
        assert (td["observation"] == 0).all()
        env.close()

    def test_seed(self):
        env = SerialEnv(2, make_cont)
        env.set_seed(0)
        r1 = env.reset()
        env.set_seed(0)
        r2 = env.reset()
        assert torch.allclose(r1["observation"], r2["observation"])
        env.close()

    def test_collector_integration(self):
        env = SerialEnv(2, make_cont)
        col = Collector(env, frames_per_batch=20, total_frames=40)
        for b in col:
            assert b.batch_size == torch.Size([2, 10])
        col.shutdown()


class TestParallelEnv:
    def test_shapes(self):
        env = ParallelEnv(2, make_counting)
        try:
            td = env.reset()
            assert td.batch_size == torch.Size([2])
            td.set("action", torch.ones(2, 1, dtype=torch.bool))
            td = env.step(td)
            assert td.get(("next", "observation")).shape == (2, 1)
            assert td.get(("next", "reward")).shape == (2, 1)
        finally:
            env.close()

    def test_step_and_maybe_reset(self):
        env = ParallelEnv(2, make_counting)
        try:
            td = env.reset()
            for _ in range(5):
                td.set("action", torch.ones(2, 1, dtype=torch.bool))
                td, td_next = env.step_and_maybe_reset(td)
                td = td_next
            assert (td["observation"] == 0).all()
        finally:
            env.close()

    def test_rollout(self):
        env = ParallelEnv(2, make_cont)
        try:
            r = env.rollout(4, break_when_any_done=False)
            assert r.batch_size == torch.Size([2, 4])
        finally:
            env.close()

    def test_seed_determinism(self):
        env = ParallelEnv(2, make_cont)
        try:
            env.set_seed(3)
            r1 = env.reset()["observation"]
            env.set_seed(3)
            r2 = env.reset()["observation"]
            assert torch.allclose(r1, r2)
        finally:
            env.close()

    def test_collector_integration(self):
        env = ParallelEnv(2, make_cont)
        col = Collector(env, frames_per_batch=16, total_frames=32)
        n = 0
        for b in col:
            n += b.numel()
        assert n == 32
        col.shutdown()


class TestParallelEnvSharedMemory:
    def test_shared_buffer_active(self):
        env = ParallelEnv(2, make_counting, shared_memory=True)
        try:
            assert env._shared is not None
            assert env._shared.is_shared()
            td = env.reset()
            td.set("action", torch.ones(2, 1, dtype=torch.bool))
            td, nxt = env.step_and_maybe_reset(td)
            assert (td.get(("next", "observation")) == 1).all()
        finally:
            env.close()

    def test_no_buffer_mode_matches(self):
        env_sh = ParallelEnv(2, make_counting, shared_memory=True)
        env_pk = ParallelEnv(2, make_counting, shared_memory=False)
        try:
            td1 = env_sh.reset()
            td2 = env_pk.reset()
            for _ in range(4):
                act = torch.ones(2, 1, dtype=torch.bool)
                td1.set("action", act)
                td2.set("action", act.clone())
                td1, n1 = env_sh.step_and_maybe_reset(td1)
                td2, n2 = env_pk.step_and_maybe_reset(td2)
                assert torch.equal(
                    td1.get(("next", "observation")), td2.get(("next", "observation"))
                )
                td1, td2 = n1, n2
        finally:
            env_sh.close()
            env_pk.close()


# ---------------------------------------------------------------------- #
# HIP-IPC tensor sharing + event sync (reference batched_envs.py
# 3349-3361: CUDA tensors shared across processes with event handshake;
# on this pool HSA_ENABLE_IPC_MODE_LEGACY=0 routes through dmabuf IPC)
# ---------------------------------------------------------------------- #
def _hip_ipc_child(q_in, q_out):
    import torch

    t, evt = q_in.get(timeout=60)
    evt.synchronize()  # producer's writes visible
    assert float(t[0]) == 1.0
    t.add_(41.0)  # write back through the SAME device memory
    done = torch.cuda.Event(interprocess=False)
    done.record()
    done.synchronize()
    q_out.put(True)


@pytest.mark.gpu
@pytest.mark.timeout(180)
def test_hip_ipc_tensor_sharing_with_event_sync():
    import torch.multiprocessing as tmp

    ctx = tmp.get_context("spawn")
    q_in, q_out = ctx.Queue(), ctx.Queue()
    proc = ctx.Process(target=_hip_ipc_child, args=(q_in, q_out))
    proc.start()
    t = torch.zeros(4, device="cuda")
    t += 1.0
    evt = torch.cuda.Event(interprocess=True)
    evt.record()
    q_in.put((t, evt))
    assert q_out.get(timeout=120) is True
    proc.join(30)
    torch.cuda.synchronize()
    assert float(t[0]) == 42.0  # child's write visible through IPC
