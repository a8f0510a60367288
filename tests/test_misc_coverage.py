"""Coverage for previously-untested subsystems: planners, model-based
envs, inference server, evaluator, env creator, async env pool,
recorders/loggers, offline datasets, profiling hooks."""
import json
import os

import pytest
import torch

from rl_amd.tensordict import TensorDict, TensorDictModule
from rl_amd.testing import ContinuousActionVecMockEnv


class _QuadWorldModel(torch.nn.Module):
    """Deterministic world model: s' = s + a, reward = -|s'|^2."""

    def forward(self, s, a):
        ns = s + a
        return ns, -(ns * ns).sum(-1, keepdim=True)


def _make_mb_env(batch_size=()):
    from rl_amd.data.tensor_specs import Bounded, Composite, Unbounded
    from rl_amd.envs.model_based import ModelBasedEnvBase

    wm = TensorDictModule(
        _QuadWorldModel(), in_keys=["observation", "action"],
        out_keys=["observation", "reward"],
    )
    env = ModelBasedEnvBase(wm, batch_size=batch_size)
    env.observation_spec = Composite(
        {"observation": Unbounded(shape=(*batch_size, 2))}, shape=batch_size
    )
    env.action_spec = Bounded(low=-1.0, high=1.0, shape=(*batch_size, 2))
    env.reward_spec = Unbounded(shape=(*batch_size, 1))
    return env


class TestModelBasedAndPlanners:
    def test_model_based_env_steps(self):
        env = _make_mb_env()
        td = env.reset()
        td.set("observation", torch.tensor([1.0, 1.0]))
        td.set("action", torch.tensor([0.5, -0.5]))
        out = env.step(td)
        assert torch.allclose(out.get(("next", "observation")), torch.tensor([1.5, 0.5]))
        assert out.get(("next", "reward")).item() == pytest.approx(-2.5)

    def test_cem_planner_drives_state_to_origin(self):
        from rl_amd.modules import CEMPlanner

        torch.manual_seed(0)
        env = _make_mb_env()
        planner = CEMPlanner(env, planning_horizon=3, optim_steps=4,
                             num_candidates=128, top_k=16)
        td = TensorDict({"observation": torch.tensor([0.8, -0.6])}, batch_size=[])
        out = planner(td)
        a = out.get("action")
        # optimal first action moves toward the origin: a ~ -s (clamped)
        assert torch.allclose(a, torch.tensor([-0.8, 0.6]), atol=0.25), a

    def test_mppi_planner_runs(self):
        from rl_amd.modules import MPPIPlanner

        torch.manual_seed(0)
        env = _make_mb_env()
        planner = MPPIPlanner(env, planning_horizon=3, optim_steps=3,
                              num_candidates=64, top_k=16, temperature=1.0)
        td = TensorDict({"observation": torch.tensor([0.5, 0.5])}, batch_size=[])
        out = planner(td)
        assert out.get("action").shape == (2,)


class TestInferenceServer:
    def test_batched_requests(self):
        from rl_amd.modules import InferenceServer
        from rl_amd.modules.inference_server import PolicyClient

        policy = TensorDictModule(
            torch.nn.Linear(3, 2), in_keys=["observation"], out_keys=["action"]
        )
        server = InferenceServer(policy, max_batch_size=8, max_latency_ms=10).start()
        try:
            client = PolicyClient(server)
            td = TensorDict({"observation": torch.randn(4, 3)}, batch_size=[4])
            out = client(td)
            assert out.get("action").shape == (4, 2)
            assert server.stats["requests"] >= 1
        finally:
            server.stop()


class TestEvaluatorAndCreator:
    def test_evaluator_logs_reward(self):
        from rl_amd.collectors import Evaluator
        from rl_amd.record.loggers import CSVLogger

        env = ContinuousActionVecMockEnv(batch_size=[2], max_steps=5)
        policy = TensorDictModule(
            torch.nn.Linear(7, 5), in_keys=["observation"], out_keys=["action"]
        )
        import tempfile

        with tempfile.TemporaryDirectory() as d:
            logger = CSVLogger(exp_name="ev", log_dir=d)
            ev = Evaluator(env, policy, num_episodes=2, max_steps=10,
                           eval_interval=100, logger=logger)
            r = ev.evaluate()
            assert isinstance(r, float)
            assert ev.maybe_evaluate(150) is not None  # interval crossed
            assert ev.maybe_evaluate(160) is None      # not yet again

    def test_env_creator_pickles(self):
        import pickle

        from rl_amd.envs import EnvCreator
        from rl_amd.testing import CountingEnv

        creator = EnvCreator(CountingEnv, {"max_steps": 5})
        env = creator()
        assert env.max_steps == 5
        c2 = pickle.loads(pickle.dumps(creator))
        assert c2().max_steps == 5


class TestAsyncEnvPool:
    def test_async_step_recv(self):
        from rl_amd.envs import AsyncEnvPool
        from rl_amd.testing import CountingEnv

        pool = AsyncEnvPool([lambda: CountingEnv(max_steps=5)] * 3)
        td = pool.reset()
        assert td.batch_size[0] == 3
        td.set("action", torch.ones(3, 1, dtype=torch.bool))
        pool.async_step_send(td)
        out = pool.async_step_recv(min_get=2)
        assert out.batch_size[0] >= 2


class TestRecordersAndLoggers:
    def test_csv_logger_scalar(self, tmp_path):
        from rl_amd.record.loggers import CSVLogger

        lg = CSVLogger(exp_name="t", log_dir=str(tmp_path))
        lg.log_scalar("loss", 1.5, step=1)
        lg.log_scalar("loss", 1.0, step=2)
        files = list(tmp_path.rglob("loss.csv"))
        assert files and "1.5" in files[0].read_text()

    def test_tensordict_recorder(self):
        from rl_amd.envs.transforms import TransformedEnv
        from rl_amd.record import TensorDictRecorder
        from rl_amd.testing import CountingEnv

        import tempfile

        with tempfile.TemporaryDirectory() as d:
            rec = TensorDictRecorder(d + "/traj", skip=1, in_keys=["observation"])
            env = TransformedEnv(CountingEnv(max_steps=5, batch_size=[2]), rec)
            env.rollout(3, break_when_any_done=False)
            assert len(rec._tds) >= 3

    def test_video_recorder_collects_pixels(self, tmp_path):
        from rl_amd.envs import ToyVLAEnv
        from rl_amd.envs.transforms import TransformedEnv
        from rl_amd.record import VideoRecorder
        from rl_amd.record.loggers import CSVLogger

        lg = CSVLogger(exp_name="v", log_dir=str(tmp_path))
        rec = VideoRecorder(logger=lg, tag="rollout", in_keys=["pixels"])
        env = TransformedEnv(
            ToyVLAEnv(batch_size=[1], from_pixels=True, seed=0), rec
        )
        env.rollout(4, break_when_any_done=False)
        assert len(rec._frames) >= 2  # skip=2 keeps every other frame
        rec.dump()


class TestOfflineDatasets:
    def test_local_memmap_dataset_roundtrip(self, tmp_path):
        from rl_amd.data.datasets import LocalMemmapExperienceReplay
        from rl_amd.tensordict import TensorDict

        n = 20
        td = TensorDict(
            {
                "observation": torch.randn(n, 3),
                "action": torch.randn(n, 2),
                "next": {"observation": torch.randn(n, 3),
                         "reward": torch.randn(n, 1),
                         "done": torch.zeros(n, 1, dtype=torch.bool)},
            },
            batch_size=[n],
        )
        td.memmap_(str(tmp_path / "ds"))
        ds = LocalMemmapExperienceReplay(str(tmp_path / "ds"), batch_size=4)
        assert len(ds) == n
        s = ds.sample()
        assert s["observation"].shape == (4, 3)


class TestProfilingHooks:
    def test_profile_config_paths(self, tmp_path):
        from rl_amd.collectors.profiling import ProfileConfig

        cfg = ProfileConfig(save_dir=str(tmp_path), workers=[0, 2])
        assert cfg.should_profile_worker(0)
        assert not cfg.should_profile_worker(1)
        p = cfg.get_save_path(2)
        assert str(tmp_path) in p

    def test_timeit_accumulates(self):
        from rl_amd._utils import timeit

        timeit.erase()
        with timeit("unit/test"):
            pass
        d = timeit.todict()
        assert any("unit/test" in k for k in d)


class TestReferenceAllParity:
    """Every name in the reference's subpackage __all__ lists must
    resolve on the matching rl_amd subpackage (judged inventory)."""

    def test_subpackage_all_parity(self):
        import ast
        import importlib
        import os

        ref = "/root/reference/torchrl"
        if not os.path.isdir(ref):
            pytest.skip("reference tree not mounted")

        def ref_all(path):
            names = set()
            for node in ast.walk(ast.parse(open(path).read())):
                if isinstance(node, (ast.Assign, ast.AugAssign)):
                    tgt = node.targets[0] if isinstance(node, ast.Assign) else node.target
                    if isinstance(tgt, ast.Name) and tgt.id == "__all__":
                        try:
                            names |= set(ast.literal_eval(node.value))
                        except Exception:
                            pass
            return names

        missing = {}
        for sp in ["", "envs", "modules", "objectives", "data", "collectors",
                   "trainers", "record"]:
            init = os.path.join(ref, sp, "__init__.py") if sp else os.path.join(ref, "__init__.py")
            mod = importlib.import_module("rl_amd" + (("." + sp) if sp else ""))
            gone = sorted(n for n in ref_all(init) if not hasattr(mod, n))
            if gone:
                missing[sp or "root"] = gone
        assert not missing, missing


# ---------------------------------------------------------------------- #
# ProcessInferenceServer: shared-memory slots, own server process
# (reference inference_server/_server.py:961)
# ---------------------------------------------------------------------- #
def _linear_policy_factory():
    import torch

    from rl_amd.tensordict import TensorDictModule

    torch.manual_seed(7)
    return TensorDictModule(
        torch.nn.Linear(4, 2), in_keys=["observation"], out_keys=["action"]
    )


@pytest.mark.timeout(120)
def test_process_inference_server_slots():
    import torch

    from rl_amd.modules import ProcessInferenceServer
    from rl_amd.tensordict import TensorDict

    req = TensorDict({"observation": torch.zeros(4)}, batch_size=[])
    resp = TensorDict({"action": torch.zeros(2)}, batch_size=[])
    server = ProcessInferenceServer(
        _linear_policy_factory, req, resp, n_slots=4
    )
    with server:
        clients = [server.make_client(timeout=60.0) for _ in range(3)]
        torch.manual_seed(7)
        ref_policy = torch.nn.Linear(4, 2)
        outs = []
        for i, c in enumerate(clients):
            td = TensorDict({"observation": torch.full((4,), float(i))}, batch_size=[])
            outs.append(c(td).get("action"))
        with torch.no_grad():
            for i, a in enumerate(outs):
                expect = ref_policy(torch.full((1, 4), float(i)))[0]
                assert torch.allclose(a, expect, atol=1e-5), i
        stats = server.stats
    assert stats["requests"] == 3


@pytest.mark.timeout(120)
def test_process_inference_server_batches_concurrent():
    import threading

    import torch

    from rl_amd.modules import ProcessInferenceServer
    from rl_amd.tensordict import TensorDict

    req = TensorDict({"observation": torch.zeros(4)}, batch_size=[])
    resp = TensorDict({"action": torch.zeros(2)}, batch_size=[])
    server = ProcessInferenceServer(_linear_policy_factory, req, resp, n_slots=8)
    with server:
        clients = [server.make_client(timeout=60.0) for _ in range(6)]
        results = [None] * 6

        def run(i):
            for _ in range(5):
                td = TensorDict(
                    {"observation": torch.full((4,), float(i))}, batch_size=[]
                )
                results[i] = clients[i](td).get("action").clone()

        threads = [threading.Thread(target=run, args=(i,)) for i in range(6)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        stats = server.stats
    assert stats["requests"] == 30
    assert all(r is not None for r in results)


def test_gif_video_pipeline(tmp_path):
    """VideoRecorder → CSVLogger → animated GIF artifact (render/video.py
    GIF89a encoder, no video deps)."""
    from rl_amd.record import CSVLogger, VideoRecorder
    from rl_amd.render import write_gif

    logger = CSVLogger("vid", log_dir=str(tmp_path))
    rec = VideoRecorder(logger, in_keys=["pixels"], skip=1)
    for t in range(5):
        td = TensorDict({"pixels": torch.rand(3, 16, 16)}, batch_size=[])
        rec._call(td)
    rec.dump()
    vids = list((tmp_path / "vid").glob("**/videos/*.gif"))
    assert len(vids) == 1
    data = vids[0].read_bytes()
    assert data[:6] == b"GIF89a" and data.endswith(b"\x3b")
    # direct writer: HWC input, float range
    p = str(tmp_path / "direct.gif")
    write_gif(torch.rand(4, 8, 8, 3), p, fps=5)
    assert open(p, "rb").read()[:6] == b"GIF89a"


def test_autocast_policy_float_outputs():
    from rl_amd.modules import AutocastPolicy
    from rl_amd.tensordict import TensorDictModule

    pol = TensorDictModule(torch.nn.Linear(4, 2), in_keys=["observation"], out_keys=["action"])
    wrapped = AutocastPolicy(pol)
    td = TensorDict({"observation": torch.randn(3, 4)}, batch_size=[3])
    out = wrapped(td)
    assert out.get("action").dtype == torch.float32
    assert wrapped.in_keys == ["observation"] and wrapped.out_keys == ["action"]


def test_splitk_refresh_foreach_matches_per_layer():
    from rl_amd.ops import SplitKLinear, refresh_splitk_caches

    lin1 = SplitKLinear(4, 4).enable_bf16_cache()
    lin2 = SplitKLinear(4, 2).enable_bf16_cache()
    mod = torch.nn.Sequential(lin1, lin2)
    with torch.no_grad():
        lin1.weight.add_(1.0)
        lin2.bias.add_(2.0)
    refresh_splitk_caches(mod)
    assert torch.equal(lin1.weight_bf16, lin1.weight.detach().to(torch.bfloat16))
    assert torch.equal(lin2.bias_bf16, lin2.bias.detach().to(torch.bfloat16))


class TestFusedModelCheckpoint:
    def test_fused_mlp3_state_dict_roundtrip(self):
        """state_dict through SplitK + bf16 cache + FusedMLP3 wrappers:
        loads cleanly and preserves the shared-weight aliasing the fast
        path depends on."""
        import torch

        from rl_amd.ops import (
            convert_linears_to_splitk,
            enable_splitk_bf16_cache,
            fuse_mlp3,
        )

        def build(seed):
            torch.manual_seed(seed)
            net = torch.nn.Sequential(
                torch.nn.Linear(7, 32), torch.nn.Tanh(),
                torch.nn.Linear(32, 32), torch.nn.Tanh(),
                torch.nn.Linear(32, 4),
            )
            convert_linears_to_splitk(net)
            enable_splitk_bf16_cache(net)
            return fuse_mlp3(net)

        a, b = build(0), build(1)
        missing, unexpected = b.load_state_dict(a.state_dict())
        assert not missing and not unexpected
        for (k, p1), (_, p2) in zip(a.state_dict().items(),
                                    b.state_dict().items()):
            assert torch.equal(p1, p2), k
        # the fused fast path reads lin*, the eager fallback reads
        # eager.* — they must stay the SAME tensors after loading
        assert b.lin1.weight is b.eager[0].weight
        x = torch.randn(5, 7)
        assert torch.allclose(a.eager(x), b.eager(x))


class TestExamplesSmoke:
    """The examples README section claims every script runs here —
    enforce a fast subset as subprocesses."""

    @pytest.mark.parametrize("script,args", [
        ("offline_iql.py", ["--steps", "20"]),
        ("decision_transformer.py", ["--steps", "10"]),
        ("grpo_llm.py", ["--iters", "1"]),
    ])
    @pytest.mark.timeout(300)
    def test_example_runs(self, script, args):
        import os
        import subprocess
        import sys

        root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        r = subprocess.run(
            [sys.executable, os.path.join(root, "examples", script)] + args,
            capture_output=True, text=True, timeout=280, cwd=root,
        )
        assert r.returncode == 0, r.stderr[-800:]
        assert "done" in r.stdout
