"""Native-op tests: C++ segment trees (CPU) and HIP kernels vs the plain
PyTorch fp32 references (GPU-marked)."""
import numpy as np
import math
import pytest
import torch

from rl_amd.objectives.value import functional as F


def _has_ext():
    try:
        import rl_amd._C  # noqa: F401

        return True
    except ImportError:
        return False


pytestmark = pytest.mark.skipif(not _has_ext(), reason="rl_amd._C not built")


class TestCppSegmentTree:
    def test_sum_tree_matches_torch_tree(self):
        import rl_amd._C as C
        from rl_amd.data import SumSegmentTree

        cpp = C.SumSegmentTreeFp64(500)
        ref = SumSegmentTree(500)
        torch.manual_seed(0)
        for _ in range(10):
            idx = torch.randint(0, 500, (64,))
            val = torch.rand(64).double()
            cpp.update(idx, val)
            ref.update(idx, val)
        assert cpp.query(0, 500) == pytest.approx(ref.query(0, 500).item())
        assert cpp.query(17, 210) == pytest.approx(ref.query(17, 210).item())
        mass = torch.rand(256).double() * cpp.query(0, 500)
        assert (cpp.scan_lower_bound(mass) == ref.scan_lower_bound(mass)).all()

    def test_duplicate_semantics(self):
        import rl_amd._C as C

        t = C.SumSegmentTreeFp64(8)
        t.update(torch.tensor([2, 2, 2]), torch.tensor([1.0, 5.0, 9.0]))
        assert t.at(2) == 9.0

    def test_min_tree(self):
        import rl_amd._C as C

        t = C.MinSegmentTreeFp64(16)
        t.update(torch.arange(8), torch.arange(8).double() + 2)
        assert t.query(0, 8) == 2.0
        assert t.query(3, 8) == 5.0

    def test_pickle(self):
        import pickle

        import rl_amd._C as C

        t = C.SumSegmentTreeFp64(32)
        t.update(torch.arange(10), torch.rand(10).double())
        t2 = pickle.loads(pickle.dumps(t))
        assert t2.query(0, 32) == pytest.approx(t.query(0, 32))

    def test_safetanh(self):
        import rl_amd._C as C

        x = torch.tensor([-50.0, 0.0, 50.0])
        y = C.safetanh(x, 1e-6)
        assert y[0] == pytest.approx(-1 + 1e-6)
        assert y[2] == pytest.approx(1 - 1e-6)
        x2 = C.safeatanh(y, 1e-6)
        assert torch.isfinite(x2).all()


@pytest.mark.gpu
class TestHipValueScan:
    def _data(self, B=32, T=200, device="cuda", dtype=torch.float32):
        torch.manual_seed(0)
        val = torch.randn(B, T, 1, device=device, dtype=dtype)
        nval = torch.randn(B, T, 1, device=device, dtype=dtype)
        r = torch.randn(B, T, 1, device=device, dtype=dtype)
        done = torch.rand(B, T, 1, device=device) < 0.1
        term = done & (torch.rand(B, T, 1, device=device) < 0.5)
        return val, nval, r, done, term

    def test_gae_fp32_matches_oracle(self):
        from rl_amd import ops

        val, nval, r, done, term = self._data()
        adv_ref, vt_ref = F.generalized_advantage_estimate(
            0.99, 0.95, val, nval, r, done, term
        )
        adv, vt = ops.gae(0.99, 0.95, val, nval, r, done, term)
        assert torch.allclose(adv, adv_ref, atol=1e-4), (adv - adv_ref).abs().max()
        assert torch.allclose(vt, vt_ref, atol=1e-4)

    def test_gae_long_t(self):
        from rl_amd import ops

        val, nval, r, done, term = self._data(B=8, T=2000)
        adv_ref, _ = F.generalized_advantage_estimate(0.99, 0.95, val, nval, r, done, term)
        adv, _ = ops.gae(0.99, 0.95, val, nval, r, done, term)
        assert torch.allclose(adv, adv_ref, atol=1e-3)

    def test_gae_bf16(self):
        from rl_amd import ops

        val, nval, r, done, term = self._data(dtype=torch.bfloat16)
        adv, vt = ops.gae(0.99, 0.95, val, nval, r, done, term)
        adv_ref, vt_ref = F.generalized_advantage_estimate(
            0.99, 0.95, val.float(), nval.float(), r.float(), done, term
        )
        assert (adv.float() - adv_ref).abs().max() < 0.1

    def test_revscan(self):
        from rl_amd import ops

        torch.manual_seed(1)
        a = torch.rand(16, 300, device="cuda") * 0.95
        b = torch.randn(16, 300, device="cuda")
        y = ops.revscan(a, b)
        y_ref = F._reverse_scan(b.unsqueeze(-1), a.unsqueeze(-1)).squeeze(-1)
        assert torch.allclose(y, y_ref, atol=1e-4)

    def test_vtrace_matches_oracle(self):
        from rl_amd import ops

        val, nval, r, done, term = self._data()
        lp = torch.randn_like(val) * 0.2
        lm = torch.randn_like(val) * 0.2
        adv_ref, vs_ref = F.vtrace_advantage_estimate(
            0.99, lp, lm, val, nval, r, done, term
        )
        adv, vs = ops.vtrace(0.99, lp, lm, val, nval, r, done, term)
        assert torch.allclose(vs, vs_ref, atol=1e-4), (vs - vs_ref).abs().max()
        assert torch.allclose(adv, adv_ref, atol=1e-4), (adv - adv_ref).abs().max()


@pytest.mark.gpu
class TestHipSegmentTree:
    def test_device_tree_matches_cpu(self):
        from rl_amd.ops import DeviceSumTree

        torch.manual_seed(0)
        tree = DeviceSumTree(1000, device="cuda")
        ref = np.zeros(1000)
        for _ in range(10):
            idx = torch.randint(0, 1000, (128,), device="cuda")
            val = torch.rand(128, device="cuda").double() + 0.01
            tree.update(idx, val)
            for i, v in zip(idx.cpu().tolist(), val.cpu().tolist()):
                ref[i] = v
        assert tree.total().item() == pytest.approx(ref.sum(), rel=1e-9)
        assert tree.min().item() == pytest.approx(
            ref[ref > 0].min() if (ref > 0).any() else np.inf
        )

    def test_device_scan_lower_bound(self):
        from rl_amd.ops import DeviceSumTree

        torch.manual_seed(0)
        tree = DeviceSumTree(4096, device="cuda")
        vals = torch.rand(4096, device="cuda").double() + 1e-3
        tree.update(torch.arange(4096, device="cuda"), vals)
        cs = vals.cpu().numpy().cumsum()
        mass = torch.rand(2048, device="cuda").double() * tree.total()
        found = tree.scan_lower_bound(mass).cpu().numpy()
        expected = np.searchsorted(cs, mass.cpu().numpy(), side="right")
        assert (found == expected).all()

    def test_device_tree_duplicates(self):
        from rl_amd.ops import DeviceSumTree

        tree = DeviceSumTree(16, device="cuda")
        tree.update(
            torch.tensor([3, 3, 3], device="cuda"),
            torch.tensor([1.0, 2.0, 7.0], device="cuda").double(),
        )
        assert tree.get(torch.tensor([3])).item() == 7.0
        assert tree.total().item() == 7.0


@pytest.mark.gpu
class TestGAEIntegration:
    def test_gae_estimator_uses_kernel_on_gpu(self):
        from rl_amd.modules import MLP, ValueOperator
        from rl_amd.objectives.value.advantages import GAE
        from rl_amd.tensordict import TensorDict

        device = "cuda"
        critic = ValueOperator(
            MLP(in_features=4, out_features=1, num_cells=[16], device=device),
            in_keys=["observation"],
        )
        B, T = 16, 32
        td = TensorDict(
            {
                "observation": torch.randn(B, T, 4, device=device),
                "next": {
                    "observation": torch.randn(B, T, 4, device=device),
                    "reward": torch.randn(B, T, 1, device=device),
                    "done": torch.rand(B, T, 1, device=device) < 0.1,
                    "terminated": torch.rand(B, T, 1, device=device) < 0.05,
                },
            },
            batch_size=[B, T],
        )
        est = GAE(gamma=0.99, lmbda=0.95, value_network=critic)
        est(td)
        assert torch.isfinite(td.get("advantage")).all()
        # cross-check against non-vectorized CPU oracle
        est_ref = GAE(gamma=0.99, lmbda=0.95, value_network=critic, vectorized=False)
        td2 = td.exclude("advantage", "value_target")
        est_ref(td2)
        assert torch.allclose(
            td.get("advantage"), td2.get("advantage"), atol=1e-3
        )


@pytest.mark.gpu
class TestFusedActor:
    def _actor(self, obs_dim=17, act_dim=6, hidden=64, device="cuda"):
        from rl_amd.modules import (
            MLP,
            NormalParamExtractor,
            ProbabilisticActor,
            TanhNormal,
        )
        from rl_amd.tensordict import TensorDictModule

        net = torch.nn.Sequential(
            MLP(in_features=obs_dim, out_features=2 * act_dim, num_cells=[hidden, hidden], device=device),
            NormalParamExtractor(),
        )
        mod = TensorDictModule(net, in_keys=["observation"], out_keys=["loc", "scale"])
        return ProbabilisticActor(
            mod, in_keys=["loc", "scale"], distribution_class=TanhNormal, return_log_prob=True
        )

    def test_loc_scale_match_eager(self):
        from rl_amd import ops
        from rl_amd.tensordict import TensorDict

        torch.manual_seed(0)
        actor = self._actor()
        fused = ops.FusedTanhNormalActor(actor)
        obs = torch.randn(256, 17, device="cuda")
        with torch.no_grad():
            td_f = fused(TensorDict({"observation": obs}, batch_size=[256]))
            td_e = actor(TensorDict({"observation": obs}, batch_size=[256]))
        assert torch.allclose(td_f.get("loc"), td_e.get("loc"), atol=1e-4), (
            (td_f.get("loc") - td_e.get("loc")).abs().max()
        )
        assert torch.allclose(td_f.get("scale"), td_e.get("scale"), atol=1e-4)

    def test_logprob_matches_eager_dist(self):
        """The fused kernel's log-prob must equal TanhNormal.log_prob of
        the same action under the same (loc, scale)."""
        from rl_amd import ops
        from rl_amd.modules import TanhNormal
        from rl_amd.tensordict import TensorDict

        torch.manual_seed(1)
        actor = self._actor()
        fused = ops.FusedTanhNormalActor(actor)
        obs = torch.randn(512, 17, device="cuda")
        with torch.no_grad():
            td = fused(TensorDict({"observation": obs}, batch_size=[512]))
            dist = TanhNormal(td.get("loc"), td.get("scale"))
            lp_ref = dist.log_prob(td.get("action"))
        assert (td.get("sample_log_prob") - lp_ref).abs().max() < 1e-3

    def test_action_in_bounds(self):
        from rl_amd import ops
        from rl_amd.tensordict import TensorDict

        actor = self._actor()
        fused = ops.FusedTanhNormalActor(actor)
        obs = torch.randn(1024, 17, device="cuda") * 5
        with torch.no_grad():
            td = fused(TensorDict({"observation": obs}, batch_size=[1024]))
        assert td.get("action").abs().max() <= 1.0


@pytest.mark.gpu
class TestSplitKWgrad:
    """wgrad_splitk vs fp32 matmul oracle (kernel: csrc/wgrad.hip)."""

    @pytest.mark.parametrize(
        "K,N,M", [(16384, 64, 64), (16384, 64, 17), (16384, 12, 64), (10000, 64, 64), (4096, 100, 80)]
    )
    def test_wgrad_matches_fp32(self, K, N, M):
        from rl_amd import _C

        torch.manual_seed(0)
        dy = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
        x = torch.randn(K, M, device="cuda", dtype=torch.bfloat16)
        dw, db = _C.wgrad_splitk(dy, x, True)
        ref_dw = dy.float().t() @ x.float()
        ref_db = dy.float().sum(0)
        assert torch.allclose(dw, ref_dw, rtol=1e-3, atol=1e-2), (
            (dw - ref_dw).abs().max().item()
        )
        assert torch.allclose(db, ref_db, rtol=1e-3, atol=1e-2)

    def test_splitk_linear_autograd(self):
        from rl_amd.ops import SplitKLinear

        torch.manual_seed(0)
        lin_ref = torch.nn.Linear(64, 64, device="cuda")
        lin_sk = SplitKLinear(64, 64, device="cuda")
        with torch.no_grad():
            lin_sk.weight.copy_(lin_ref.weight)
            lin_sk.bias.copy_(lin_ref.bias)
        x = torch.randn(8192, 64, device="cuda")
        with torch.autocast("cuda", dtype=torch.bfloat16, cache_enabled=False):
            y_ref = lin_ref(x).float().pow(2).sum()
            y_sk = lin_sk(x).float().pow(2).sum()
        y_ref.backward()
        y_sk.backward()
        assert torch.allclose(y_ref, y_sk, rtol=1e-3)
        assert torch.allclose(
            lin_ref.weight.grad, lin_sk.weight.grad, rtol=2e-2, atol=2e-1
        ), (lin_ref.weight.grad - lin_sk.weight.grad).abs().max().item()
        assert torch.allclose(lin_ref.bias.grad, lin_sk.bias.grad, rtol=2e-2, atol=2e-1)
        # sharper statement than the loose pairwise tolerance: against a
        # FP32 oracle the split-K kernel (fp32 accumulation) must be at
        # least as accurate as torch's own autocast-bf16 backward
        lin_32 = torch.nn.Linear(64, 64, device="cuda")
        with torch.no_grad():
            lin_32.weight.copy_(lin_ref.weight)
            lin_32.bias.copy_(lin_ref.bias)
        lin_32(x).pow(2).sum().backward()
        err_ref = (lin_ref.weight.grad - lin_32.weight.grad).abs().max()
        err_sk = (lin_sk.weight.grad - lin_32.weight.grad).abs().max()
        assert err_sk <= 1.5 * err_ref + 1e-3, (
            f"splitk wgrad error {err_sk:.4f} vs torch-bf16 error {err_ref:.4f}"
        )
        berr_ref = (lin_ref.bias.grad - lin_32.bias.grad).abs().max()
        berr_sk = (lin_sk.bias.grad - lin_32.bias.grad).abs().max()
        assert berr_sk <= 1.5 * berr_ref + 1e-3


@pytest.mark.gpu
class TestFusedEnvStep:
    def test_matches_eager_step(self):
        from rl_amd.envs.custom.synthetic import HalfCheetahVec

        torch.manual_seed(0)
        B = 64
        env_e = HalfCheetahVec(batch_size=[B], device="cuda", dtype=torch.float32)
        env_f = HalfCheetahVec(batch_size=[B], device="cuda", dtype=torch.float32)
        env_e.set_seed(0)
        env_f.set_seed(0)
        td_e = env_e.reset()
        td_f = env_f.reset()
        # align the initial state exactly
        env_f._state.copy_(env_e._state)
        env_f._t.copy_(env_e._t)
        env_f.enable_capture_mode(True)  # capture-safe + cuda → fused path
        env_e._capture_safe = True
        env_e._fused_step = lambda td: None  # force eager in-place path
        for i in range(12):
            act = torch.rand(B, 6, device="cuda") * 2 - 1
            td_e.set("action", act)
            td_f.set("action", act.clone())
            out_e = env_e.step(td_e.clone(False))
            out_f = env_f.step(td_f.clone(False))
            for k in [("next", "observation"), ("next", "reward")]:
                a, b = out_e.get(k), out_f.get(k)
                assert torch.allclose(a, b, rtol=1e-4, atol=1e-5), (i, k, (a - b).abs().max())
            assert torch.equal(out_e.get(("next", "done")), out_f.get(("next", "done")))


@pytest.mark.gpu
class TestStoreDirectEnvStep:
    def test_auto_reset_semantics(self):
        """synthetic_env_step_into must write the TERMINAL obs to the
        store while carrying the RESET state (step_and_maybe_reset
        contract), zeroing t at truncation."""
        from rl_amd import _C
        from rl_amd.envs.custom.synthetic import HalfCheetahVec

        torch.manual_seed(0)
        B, T = 32, 4
        env = HalfCheetahVec(batch_size=[B], device="cuda", dtype=torch.float32)
        env.set_seed(0)
        env.reset()
        env.enable_capture_mode(True)
        S, A = env.obs_dim, env.act_dim
        # force truncation on the next step
        env._t.fill_(float(env.max_steps) - 1)
        prev = torch.empty(B, T, S, device="cuda")
        nxt = torch.empty(B, T, S, device="cuda")
        rew = torch.empty(B, T, 1, device="cuda")
        done = torch.empty(B, T, 1, dtype=torch.bool, device="cuda")
        state_before = env._state.clone()
        action = torch.rand(B, A, device="cuda") * 2 - 1
        noise = torch.randn(B, S, device="cuda") * 0.1
        t0 = 1
        _C.synthetic_env_step_into(
            env._state, action, env.A, env.B, env._t.reshape(-1),
            nxt[:, t0], prev[:, t0], rew[:, t0], done[:, t0], noise,
            float(env.max_steps),
        )
        assert done[:, t0].all()  # truncated
        assert torch.equal(prev[:, t0], state_before)  # pre-step obs stored
        # terminal obs stored ≠ carried state (which took the reset noise)
        expected_terminal = torch.tanh(
            state_before @ env.A + action.clamp(-1, 1) @ env.B
        )
        assert torch.allclose(nxt[:, t0], expected_terminal, atol=1e-5)
        assert torch.allclose(env._state, noise, atol=1e-6)  # reset carried
        assert (env._t == 0).all()  # t zeroed
        # non-truncating step: state advances normally, t increments
        _C.synthetic_env_step_into(
            env._state, action, env.A, env.B, env._t.reshape(-1),
            nxt[:, 2], prev[:, 2], rew[:, 2], done[:, 2], noise,
            float(env.max_steps),
        )
        assert not done[:, 2].any()
        assert (env._t == 1).all()
        assert torch.allclose(env._state, nxt[:, 2], atol=1e-6)


@pytest.mark.gpu
class TestTanhNormalLogProb:
    def test_matches_eager_with_grads(self):
        from rl_amd import ops
        from rl_amd.modules import TanhNormal

        torch.manual_seed(0)
        N, A = 4096, 6
        loc = torch.randn(N, A, device="cuda", requires_grad=True)
        scale = (torch.rand(N, A, device="cuda") * 0.9 + 0.1).requires_grad_()
        with torch.no_grad():
            action = TanhNormal(loc, scale).sample()
        lp_f = ops.tanh_normal_logprob(loc, scale, action)
        lp_e = TanhNormal(loc, scale).log_prob(action)
        assert torch.allclose(lp_f, lp_e, rtol=1e-4, atol=1e-4), (
            (lp_f - lp_e).abs().max().item()
        )
        g = torch.randn_like(lp_f)
        dl_f, ds_f = torch.autograd.grad(lp_f, (loc, scale), g, retain_graph=True)
        loc2 = loc.detach().clone().requires_grad_()
        scale2 = scale.detach().clone().requires_grad_()
        lp_e2 = TanhNormal(loc2, scale2).log_prob(action)
        dl_e, ds_e = torch.autograd.grad(lp_e2, (loc2, scale2), g)
        assert torch.allclose(dl_f, dl_e, rtol=1e-3, atol=1e-3), (
            (dl_f - dl_e).abs().max().item()
        )
        assert torch.allclose(ds_f, ds_e, rtol=1e-3, atol=1e-3), (
            (ds_f - ds_e).abs().max().item()
        )


@pytest.mark.gpu
class TestTanhNormalEntropy:
    def test_matches_eager_reparam_grads(self):
        from rl_amd import ops
        from rl_amd.modules import TanhNormal

        torch.manual_seed(0)
        N, A = 4096, 6
        loc = torch.randn(N, A, device="cuda", requires_grad=True)
        scale = (torch.rand(N, A, device="cuda") * 0.9 + 0.1).requires_grad_()
        eps = torch.randn(N, A, device="cuda")
        ent_f = ops.tanh_normal_entropy(loc, scale, eps)
        # eager: same sample via explicit reparameterization
        loc2 = loc.detach().clone().requires_grad_()
        scale2 = scale.detach().clone().requires_grad_()
        dist = TanhNormal(loc2, scale2)
        x = torch.tanh(loc2 + scale2 * eps)
        ent_e = -dist.log_prob(x)
        assert torch.allclose(ent_f, ent_e, rtol=1e-3, atol=1e-3), (
            (ent_f - ent_e).abs().max().item()
        )
        g = torch.randn_like(ent_f)
        dl_f, ds_f = torch.autograd.grad(ent_f, (loc, scale), g)
        dl_e, ds_e = torch.autograd.grad(ent_e, (loc2, scale2), g)
        assert torch.allclose(dl_f, dl_e, rtol=1e-3, atol=2e-3), (
            (dl_f - dl_e).abs().max().item()
        )
        assert torch.allclose(ds_f, ds_e, rtol=1e-3, atol=2e-3), (
            (ds_f - ds_e).abs().max().item()
        )


@pytest.mark.gpu
class TestFusedMLP3:
    def test_matches_eager_fwd_bwd(self):
        from rl_amd.ops import (
            convert_linears_to_splitk,
            enable_splitk_bf16_cache,
            fuse_mlp3,
            refresh_splitk_caches,
        )

        torch.manual_seed(0)
        O, H, A2, N = 17, 64, 12, 8192
        net = torch.nn.Sequential(
            torch.nn.Linear(O, H), torch.nn.Tanh(),
            torch.nn.Linear(H, H), torch.nn.Tanh(),
            torch.nn.Linear(H, A2),
        ).cuda()
        ref = torch.nn.Sequential(
            torch.nn.Linear(O, H), torch.nn.Tanh(),
            torch.nn.Linear(H, H), torch.nn.Tanh(),
            torch.nn.Linear(H, A2),
        ).cuda()
        ref.load_state_dict(net.state_dict())
        convert_linears_to_splitk(net)
        enable_splitk_bf16_cache(net)
        fused = fuse_mlp3(net)
        assert type(fused).__name__ == "FusedMLP3"
        x = torch.randn(N, O, device="cuda")
        with torch.autocast("cuda", dtype=torch.bfloat16, cache_enabled=False):
            y_f = fused(x.to(torch.bfloat16))
            y_e = ref(x.to(torch.bfloat16))
        assert torch.allclose(y_f.float(), y_e.float(), rtol=2e-2, atol=2e-2), (
            (y_f.float() - y_e.float()).abs().max().item()
        )
        # backward: same upstream grad, compare master-weight grads
        g = torch.randn_like(y_f, dtype=torch.float32)
        y_f.float().backward(g)
        y_e.float().backward(g)
        fused_linears = [fused.lin1, fused.lin2, fused.lin3]
        ref_linears = [m for m in ref if isinstance(m, torch.nn.Linear)]
        for lf, le in zip(fused_linears, ref_linears):
            assert lf.weight.grad is not None
            scale = le.weight.grad.abs().max().clamp_min(1e-3)
            err = (lf.weight.grad - le.weight.grad).abs().max() / scale
            assert err < 5e-2, err.item()
            berr = (lf.bias.grad - le.bias.grad).abs().max() / le.bias.grad.abs().max().clamp_min(1e-3)
            assert berr < 5e-2, berr.item()


@pytest.mark.gpu
class TestFusedRollout:
    def test_matches_per_step_kernels(self):
        """The whole-rollout mega-kernel must reproduce the per-step
        fused_actor_into + synthetic_env_step_into pair bit-for-bit
        (same eps/noise streams)."""
        from rl_amd import _C
        from rl_amd.envs.custom.synthetic import HalfCheetahVec
        from rl_amd.modules import MLP, NormalParamExtractor

        torch.manual_seed(0)
        B, T = 64, 12
        env1 = HalfCheetahVec(batch_size=[B], device="cuda", dtype=torch.float32)
        env2 = HalfCheetahVec(batch_size=[B], device="cuda", dtype=torch.float32)
        env1.set_seed(0); env2.set_seed(0)
        env1.reset(); env2.reset()
        env2._state.copy_(env1._state); env2._t.copy_(env1._t)
        env1.enable_capture_mode(True); env2.enable_capture_mode(True)
        S, A = env1.obs_dim, env1.act_dim
        net = torch.nn.Sequential(
            MLP(in_features=S, out_features=2 * A, num_cells=[64, 64],
                activation_class=torch.nn.Tanh, device="cuda"),
            NormalParamExtractor(),
        )
        lins = [m for m in net.modules() if isinstance(m, torch.nn.Linear)]
        w1, b1 = lins[0].weight, lins[0].bias
        w2, b2 = lins[1].weight, lins[1].bias
        w3, b3 = lins[2].weight, lins[2].bias
        isb = 0.5413248546129181
        eps = torch.randn(T, B, A, device="cuda")
        noise = torch.randn(T, B, S, device="cuda") * 0.1

        def alloc():
            return dict(
                obs=torch.zeros(B, T, S, device="cuda"),
                act=torch.zeros(B, T, A, device="cuda"),
                lp=torch.zeros(B, T, device="cuda"),
                nobs=torch.zeros(B, T, S, device="cuda"),
                rew=torch.zeros(B, T, 1, device="cuda"),
                done=torch.zeros(B, T, 1, dtype=torch.bool, device="cuda"),
            )

        st1, st2 = alloc(), alloc()
        # force a mid-rollout reset
        env1._t.fill_(float(env1.max_steps) - 5)
        env2._t.fill_(float(env2.max_steps) - 5)
        _C.fused_rollout(env1._state, env1._t.reshape(-1), w1, b1, w2, b2,
                         w3, b3, env1.A, env1.B, eps, noise,
                         st1["obs"], st1["act"], st1["lp"], st1["nobs"],
                         st1["rew"], st1["done"], float(env1.max_steps), isb, 1e-4,
                         [])
        for t in range(T):
            _C.fused_actor_into(env2._state, w1, b1, w2, b2, w3, b3, eps[t],
                                st2["act"][:, t], st2["lp"][:, t], isb, 1e-4)
            _C.synthetic_env_step_into(
                env2._state, st2["act"][:, t], env2.A, env2.B,
                env2._t.reshape(-1), st2["nobs"][:, t], st2["obs"][:, t],
                st2["rew"][:, t], st2["done"][:, t], noise[t],
                float(env2.max_steps))
        for k in st1:
            a, b = st1[k], st2[k]
            if a.dtype == torch.bool:
                assert torch.equal(a, b), k
            else:
                assert torch.allclose(a, b, atol=1e-5), (k, (a - b).abs().max())
        assert torch.allclose(env1._state, env2._state, atol=1e-5)
        assert torch.equal(env1._t, env2._t)
        assert st1["done"][:, 4].all()  # the forced reset happened


@pytest.mark.gpu
class TestPPOClipFused:
    """csrc/loss_ops.hip fused ClipPPO objective vs the eager chain."""

    def _eager(self, lw, adv, lo, hi, normalize):
        if normalize:
            adv = (adv - adv.mean()) / adv.std().clamp_min(1e-6)
        ratio = lw.exp()
        gain1 = ratio * adv
        rc = lw.clamp(lo, hi).exp()
        gain = torch.minimum(gain1, rc * adv)
        ess = lw.exp().sum().pow(2) / lw.mul(2).exp().sum().clamp_min(1e-12)
        clip_frac = (rc != ratio).float().mean()
        return -gain.mean(), ess / lw.numel(), clip_frac

    @pytest.mark.parametrize("normalize", [False, True])
    def test_matches_eager(self, normalize):
        from rl_amd import ops

        torch.manual_seed(0)
        N = 4097
        lw = (0.3 * torch.randn(N, device="cuda")).requires_grad_()
        adv = torch.randn(N, device="cuda")
        lo, hi = math.log1p(-0.2), math.log1p(0.2)
        loss, ess, cf = ops.ppo_clip_objective(lw, adv, lo, hi, normalize)
        lw2 = lw.detach().clone().requires_grad_()
        e_loss, e_ess, e_cf = self._eager(lw2, adv, lo, hi, normalize)
        assert torch.allclose(loss, e_loss, atol=1e-5), (loss, e_loss)
        assert torch.allclose(ess, e_ess, rtol=1e-4)
        assert torch.allclose(cf, e_cf)
        g = torch.randn((), device="cuda")
        loss.backward(g)
        e_loss.backward(g)
        assert torch.allclose(lw.grad, lw2.grad, atol=1e-6), (
            (lw.grad - lw2.grad).abs().max()
        )

    def test_loss_module_fused_vs_eager(self):
        """ClipPPOLoss end-to-end: fused HIP path vs forced-eager."""
        from rl_amd import ops
        from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal, ValueOperator
        from rl_amd.objectives import ClipPPOLoss
        from rl_amd.data.tensor_specs import Bounded
        from rl_amd.tensordict import TensorDict, TensorDictModule

        torch.manual_seed(0)
        dev = "cuda"
        actor = ProbabilisticActor(
            TensorDictModule(
                torch.nn.Sequential(
                    MLP(in_features=4, out_features=4, num_cells=[16], device=dev),
                    NormalParamExtractor(),
                ),
                in_keys=["observation"],
                out_keys=["loc", "scale"],
            ),
            in_keys=["loc", "scale"],
            distribution_class=TanhNormal,
            return_log_prob=True,
            spec=Bounded(-1.0, 1.0, shape=(2,), device=dev),
        )
        critic = ValueOperator(
            MLP(in_features=4, out_features=1, num_cells=[16], device=dev),
            in_keys=["observation"],
        )
        N = 256
        td = TensorDict(
            {
                "observation": torch.randn(N, 4, device=dev),
                "action": torch.rand(N, 2, device=dev) * 1.6 - 0.8,
                "sample_log_prob": torch.randn(N, device=dev) * 0.1,
                "advantage": torch.randn(N, 1, device=dev),
                "value_target": torch.randn(N, 1, device=dev),
            },
            batch_size=[N],
        )
        loss_mod = ClipPPOLoss(actor, critic, normalize_advantage=True).to(dev)
        out_fused = loss_mod(td.clone())
        # force the eager path
        old = ops.HAS_HIP_EXT
        ops.HAS_HIP_EXT = False
        try:
            out_eager = loss_mod(td.clone())
        finally:
            ops.HAS_HIP_EXT = old
        for k in ("loss_objective", "clip_fraction", "ESS", "loss_critic"):
            a, b = out_fused.get(k), out_eager.get(k)
            assert torch.allclose(a, b, atol=2e-5), (k, a, b)
        # gradients through the actor match
        ga = torch.autograd.grad(
            out_fused.get("loss_objective"), list(actor.parameters()),
            retain_graph=False, allow_unused=True,
        )
        ops.HAS_HIP_EXT = False
        try:
            out_eager2 = loss_mod(td.clone())
        finally:
            ops.HAS_HIP_EXT = old
        gb = torch.autograd.grad(
            out_eager2.get("loss_objective"), list(actor.parameters()),
            allow_unused=True,
        )
        for x, y in zip(ga, gb):
            if x is None:
                assert y is None
                continue
            assert torch.allclose(x, y, atol=1e-5), (x - y).abs().max()


@pytest.mark.gpu
class TestSmoothL1Fused:
    @pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
    def test_matches_eager(self, dtype):
        from rl_amd import ops

        torch.manual_seed(1)
        N = 3001
        v = (torch.randn(N, device="cuda") * 2).to(dtype).requires_grad_()
        t = torch.randn(N, device="cuda") * 2
        loss = ops.smooth_l1_mean(v, t)
        v2 = v.detach().clone().requires_grad_()
        e = torch.nn.functional.smooth_l1_loss(v2.float(), t, reduction="mean")
        assert torch.allclose(loss, e, atol=1e-5)
        loss.backward()
        e.backward()
        atol = 1e-6 if dtype == torch.float32 else 1e-2
        assert torch.allclose(v.grad.float(), v2.grad.float(), atol=atol)


@pytest.mark.gpu
class TestPPOHeadMega:
    """csrc/loss_ops.hip mega-fused head loss vs the eager chain."""

    def _eager(self, head, action, prev, adv, eps, bias, lb, lo, hi, coeff,
               normalize):
        A = action.shape[-1]
        loc, sraw = head.float().chunk(2, -1)
        scale = torch.nn.functional.softplus(sraw + bias).clamp_min(lb)
        if normalize:
            adv = (adv - adv.mean()) / adv.std().clamp_min(1e-6)
        lim = 1.0 - 1.1920929e-7
        y = action.clamp(-lim, lim)
        u = torch.atanh(y)
        z = (u - loc) / scale
        jac = 2.0 * (math.log(2.0) - u - torch.nn.functional.softplus(-2 * u))
        lp = (-0.5 * z * z - scale.log() - 0.5 * math.log(2 * math.pi) - jac).sum(-1)
        lw = lp - prev
        ratio = lw.exp()
        rc = lw.clamp(lo, hi).exp()
        gain = torch.minimum(ratio * adv, rc * adv)
        x = torch.tanh(loc + scale * eps)
        ent = (0.5 * eps * eps + scale.log() + 0.5 * math.log(2 * math.pi)
               + torch.log1p(-x * x)).sum(-1)
        ess = lw.exp().sum().pow(2) / lw.mul(2).exp().sum().clamp_min(1e-12)
        cf = (rc != ratio).float().mean()
        return (-gain.mean(), -coeff * ent.mean(), ent.mean().detach(),
                ess / lw.numel(), cf)

    @pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
    @pytest.mark.parametrize("normalize", [False, True])
    def test_matches_eager(self, dtype, normalize):
        from rl_amd import ops

        torch.manual_seed(0)
        N, A = 2049, 6
        head = (0.4 * torch.randn(N, 2 * A, device="cuda")).to(dtype).requires_grad_()
        action = (torch.rand(N, A, device="cuda") * 1.8 - 0.9)
        prev = torch.randn(N, device="cuda") * 0.1
        adv = torch.randn(N, device="cuda")
        eps = torch.randn(N, A, device="cuda")
        bias, lb = 0.5413248546129181, 1e-4
        lo, hi = math.log1p(-0.2), math.log1p(0.2)
        o_f = ops.ppo_head_loss(head, action, prev, adv, eps, sp_bias=bias,
                                scale_lb=lb, lo=lo, hi=hi, entropy_coeff=0.01,
                                normalize=normalize)[:6]
        head2 = head.detach().clone().requires_grad_()
        o_e = self._eager(head2, action, prev, adv, eps, bias, lb, lo, hi,
                          0.01, normalize)
        tol = 1e-5 if dtype == torch.float32 else 5e-3
        for i, (a, b) in enumerate(zip(o_f, o_e)):
            assert torch.allclose(a, b, atol=tol, rtol=1e-3), (i, a, b)
        # 6th output: kernel-side pre-summed actor loss
        assert torch.allclose(o_f[5], o_f[0] + o_f[1], atol=1e-6)
        (o_f[0] + o_f[1]).backward()
        (o_e[0] + o_e[1]).backward()
        gtol = 1e-5 if dtype == torch.float32 else 1e-2
        assert torch.allclose(head.grad.float(), head2.grad.float(),
                              atol=gtol), (head.grad - head2.grad).abs().max()

    def test_objective_only_grad(self):
        """g_ent=None path (autograd.grad on loss_objective alone)."""
        from rl_amd import ops

        torch.manual_seed(1)
        N, A = 513, 3
        head = (0.3 * torch.randn(N, 2 * A, device="cuda")).requires_grad_()
        action = torch.rand(N, A, device="cuda") * 1.6 - 0.8
        prev = torch.randn(N, device="cuda") * 0.1
        adv = torch.randn(N, device="cuda")
        eps = torch.randn(N, A, device="cuda")
        lo, hi = math.log1p(-0.2), math.log1p(0.2)
        o = ops.ppo_head_loss(head, action, prev, adv, eps,
                              sp_bias=0.5413248546129181, scale_lb=1e-4,
                              lo=lo, hi=hi, entropy_coeff=0.01,
                              normalize=True)[:6]
        (g,) = torch.autograd.grad(o[0], head)
        head2 = head.detach().clone().requires_grad_()
        o_e = TestPPOHeadMega._eager(self, head2, action, prev, adv, eps,
                                     0.5413248546129181, 1e-4, lo, hi, 0.01,
                                     True)
        (g_e,) = torch.autograd.grad(o_e[0], head2)
        assert torch.allclose(g, g_e, atol=1e-5), (g - g_e).abs().max()


@pytest.mark.gpu
class TestActorCriticFused:
    """Dual-network fwd2/bwd2 + 6-layer batched wgrad vs two separate
    FusedMLP3 passes (csrc/fused_mlp.hip)."""

    def _mk(self, O, H, A2, seed):
        from rl_amd.ops import convert_linears_to_splitk, enable_splitk_bf16_cache, fuse_mlp3

        torch.manual_seed(seed)
        net = torch.nn.Sequential(
            torch.nn.Linear(O, H), torch.nn.Tanh(),
            torch.nn.Linear(H, H), torch.nn.Tanh(),
            torch.nn.Linear(H, A2),
        ).cuda()
        convert_linears_to_splitk(net)
        enable_splitk_bf16_cache(net)
        return fuse_mlp3(net)

    def test_matches_separate(self):
        from rl_amd import ops

        O, H, N = 17, 64, 4096
        actor = self._mk(O, H, 12, 0)
        critic = self._mk(O, H, 1, 1)
        x = torch.randn(N, O, device="cuda")
        head, value = ops.actor_critic_mlp3(x, actor, critic)
        h_ref = actor(x)
        v_ref = critic(x)
        assert torch.equal(head, h_ref)  # same kernels, same math
        assert torch.equal(value, v_ref)
        # backward: compare master-weight grads against separate passes
        g_h = torch.randn_like(head, dtype=torch.float32)
        g_v = torch.randn_like(value, dtype=torch.float32)
        (head.float() * g_h).sum().backward(retain_graph=True)
        (value.float() * g_v).sum().backward()
        fused_grads = [p.grad.clone() for p in list(actor.parameters()) + list(critic.parameters())]
        for p in list(actor.parameters()) + list(critic.parameters()):
            p.grad = None
        (h_ref.float() * g_h).sum().backward(retain_graph=True)
        (v_ref.float() * g_v).sum().backward()
        for g1, p in zip(fused_grads, list(actor.parameters()) + list(critic.parameters())):
            assert torch.allclose(g1, p.grad, atol=1e-4, rtol=1e-3), (
                (g1 - p.grad).abs().max()
            )


@pytest.mark.gpu
def test_grad_scale_clip_equals_explicit_clip():
    """GraphedPPO's clip-via-grad_scale: fused Adam with grad_scale =
    max(1, norm/max_norm) must match clip_grad_norm_ + step."""
    from rl_amd import ops

    torch.manual_seed(0)
    p1 = torch.randn(100, device="cuda", requires_grad=True)
    p2 = p1.detach().clone().requires_grad_()
    g = torch.randn(100, device="cuda") * 3
    max_norm = 1.0

    o1 = torch.optim.Adam([p1], lr=1e-2, fused=True)
    p1.grad = g.clone()
    scale = torch.ones((), device="cuda")
    o1.grad_scale = scale
    assert ops.fused_grad_clip_scale_([p1], max_norm, scale)
    o1.step()

    o2 = torch.optim.Adam([p2], lr=1e-2, fused=True)
    p2.grad = g.clone()
    torch.nn.utils.clip_grad_norm_([p2], max_norm)
    o2.step()
    assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


@pytest.mark.gpu
def test_ppo_head_with_critic_fused():
    """Critic smooth-L1 riding in the head kernels: outputs and BOTH
    gradients (d head, d value) match the separate eager chain."""
    from rl_amd import ops

    torch.manual_seed(3)
    N, A = 1537, 6
    head = (0.4 * torch.randn(N, 2 * A, device="cuda")).to(torch.bfloat16).requires_grad_()
    value = torch.randn(N, device="cuda").to(torch.bfloat16).requires_grad_()
    target = torch.randn(N, device="cuda")
    action = torch.rand(N, A, device="cuda") * 1.6 - 0.8
    prev = torch.randn(N, device="cuda") * 0.1
    adv = torch.randn(N, device="cuda")
    eps = torch.randn(N, A, device="cuda")
    lo, hi = math.log1p(-0.2), math.log1p(0.2)
    kw = dict(sp_bias=0.5413248546129181, scale_lb=1e-4, lo=lo, hi=hi,
              entropy_coeff=0.01, normalize=True)
    o = ops.ppo_head_loss(head, action, prev, adv, eps, value=value,
                          value_target=target, critic_scale=0.5, **kw)
    l_obj, l_ent, _, _, _, l_act, l_crit, l_tot = o
    # reference: separate fused paths on cloned leaves
    head2 = head.detach().clone().requires_grad_()
    value2 = value.detach().clone().requires_grad_()
    o2 = ops.ppo_head_loss(head2, action, prev, adv, eps, **kw)
    crit_ref = ops.smooth_l1_mean(value2, target, 0.5)
    assert torch.allclose(l_crit, crit_ref, atol=1e-5)
    assert torch.allclose(l_tot, o2[5] + crit_ref, atol=1e-5)
    l_tot.backward()
    (o2[5] + crit_ref).backward()
    assert torch.allclose(head.grad.float(), head2.grad.float(), atol=1e-5)
    assert torch.allclose(value.grad.float(), value2.grad.float(), atol=1e-5)


@pytest.mark.gpu
def test_value_pair_eval_matches_two_calls():
    from rl_amd import ops
    from rl_amd.ops import convert_linears_to_splitk, enable_splitk_bf16_cache, fuse_mlp3

    torch.manual_seed(0)
    net = torch.nn.Sequential(
        torch.nn.Linear(17, 64), torch.nn.Tanh(),
        torch.nn.Linear(64, 64), torch.nn.Tanh(),
        torch.nn.Linear(64, 1),
    ).cuda()
    convert_linears_to_splitk(net)
    enable_splitk_bf16_cache(net)
    fused = fuse_mlp3(net)
    x0 = torch.randn(3000, 17, device="cuda")
    x1 = torch.randn(3000, 17, device="cuda")
    y0, y1 = ops.value_pair_eval(fused, x0, x1)
    with torch.no_grad():
        r0, r1 = fused(x0), fused(x1)
    assert torch.equal(y0, r0) and torch.equal(y1, r1)


@pytest.mark.gpu
def test_fused_rollout_mfma_vs_reference():
    """MFMA rollout variant (policy on the matrix cores with bf16
    weight caches): every step's action/log-prob must match a torch
    reference computed from the STORED per-step observation (so state
    drift cannot compound), and the fp32 env transition must match
    exactly."""
    import rl_amd._C as _C
    from rl_amd.envs.custom.synthetic import HalfCheetahVec

    torch.manual_seed(0)
    B, T, H = 256, 6, 64
    env = HalfCheetahVec(batch_size=[B], device="cuda")
    env.reset()
    S, A = env.obs_dim, env.act_dim
    w1 = torch.randn(H, S, device="cuda") * 0.2
    b1 = torch.randn(H, device="cuda") * 0.1
    w2 = torch.randn(H, H, device="cuda") * 0.2
    b2 = torch.randn(H, device="cuda") * 0.1
    w3 = torch.randn(2 * A, H, device="cuda") * 0.2
    b3 = torch.randn(2 * A, device="cuda") * 0.1
    bf = [t.to(torch.bfloat16) for t in (w1, b1, w2, b2, w3, b3)]
    eps = torch.randn(T, B, A, device="cuda")
    noise = torch.randn(T, B, S, device="cuda") * 0.1
    st = dict(
        obs=torch.zeros(B, T, S, device="cuda"),
        act=torch.zeros(B, T, A, device="cuda"),
        lp=torch.zeros(B, T, device="cuda"),
        nobs=torch.zeros(B, T, S, device="cuda"),
        rew=torch.zeros(B, T, 1, device="cuda"),
        done=torch.zeros(B, T, 1, dtype=torch.bool, device="cuda"),
    )
    isb = 0.5413248546129181
    _C.fused_rollout(env._state, env._t.reshape(-1), w1, b1, w2, b2, w3, b3,
                     env.A, env.B, eps, noise, st["obs"], st["act"],
                     st["lp"], st["nobs"], st["rew"], st["done"],
                     float(env.max_steps), isb, 1e-4, bf)
    for t in range(T):
        obs = st["obs"][:, t]
        h1 = torch.tanh(
            obs.to(torch.bfloat16).float() @ bf[0].float().t() + bf[1].float()
        ).to(torch.bfloat16)
        h2 = torch.tanh(h1.float() @ bf[2].float().t() + bf[3].float()).to(
            torch.bfloat16
        )
        head = h2.float() @ bf[4].float().t() + bf[5].float()
        loc, sraw = head.chunk(2, -1)
        scale = torch.nn.functional.softplus(sraw + isb).clamp_min(1e-4)
        u = loc + scale * eps[t]
        act_ref = torch.tanh(u).clamp(-1 + 1.2e-7, 1 - 1.2e-7)
        assert torch.allclose(st["act"][:, t], act_ref, atol=3e-3), (
            t, (st["act"][:, t] - act_ref).abs().max()
        )
        lp_ref = (
            -0.5 * eps[t] ** 2 - scale.log() - 0.5 * math.log(2 * math.pi)
            - 2.0 * (math.log(2.0) - u - torch.nn.functional.softplus(-2 * u))
        ).sum(-1)
        assert torch.allclose(st["lp"][:, t], lp_ref, atol=2e-2, rtol=1e-3), (
            t, (st["lp"][:, t] - lp_ref).abs().max()
        )
        # env transition is fp32 on the STORED action: tight tolerance
        nobs_ref = torch.tanh(obs @ env.A + st["act"][:, t] @ env.B)
        assert torch.allclose(st["nobs"][:, t], nobs_ref, atol=1e-5)


@pytest.mark.gpu
def test_merged_acloss_matches_split_path():
    """The whole-minibatch merged Function (acloss kernels) vs the
    split fwd2 + head-loss path: all 8 scalars and every master-weight
    gradient."""
    import os

    from rl_amd import ops
    from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal, ValueOperator
    from rl_amd.objectives import ClipPPOLoss
    from rl_amd.data.tensor_specs import Bounded
    from rl_amd.tensordict import TensorDict, TensorDictModule
    from rl_amd.ops import convert_linears_to_splitk, enable_splitk_bf16_cache, fuse_mlp3

    def build(seed):
        torch.manual_seed(seed)
        dev = "cuda"
        actor = ProbabilisticActor(
            TensorDictModule(
                torch.nn.Sequential(
                    MLP(in_features=17, out_features=12, num_cells=[64, 64], device=dev),
                    NormalParamExtractor(),
                ),
                in_keys=["observation"], out_keys=["loc", "scale"],
            ),
            in_keys=["loc", "scale"], distribution_class=TanhNormal,
            return_log_prob=True,
            spec=Bounded(-1.0, 1.0, shape=(6,), device=dev),
        )
        critic = ValueOperator(
            MLP(in_features=17, out_features=1, num_cells=[64, 64], device=dev),
            in_keys=["observation"],
        )
        convert_linears_to_splitk(actor)
        convert_linears_to_splitk(critic)
        enable_splitk_bf16_cache(actor)
        enable_splitk_bf16_cache(critic)
        actor.module[0].module[0] = fuse_mlp3(actor.module[0].module[0])
        critic.module = fuse_mlp3(critic.module)
        return ClipPPOLoss(actor, critic, normalize_advantage=True,
                           critic_coeff=0.5).to(dev)

    N = 2048
    torch.manual_seed(7)
    td = TensorDict(
        {
            "observation": torch.randn(N, 17, device="cuda"),
            "action": torch.rand(N, 6, device="cuda") * 1.6 - 0.8,
            "sample_log_prob": torch.randn(N, device="cuda") * 0.1,
            "advantage": torch.randn(N, 1, device="cuda"),
            "value_target": torch.randn(N, 1, device="cuda"),
        },
        batch_size=[N],
    )
    outs, grads = {}, {}
    for mode in ("1", "0"):
        os.environ["RL_AMD_MERGED_LOSS"] = mode
        loss_mod = build(3)
        torch.manual_seed(11)  # same eps draw
        out = loss_mod(td.clone())
        total = out.get("_loss_total", None)
        if total is None:
            total = (out.get("loss_objective") + out.get("loss_entropy")
                     + out.get("loss_critic"))
        g = torch.autograd.grad(total, list(loss_mod.parameters()))
        outs[mode] = {k: out.get(k).detach() for k in
                      ("loss_objective", "loss_entropy", "loss_critic",
                       "ESS", "clip_fraction", "entropy")}
        grads[mode] = g
    os.environ.pop("RL_AMD_MERGED_LOSS", None)
    for k in outs["1"]:
        assert torch.allclose(outs["1"][k], outs["0"][k], atol=2e-4, rtol=1e-3), (
            k, outs["1"][k], outs["0"][k]
        )
    for g1, g0 in zip(grads["1"], grads["0"]):
        assert torch.allclose(g1, g0, atol=2e-3, rtol=1e-2), (g1 - g0).abs().max()


@pytest.mark.gpu
def test_wgrad_sq_partials_clip_coefficient():
    """Clip coefficient from the wgrad reduce's sq partials must equal
    the true global gradient norm's coefficient."""
    import rl_amd._C as _C
    from rl_amd import ops
    from rl_amd.ops import convert_linears_to_splitk, enable_splitk_bf16_cache, fuse_mlp3

    torch.manual_seed(5)

    def mk(a2, seed):
        torch.manual_seed(seed)
        net = torch.nn.Sequential(
            torch.nn.Linear(17, 64), torch.nn.Tanh(),
            torch.nn.Linear(64, 64), torch.nn.Tanh(),
            torch.nn.Linear(64, a2),
        ).cuda()
        convert_linears_to_splitk(net)
        enable_splitk_bf16_cache(net)
        return fuse_mlp3(net)

    actor, critic = mk(12, 0), mk(1, 1)
    N = 2048
    x = torch.randn(N, 17, device="cuda")
    action = torch.rand(N, 6, device="cuda") * 1.6 - 0.8
    prev = torch.randn(N, device="cuda") * 0.1
    adv = torch.randn(N, device="cuda")
    vt = torch.randn(N, device="cuda")
    eps = torch.randn(N, 6, device="cuda")
    sq = torch.empty(6 * 512, device="cuda")
    out = ops.actor_critic_loss(
        x, actor, critic, action, prev, adv, vt, eps,
        sp_bias=0.5413248546129181, scale_lb=1e-4,
        lo=math.log1p(-0.2), hi=math.log1p(0.2),
        entropy_coeff=0.01, critic_scale=0.5, normalize=True,
        gradsq_out=sq,
    )
    out[7].backward()  # loss_total
    params = list(actor.parameters()) + list(critic.parameters())
    true_norm = torch.sqrt(sum(p.grad.float().pow(2).sum() for p in params))
    coef = torch.empty((), device="cuda")
    _C.wgrad_clip_finalize(sq, 1.0, coef, True)
    expect = torch.clamp((true_norm + 1e-6) / 1.0, min=1.0)
    assert torch.allclose(coef, expect, rtol=1e-4), (coef, expect)


@pytest.mark.gpu
def test_multi_shuffle_is_consistent_permutation():
    """Feistel shuffle: a real permutation of [0, n), identical across
    tensors, fresh for fresh keys."""
    from rl_amd import ops
    from rl_amd.tensordict import TensorDict

    for n in (4096, 5000):  # power of two + cycle-walking case
        idx = torch.arange(n, device="cuda", dtype=torch.float32)
        td = TensorDict(
            {"a": idx.clone(), "b": torch.stack([idx, idx * 2], -1).contiguous()},
            batch_size=[n],
        )
        keys = torch.randint(-2**31, 2**31 - 1, (4,), device="cuda", dtype=torch.int32)
        out = ops.multi_shuffle_td(td, keys)
        a = out.get("a")
        # permutation property
        assert torch.equal(torch.sort(a).values, idx)
        assert not torch.equal(a, idx)  # actually shuffled
        # same permutation applied to every tensor
        assert torch.equal(out.get("b")[:, 0], a)
        assert torch.equal(out.get("b")[:, 1], a * 2)
        # fresh keys -> different permutation
        keys2 = keys + 12345
        out2 = ops.multi_shuffle_td(td, keys2)
        assert not torch.equal(out2.get("a"), a)


@pytest.mark.gpu
def test_adv_stats_batch_matches_torch():
    from rl_amd import ops

    torch.manual_seed(0)
    n_mb, mb = 4, 16384
    adv = torch.randn(n_mb * mb, device="cuda") * 3 + 1
    stats = ops.adv_stats_batch(adv, n_mb)
    for i in range(n_mb):
        sl = adv[i * mb : (i + 1) * mb]
        assert torch.allclose(stats[i, 0], sl.mean(), atol=1e-4)
        assert torch.allclose(stats[i, 1], 1.0 / sl.std().clamp_min(1e-6),
                              rtol=1e-4)


@pytest.mark.gpu
def test_gae_paired_value_eval_matches_eager_critic():
    """GAE through a cached FusedMLP3 critic (ONE paired launch for
    value/next_value) must match GAE through the identical eager
    critic."""
    import copy

    from rl_amd.modules import MLP, ValueOperator
    from rl_amd.objectives.value.advantages import GAE
    from rl_amd.ops import convert_linears_to_splitk, enable_splitk_bf16_cache, fuse_mlp3
    from rl_amd.tensordict import TensorDict

    torch.manual_seed(0)
    dev = "cuda"
    critic = ValueOperator(
        MLP(in_features=17, out_features=1, num_cells=[64, 64], device=dev),
        in_keys=["observation"],
    )
    critic_ref = copy.deepcopy(critic)
    convert_linears_to_splitk(critic)
    enable_splitk_bf16_cache(critic)
    critic.module = fuse_mlp3(critic.module)
    B, T = 32, 16
    td = TensorDict(
        {
            "observation": torch.randn(B, T, 17, device=dev),
            "next": {
                "observation": torch.randn(B, T, 17, device=dev),
                "reward": torch.randn(B, T, 1, device=dev),
                "done": torch.rand(B, T, 1, device=dev) < 0.05,
                "terminated": torch.rand(B, T, 1, device=dev) < 0.02,
            },
        },
        batch_size=[B, T],
    )
    g1 = GAE(gamma=0.99, lmbda=0.95, value_network=critic, vectorized=True)
    g2 = GAE(gamma=0.99, lmbda=0.95, value_network=critic_ref, vectorized=True)
    with torch.no_grad():
        a1 = g1(td.clone()).get("advantage")
        with torch.autocast("cuda", dtype=torch.bfloat16, cache_enabled=False):
            a2 = g2(td.clone()).get("advantage")
    # identical weights; paired path runs bf16 caches vs eager autocast
    assert torch.allclose(a1, a2, atol=3e-2, rtol=1e-2), (a1 - a2).abs().max()
