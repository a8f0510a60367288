"""rl_amd.ops — dispatch layer over the native extension (rl_amd._C).

On a HIP device the fused CDNA4 kernels run; on CPU the torch scan
implementations (objectives/value/functional.py) are the fallback.  If a
GPU is present but the extension is missing, ops raise — GPU runs must
never silently fall back to eager torch (that would be an unmeasured
perf regression masquerading as success).
"""
from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

_C = None
_C_ERR: Optional[str] = None
try:
    from rl_amd import _C  # type: ignore
except Exception as e:  # pragma: no cover
    _C_ERR = repr(e)

HAS_EXT = _C is not None
HAS_HIP_EXT = bool(HAS_EXT and getattr(_C, "COMPILED_WITH_HIP", False))

_ALLOW_FALLBACK = os.environ.get("RL_AMD_ALLOW_EAGER_FALLBACK", "0") == "1"


def _require_ext():
    if not HAS_HIP_EXT and not _ALLOW_FALLBACK:
        raise RuntimeError(
            "rl_amd._C HIP extension not available on a GPU device "
            f"(import error: {_C_ERR}). Build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950), "
            "or set RL_AMD_ALLOW_EAGER_FALLBACK=1 to run the slow torch path."
        )
    return HAS_HIP_EXT


def ext_module():
    return _C


def _flat_time(x: torch.Tensor) -> torch.Tensor:
    """[*, T, 1] or [*, T] → contiguous [*, T] view."""
    if x.shape[-1] == 1 and x.dim() >= 2:
        x = x.squeeze(-1)
    return x.contiguous()


def gae(
    gamma: float,
    lmbda: float,
    state_value: torch.Tensor,
    next_state_value: torch.Tensor,
    reward: torch.Tensor,
    done: torch.Tensor,
    terminated: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Fused GAE: returns (advantage, value_target) with the input shape."""
    if reward.is_cuda and _require_ext():
        shape = reward.shape
        r = _flat_time(reward)
        v = _flat_time(state_value).to(r.dtype)
        nv = _flat_time(next_state_value).to(r.dtype)
        d = _flat_time(done)
        tm = _flat_time(terminated)
        adv, vt = _C.gae(r, v, nv, d, tm, float(gamma), float(lmbda))
        return adv.reshape(shape), vt.reshape(shape)
    from ..objectives.value import functional as F

    return F.vec_generalized_advantage_estimate(
        gamma, lmbda, state_value, next_state_value, reward, done, terminated
    )


def revscan(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """y[t] = b[t] + a[t]·y[t+1] over the last dim."""
    if a.is_cuda and _require_ext():
        return _C.revscan(a.contiguous(), b.contiguous())
    from ..objectives.value.functional import _reverse_scan

    return _reverse_scan(b.unsqueeze(-1), a.unsqueeze(-1)).squeeze(-1)


def vtrace(
    gamma: float,
    log_pi: torch.Tensor,
    log_mu: torch.Tensor,
    state_value: torch.Tensor,
    next_state_value: torch.Tensor,
    reward: torch.Tensor,
    done: torch.Tensor,
    terminated: torch.Tensor,
    rho_thresh: float = 1.0,
    c_thresh: float = 1.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    if reward.is_cuda and _require_ext():
        shape = reward.shape
        adv, vs = _C.vtrace(
            _flat_time(log_pi).float(),
            _flat_time(log_mu).float(),
            _flat_time(reward).float(),
            _flat_time(state_value).float(),
            _flat_time(next_state_value).float(),
            _flat_time(done),
            _flat_time(terminated),
            float(gamma),
            float(rho_thresh),
            float(c_thresh),
        )
        return adv.reshape(shape), vs.reshape(shape)
    from ..objectives.value import functional as F

    return F.vec_vtrace_advantage_estimate(
        gamma,
        log_pi,
        log_mu,
        state_value,
        next_state_value,
        reward,
        done,
        terminated,
        rho_thresh,
        c_thresh,
    )


def safetanh(x: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    if HAS_EXT:
        return _C.safetanh(x, eps)
    from ..modules.distributions.continuous import safetanh as py_safetanh

    return py_safetanh(x, eps)


class DeviceSumTree:
    """HBM-resident sum(+min) segment tree pair driven by the fused HIP
    kernels (scan descent + two-pass update).  Numerics validated against
    the torch trees in tests/test_ops.py."""

    def __init__(self, capacity: int, device, with_min: bool = True):
        _require_ext()
        size = 1
        while size < capacity:
            size *= 2
        self.capacity = capacity
        self.size = size
        self.device = torch.device(device)
        self.sum_tree = torch.zeros(2 * size, dtype=torch.float64, device=self.device)
        self.min_tree = torch.full(
            (2 * size,), float("inf"), dtype=torch.float64, device=self.device
        )
        self.with_min = with_min
        self._cnt = torch.zeros(2 * size, dtype=torch.int32, device=self.device)

    def update(self, index: torch.Tensor, value: torch.Tensor) -> None:
        index = index.to(self.device, torch.long).reshape(-1)
        value = value.to(self.device, torch.float64).reshape(-1)
        if index.numel() == 0:
            return
        # last-writer-wins dedup (same semantics as the serialized
        # reference write, cuda_segment_tree.cu:27-37)
        order = torch.arange(index.numel(), device=self.device)
        winner = torch.full((self.size,), -1, dtype=torch.long, device=self.device)
        winner.scatter_reduce_(0, index, order, reduce="amax")
        sel = winner[index] == order
        index = index[sel]
        value = value[sel]
        _C.tree_update(
            self.sum_tree, self.min_tree, self._cnt, index, value, self.size, self.with_min
        )

    def total(self) -> torch.Tensor:
        return self.sum_tree[1]

    def min(self) -> torch.Tensor:
        return self.min_tree[1]

    def scan_lower_bound(self, mass: torch.Tensor) -> torch.Tensor:
        mass = mass.to(self.device, torch.float64).reshape(-1)
        return _C.tree_scan_lower_bound(self.sum_tree, mass, self.size, self.capacity)

    def get(self, index: torch.Tensor) -> torch.Tensor:
        return self.sum_tree[self.size + index.to(self.device, torch.long)]


def gru_fused(cell, x: torch.Tensor, is_init: torch.Tensor, h0: Optional[torch.Tensor] = None):
    """Fused GRU forward over [B, T, F] with per-step resets.

    Matches rnn.gru_scan numerics (tests/test_rnn.py); forward-only."""
    _require_ext()
    B, T = x.shape[0], x.shape[1]
    H = cell.hidden_size
    gates_x = (x.float() @ cell.weight_ih.T.float() + cell.bias_ih.float()).contiguous()
    ii = is_init.squeeze(-1) if is_init.dim() == 3 else is_init
    ys, h = _C.gru_fused(
        gates_x,
        cell.weight_hh.detach().float().contiguous(),
        cell.bias_hh.detach().float().contiguous(),
        ii.contiguous(),
        h0.float().contiguous() if h0 is not None else torch.Tensor(),
    )
    return ys.to(x.dtype), h.to(x.dtype)


def lstm_fused(cell, x: torch.Tensor, is_init: torch.Tensor, h0=None, c0=None):
    _require_ext()
    H = cell.hidden_size
    gates_x = (
        x.float() @ cell.weight_ih.T.float() + cell.bias_ih.float() + cell.bias_hh.float()
    ).contiguous()
    ii = is_init.squeeze(-1) if is_init.dim() == 3 else is_init
    ys, h, c = _C.lstm_fused(
        gates_x,
        cell.weight_hh.detach().float().contiguous(),
        ii.contiguous(),
        h0.float().contiguous() if h0 is not None else torch.Tensor(),
        c0.float().contiguous() if c0 is not None else torch.Tensor(),
    )
    return ys.to(x.dtype), h.to(x.dtype), c.to(x.dtype)


def gru_train_lds_ok(H: int) -> bool:
    """True when W_hh^T fits LDS (else the scan streams it from L2 —
    still a single launch, just slower weight reads).  Informational."""
    return H * (3 * H + 2) * 2 + 8 * 6 * H * 4 <= 160 * 1024


def lstm_train_lds_ok(H: int) -> bool:
    return H * (4 * H + 2) * 2 + 4 * 8 * H * 4 <= 160 * 1024


class _GRUFusedTrainFn(torch.autograd.Function):
    """Differentiable fused GRU scan: forward = the single-launch HIP
    scan (csrc/rnn_scan.hip gru_train_fwd_kernel), backward = the
    reverse-time gate-RECOMPUTE scan (gru_bwd_kernel) + ONE GEMM for
    the [3H, H] weight gradient (dgates_h^T @ h_prev).
    Reference capability: _rnn_triton.py:329 (_gru_bwd_kernel)."""

    @staticmethod
    def forward(ctx, gates_x, w_hh, bias_hh, is_init, h0):
        # pre-transposed bf16 weight (k-major) shared by fwd and bwd;
        # row-major bf16 copy feeds the backward's carry GEMM
        wt = w_hh.detach().t().contiguous().to(torch.bfloat16)
        w_row = w_hh.detach().contiguous().to(torch.bfloat16)
        bias = bias_hh.detach().contiguous()
        ys = _C.gru_train_fwd(gates_x, wt, bias, is_init, h0)
        ctx.save_for_backward(gates_x, wt, w_row, bias, is_init, h0, ys)
        return ys, ys[:, -1]

    @staticmethod
    def backward(ctx, dys, dh_last):
        gates_x, wt, w_row, bias, is_init, h0, ys = ctx.saved_tensors
        dys = dys.contiguous()
        if dh_last is not None and dh_last.numel():
            dys = dys.clone()
            dys[:, -1] += dh_last
        dgx, dgh, hprev, dh0 = _C.gru_bwd(
            gates_x, wt, w_row, bias, is_init, h0, ys, dys
        )
        H = hprev.shape[-1]
        dW_hh = dgh.reshape(-1, 3 * H).T @ hprev.reshape(-1, H)
        db_hh = dgh.sum((0, 1))
        return dgx, dW_hh, db_hh, None, (dh0 if h0.numel() else None)


class _LSTMFusedTrainFn(torch.autograd.Function):
    """Differentiable fused LSTM scan (forward saves per-step cell
    states; backward recomputes the gates reverse-time).  Reference
    capability: _rnn_triton.py:822 (_lstm_bwd_kernel)."""

    @staticmethod
    def forward(ctx, gates_x, w_hh, is_init, h0, c0):
        wt = w_hh.detach().t().contiguous().to(torch.bfloat16)
        w_row = w_hh.detach().contiguous().to(torch.bfloat16)
        ys, cs = _C.lstm_train_fwd(gates_x, wt, is_init, h0, c0)
        ctx.save_for_backward(gates_x, wt, w_row, is_init, h0, c0, ys, cs)
        # grads flow through ys only; the cell-state sequence is exposed
        # as data (the backward kernel has no dcs input)
        ctx.mark_non_differentiable(cs)
        return ys, cs

    @staticmethod
    def backward(ctx, dys, dcs):
        gates_x, wt, w_row, is_init, h0, c0, ys, cs = ctx.saved_tensors
        dys = dys.contiguous()
        dg, hprev, dh0, dc0 = _C.lstm_bwd(
            gates_x, wt, w_row, is_init, h0, c0, ys, cs, dys
        )
        H = hprev.shape[-1]
        dW_hh = dg.reshape(-1, 4 * H).T @ hprev.reshape(-1, H)
        return (
            dg,
            dW_hh,
            None,
            (dh0 if h0.numel() else None),
            (dc0 if c0.numel() else None),
        )


def gru_train(cell, x: torch.Tensor, is_init: torch.Tensor, h0: Optional[torch.Tensor] = None):
    """Differentiable fused GRU over [B, T, F] with per-step resets.
    The x@W_ih GEMM stays a torch op (dW_ih/db_ih/dx flow through torch
    autograd); the recurrent scan runs fwd+bwd on HIP."""
    _require_ext()
    gates_x = (x.float() @ cell.weight_ih.T.float() + cell.bias_ih.float()).contiguous()
    ii = is_init.squeeze(-1) if is_init.dim() == 3 else is_init
    h0c = h0.float().contiguous() if h0 is not None else torch.Tensor()
    ys, h = _GRUFusedTrainFn.apply(
        gates_x, cell.weight_hh, cell.bias_hh, ii.contiguous(), h0c
    )
    return ys.to(x.dtype), h.to(x.dtype)


def lstm_train(cell, x: torch.Tensor, is_init: torch.Tensor, h0=None, c0=None):
    """Differentiable fused LSTM over [B, T, F] with per-step resets.
    Returns (ys, h_last, cs)."""
    _require_ext()
    gates_x = (
        x.float() @ cell.weight_ih.T.float() + cell.bias_ih.float() + cell.bias_hh.float()
    ).contiguous()
    ii = is_init.squeeze(-1) if is_init.dim() == 3 else is_init
    h0c = h0.float().contiguous() if h0 is not None else torch.Tensor()
    c0c = c0.float().contiguous() if c0 is not None else torch.Tensor()
    ys, cs = _LSTMFusedTrainFn.apply(
        gates_x, cell.weight_hh, ii.contiguous(), h0c, c0c
    )
    return ys.to(x.dtype), ys[:, -1].to(x.dtype), cs.to(x.dtype)


class FusedTanhNormalActor(torch.nn.Module):
    """Drop-in rollout policy: MLP(tanh)x2 + heads + TanhNormal sample +
    log-prob as ONE kernel (rl_amd/csrc/fused_actor.hip).

    Wraps the SAME parameters as the eager
    ``Sequential(MLP[Linear,Tanh,Linear,Tanh,Linear], NormalParamExtractor)``
    actor, so training updates flow through unchanged; only the no-grad
    rollout forward uses the fused kernel.  Falls back to the eager actor
    when grads are required or off-GPU.
    """

    def __init__(self, eager_actor, *, in_key: str = "observation", scale_lb: float = 1e-4):
        super().__init__()
        self.eager_actor = eager_actor
        self.in_key = in_key
        self.scale_lb = scale_lb
        # locate the Linear layers + extractor inside the eager actor
        import torch.nn as nn

        linears = [m for m in eager_actor.modules() if isinstance(m, nn.Linear)]
        if len(linears) != 3:
            raise ValueError("FusedTanhNormalActor expects exactly 3 Linear layers")
        self.linears = linears
        from ..modules.models.models import NormalParamExtractor

        extractors = [m for m in eager_actor.modules() if isinstance(m, NormalParamExtractor)]
        self.inv_softplus_bias = (
            extractors[0]._inv_softplus_bias if extractors else 0.5413248546129181
        )
        self.in_keys = [in_key]
        self.out_keys = ["action", "sample_log_prob", "loc", "scale"]
        self._kernel_ok = None  # probed on first GPU forward

    def forward(self, td):
        obs = td.get(self.in_key)
        if not (
            obs.is_cuda
            and not torch.is_grad_enabled()
            and HAS_HIP_EXT
            and self._kernel_ok is not False
        ):
            return self.eager_actor(td)
        w1, w2, w3 = self.linears
        A = w3.out_features // 2
        eps = torch.randn(obs.shape[0], A, device=obs.device)
        if self._kernel_ok is None:
            # dims may exceed the kernel's LDS budget (e.g. Humanoid's
            # 376-wide obs): probe once, fall back to the eager actor
            try:
                _C.fused_actor(
                    obs.contiguous().float(),
                    w1.weight.detach(), w1.bias.detach(),
                    w2.weight.detach(), w2.bias.detach(),
                    w3.weight.detach(), w3.bias.detach(),
                    eps, float(self.inv_softplus_bias), float(self.scale_lb),
                    True,
                )
                self._kernel_ok = True
            except RuntimeError:
                self._kernel_ok = False
                return self.eager_actor(td)
        action, logp, loc, scale = _C.fused_actor(
            obs.contiguous().float(),
            w1.weight.detach(), w1.bias.detach(),
            w2.weight.detach(), w2.bias.detach(),
            w3.weight.detach(), w3.bias.detach(),
            eps,
            float(self.inv_softplus_bias),
            float(self.scale_lb),
            True,
        )
        td.set("action", action)
        td.set("sample_log_prob", logp)
        td.set("loc", loc)
        td.set("scale", scale)
        return td

    def get_dist(self, td):
        return self.eager_actor.get_dist(td)

    def parameters(self, recurse: bool = True):
        return self.eager_actor.parameters(recurse)


# --------------------------------------------------------------------------- #
# Split-K weight gradient for skinny linear layers
# --------------------------------------------------------------------------- #
class _SplitKLinearFn(torch.autograd.Function):
    """F.linear with the weight gradient computed by the split-K HIP
    kernel (csrc/wgrad.hip).

    Why: measured on MI355X (rocprofv3, PPO bench r11), hipBLASLt runs
    the [64,16384]x[16384,64] wgrad of a 16k-row minibatch on ONE
    workgroup (no split-K) at ~101 us — 12.9% of the training step.  The
    split-K kernel spreads K over ~128 workgroups with fp32 atomic
    accumulation and fuses the bias gradient.
    """

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, x, weight, bias):
        x = x.contiguous()
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        return torch.nn.functional.linear(x, weight, bias)

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous()
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        # first-layer inputs are leaf observations: skip the dx GEMM
        dx = (dy @ weight).reshape(x.shape) if ctx.needs_input_grad[0] else None
        if dy2.is_cuda and dy2.dtype == torch.bfloat16 and HAS_HIP_EXT:
            outs = _C.wgrad_splitk(dy2, x2, ctx.has_bias)
            dw = outs[0].to(weight.dtype)
            db = outs[1].to(dy.dtype) if ctx.has_bias else None
        else:
            dw = (dy2.t().float() @ x2.float()).to(weight.dtype)
            db = dy2.float().sum(0).to(dy.dtype) if ctx.has_bias else None
        return dx, dw, db


class _SplitKLinearCachedFn(torch.autograd.Function):
    """Variant taking an explicitly-cached bf16 weight: under hipGraph
    capture `autocast(cache_enabled=False)` re-casts every weight on
    every call (~130 cast kernels per PPO step in the r29 profile);
    with the cache the cast runs ONCE per step, and the wgrad kernel's
    fp32 output feeds the fp32 master parameter directly (no bf16
    round-trip on the gradient)."""

    @staticmethod
    def forward(ctx, x, w_bf, weight, b_bf, bias):
        x = x.contiguous()
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        ctx.save_for_backward(x, w_bf)
        ctx.has_bias = bias is not None
        with torch.autocast("cuda", enabled=False):
            if w_bf.shape[0] == 1:
                # single-output head: hipBLASLt runs the [N,H]x[H,1]
                # gemv at ~29 us (MT1x4x256, one-ish WG); an
                # elementwise-multiply + row reduce is ~6 us
                out = (x * w_bf[0]).sum(-1, keepdim=True)
                if b_bf is not None:
                    out = out + b_bf
                return out
            return torch.nn.functional.linear(x, w_bf, b_bf)

    @staticmethod
    def backward(ctx, dy):
        x, w_bf = ctx.saved_tensors
        dy = dy.contiguous()
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        if not ctx.needs_input_grad[0]:
            dx = None
        elif w_bf.shape[0] == 1:
            # rank-1 dgrad: dy [N,1] @ w [1,H] is a broadcast multiply
            dx = (dy2 * w_bf).reshape(x.shape)
        else:
            dx = (dy @ w_bf).reshape(x.shape)
        outs = _C.wgrad_splitk(dy2, x2, ctx.has_bias)
        dw = outs[0]  # fp32 — matches the fp32 master weight
        db = outs[1] if ctx.has_bias else None
        return dx, None, dw, None, db


class SplitKLinear(torch.nn.Linear):
    """nn.Linear whose backward uses the split-K wgrad kernel on
    bf16/cuda inputs (eager semantics elsewhere).  With
    :meth:`enable_bf16_cache` the bf16 weight cast is amortized to once
    per training step (call :func:`refresh_splitk_caches` inside the
    step, after the optimizer update)."""

    _bf16_cache: bool = False

    def enable_bf16_cache(self):
        self._bf16_cache = True
        self.register_buffer("weight_bf16", self.weight.detach().to(torch.bfloat16))
        if self.bias is not None:
            self.register_buffer("bias_bf16", self.bias.detach().to(torch.bfloat16))
        return self

    def refresh_bf16_cache_(self):
        self.weight_bf16.copy_(self.weight.detach())
        if self.bias is not None:
            self.bias_bf16.copy_(self.bias.detach())

    def forward(self, x):
        if x.is_cuda and HAS_HIP_EXT:
            if self._bf16_cache and torch.is_autocast_enabled():
                return _SplitKLinearCachedFn.apply(
                    x, self.weight_bf16, self.weight,
                    getattr(self, "bias_bf16", None), self.bias,
                )
            return _SplitKLinearFn.apply(x, self.weight, self.bias)
        return super().forward(x)


def enable_splitk_bf16_cache(module: torch.nn.Module) -> None:
    for m in module.modules():
        if isinstance(m, SplitKLinear):
            m.enable_bf16_cache()
        elif hasattr(m, "enable_bf16_cache") and not isinstance(m, SplitKLinear):
            # e.g. loss ensembles with their own stacked-weight caches
            m.enable_bf16_cache()


def refresh_splitk_caches(*modules: torch.nn.Module) -> None:
    """Re-cast all cached bf16 weights — batched into a single foreach
    copy (the per-layer casts were 78 launches/step ≈ 10% of the PPO
    step in the T=64 profile); call once per training step after the
    optimizer update."""
    dsts: list = []
    srcs: list = []
    for module in modules:
        for m in module.modules():
            if isinstance(m, SplitKLinear) and m._bf16_cache:
                dsts.append(m.weight_bf16)
                srcs.append(m.weight.detach())
                if m.bias is not None:
                    dsts.append(m.bias_bf16)
                    srcs.append(m.bias.detach())
    if dsts:
        torch._foreach_copy_(dsts, srcs)
    for module in modules:
        for m in module.modules():
            if hasattr(m, "refresh_bf16_cache_") and not isinstance(m, SplitKLinear):
                m.refresh_bf16_cache_()


def convert_linears_to_splitk(module: torch.nn.Module) -> torch.nn.Module:
    """Swap every nn.Linear in ``module`` for a weight-sharing
    :class:`SplitKLinear` (in place)."""
    for name, child in module.named_children():
        if type(child) is torch.nn.Linear:
            new = SplitKLinear(
                child.in_features,
                child.out_features,
                bias=child.bias is not None,
                device=child.weight.device,
                dtype=child.weight.dtype,
            )
            new.weight = child.weight
            new.bias = child.bias
            setattr(module, name, new)
        else:
            convert_linears_to_splitk(child)
    return module


class _TanhNormalLogProbFn(torch.autograd.Function):
    """Fused TanhNormal log-prob for the PPO ratio (csrc/loss_ops.hip):
    the action is DATA (sampled during the rollout), so the gradient
    only flows to loc/scale — both analytic, one kernel each way
    (the eager chain is ~15 elementwise launches per direction)."""

    @staticmethod
    def forward(ctx, loc, scale, action):
        loc = loc.contiguous()
        scale = scale.contiguous()
        action = action.contiguous().detach()
        ctx.save_for_backward(loc, scale, action)
        return _C.tanh_normal_logprob(loc.float(), scale.float(), action.float())

    @staticmethod
    def backward(ctx, gout):
        loc, scale, action = ctx.saved_tensors
        dloc, dscale = _C.tanh_normal_logprob_bwd(
            loc.float(), scale.float(), action.float(), gout
        )
        return dloc.to(loc.dtype), dscale.to(scale.dtype), None


def tanh_normal_logprob(loc, scale, action):
    """Fused log π(a | TanhNormal(loc, scale)) with bounds (-1, 1),
    summed over the last dim.  GPU-only; used by the PPO ratio fast
    path (no gradient through the action)."""
    _require_ext()
    flat_loc = loc.reshape(-1, loc.shape[-1])
    flat_scale = scale.reshape(-1, scale.shape[-1])
    flat_act = action.reshape(-1, action.shape[-1])
    out = _TanhNormalLogProbFn.apply(flat_loc, flat_scale, flat_act)
    return out.reshape(loc.shape[:-1])


class _TanhNormalEntropyFn(torch.autograd.Function):
    """Fused reparameterized MC entropy of TanhNormal (one sample):
    analytic total gradients d(-lp)/dloc = -2x, d(-lp)/dscale =
    1/scale - 2x*eps (csrc/loss_ops.hip)."""

    @staticmethod
    def forward(ctx, loc, scale, eps):
        loc = loc.contiguous()
        scale = scale.contiguous()
        eps = eps.contiguous()
        ctx.save_for_backward(loc, scale, eps)
        return _C.tanh_normal_entropy(loc.float(), scale.float(), eps.float())

    @staticmethod
    def backward(ctx, gout):
        loc, scale, eps = ctx.saved_tensors
        dloc, dscale = _C.tanh_normal_entropy_bwd(
            loc.float(), scale.float(), eps.float(), gout
        )
        return dloc.to(loc.dtype), dscale.to(scale.dtype), None


def tanh_normal_entropy(loc, scale, eps=None):
    """Single-sample reparameterized entropy estimate of
    TanhNormal(loc, scale) with bounds (-1, 1), summed over the last
    dim.  ``eps`` defaults to a fresh standard-normal draw."""
    _require_ext()
    if eps is None:
        eps = torch.randn_like(loc)
    flat_loc = loc.reshape(-1, loc.shape[-1])
    flat_scale = scale.reshape(-1, scale.shape[-1])
    flat_eps = eps.reshape(-1, eps.shape[-1])
    out = _TanhNormalEntropyFn.apply(flat_loc, flat_scale, flat_eps)
    return out.reshape(loc.shape[:-1])


class _PPOClipFn(torch.autograd.Function):
    """Fused ClipPPO objective (csrc/loss_ops.hip; reference
    torchrl/objectives/ppo.py:1082 ClipPPOLoss.forward).  One pass
    computes loss_objective, ESS/N and clip_fraction — plus the
    optional advantage mean/std normalization, whose (mu, 1/sigma)
    stay on device and are re-applied analytically in backward.
    Replaces ~20 forward + ~10 backward elementwise/reduction
    launches per minibatch with 3-4 forward + 1 backward."""

    @staticmethod
    def forward(ctx, lw, adv, lo, hi, normalize):
        ctx.set_materialize_grads(False)
        lwf = lw.contiguous().reshape(-1)
        advf = adv.contiguous().reshape(-1)
        loss, ess, clip_frac, stats = _C.ppo_clip_fwd(lwf, advf, lo, hi, normalize)
        ctx.save_for_backward(lwf, advf, stats)
        ctx.bounds = (lo, hi)
        ctx.lw_shape = lw.shape
        ctx.mark_non_differentiable(ess, clip_frac)
        return loss, ess, clip_frac

    @staticmethod
    def backward(ctx, g_loss, g_ess, g_cf):
        lwf, advf, stats = ctx.saved_tensors
        lo, hi = ctx.bounds
        if g_loss is None:
            return None, None, None, None, None
        dlw = _C.ppo_clip_bwd(lwf, advf, stats, g_loss.contiguous(), lo, hi)
        return dlw.reshape(ctx.lw_shape), None, None, None, None


def ppo_clip_objective(log_weight, advantage, lo, hi, normalize):
    """Fused clipped-surrogate PPO objective (mean reduction) with
    on-device diagnostics.  Returns ``(loss_objective, ESS_per_sample,
    clip_fraction)`` — only the first carries gradient (to
    ``log_weight``; the advantage is treated as data).  ``normalize``
    fuses the advantage (x - mean)/std normalization into the same
    kernels."""
    _require_ext()
    return _PPOClipFn.apply(log_weight, advantage, float(lo), float(hi), bool(normalize))


class _PPOHeadLossFn(torch.autograd.Function):
    """Mega-fused TanhNormal head loss (csrc/loss_ops.hip): raw actor
    head [N, 2A] -> (loss_objective, loss_entropy, entropy_mean, ESS/N,
    clip_fraction) in one grid pass + finalize; the entire backward —
    NormalParamExtractor + log-prob + clipped surrogate + entropy — is
    ONE analytic kernel emitting d(head) in the head's dtype (bf16
    under autocast, so the MLP backward needs no casts)."""

    @staticmethod
    def forward(ctx, head, action, prev_lp, adv, eps, value, vtarget,
                sp_bias, lb, lo, hi, ent_coeff, crit_scale, normalize):
        ctx.set_materialize_grads(False)
        head = head.contiguous()
        action = action.contiguous().detach()
        prev_lp = prev_lp.contiguous().detach()
        adv = adv.contiguous().detach()
        eps = eps.contiguous()
        empty = head.new_empty(0)
        has_crit = value is not None
        v = value.contiguous().reshape(-1) if has_crit else empty
        vt = (vtarget.contiguous().reshape(-1).detach()
              if has_crit else head.new_empty(0, dtype=torch.float32))
        (loss_obj, ess, clip_frac, ent_mean, loss_ent, loss_act, loss_crit,
         loss_total, stats) = _C.ppo_head_fwd(
            head, action, eps, prev_lp, adv, v, vt, sp_bias, lb, lo, hi,
            ent_coeff, crit_scale, normalize,
        )
        ctx.save_for_backward(head, action, eps, prev_lp, adv, stats, v, vt)
        ctx.cfg = (sp_bias, lb, lo, hi, ent_coeff, crit_scale)
        ctx.has_crit = has_crit
        ctx.v_shape = value.shape if has_crit else None
        ctx.mark_non_differentiable(ess, clip_frac, ent_mean)
        return (loss_obj, loss_ent, ent_mean, ess, clip_frac, loss_act,
                loss_crit, loss_total)

    @staticmethod
    def backward(ctx, g_obj, g_ent, g_em, g_ess, g_cf, g_act, g_crit, g_tot):
        (head, action, eps, prev_lp, adv, stats, v, vt) = ctx.saved_tensors
        sp_bias, lb, lo, hi, ent_coeff, crit_scale = ctx.cfg
        empty = head.new_empty(0, dtype=torch.float32)
        gobj = g_obj.contiguous() if g_obj is not None else empty
        gent = g_ent.contiguous() if g_ent is not None else empty
        gact = g_act.contiguous() if g_act is not None else empty
        gcrit = g_crit.contiguous() if g_crit is not None else empty
        gtot = g_tot.contiguous() if g_tot is not None else empty
        dhead, dvalue = _C.ppo_head_bwd(
            head, action, eps, prev_lp, adv, stats, v, vt, gobj, gent, gact,
            gcrit, gtot, sp_bias, lb, lo, hi, ent_coeff, crit_scale,
        )
        dv = dvalue.reshape(ctx.v_shape) if ctx.has_crit else None
        return (dhead, None, None, None, None, dv, None, None, None, None,
                None, None, None, None)


def fused_grad_clip_(parameters, max_norm: float) -> bool:
    """clip_grad_norm_ in two launches (csrc/loss_ops.hip): one
    single-WG kernel for the global norm + clamped coefficient, one
    foreach multiply.  Returns False (caller should use torch's
    clip_grad_norm_) when the gradients don't fit the fused kernel."""
    if not HAS_HIP_EXT:
        return False
    grads = [p.grad for p in parameters if p.grad is not None]
    if not grads or len(grads) > 32:
        return False
    for g in grads:
        if not (g.is_cuda and g.dtype == torch.float32 and g.is_contiguous()):
            return False
    coef = _C.fused_grad_clip_coef(
        grads, float(max_norm), False, torch.empty(0)
    )
    torch._foreach_mul_(grads, coef)
    return True


def fused_grad_clip_scale_(parameters, max_norm: float,
                           out_scale: torch.Tensor) -> bool:
    """Write the INVERSE clip coefficient ``max(1, norm/max_norm)``
    into ``out_scale`` (a persistent 0-d fp32 cuda tensor) — exactly
    the ``grad_scale`` divisor torch's fused Adam consumes, so the
    clip costs two launches and NO gradient-multiply at all."""
    if not HAS_HIP_EXT:
        return False
    grads = [p.grad for p in parameters if p.grad is not None]
    if not grads or len(grads) > 32:
        return False
    for g in grads:
        if not (g.is_cuda and g.dtype == torch.float32 and g.is_contiguous()):
            return False
    _C.fused_grad_clip_coef(grads, float(max_norm), True, out_scale)
    return True


def multi_gather_td(flat_td, perm: torch.Tensor):
    """Shuffle-gather a flat TensorDict of fp32 cuda leaves in ONE
    kernel (csrc/loss_ops.hip) instead of one index kernel per key.
    Returns None when ineligible (caller falls back to flat_td[perm])."""
    if not HAS_HIP_EXT:
        return None
    try:
        items = list(flat_td.items())
    except Exception:
        return None
    if not (1 <= len(items) <= 8):
        return None
    n = perm.numel()
    srcs = []
    for _, v in items:
        if not (
            isinstance(v, torch.Tensor)
            and v.is_cuda
            and v.dtype == torch.float32
            and v.is_contiguous()
            and v.dim() >= 1
            and v.shape[0] == n
        ):
            return None
        srcs.append(v)
    outs = _C.multi_gather(perm.contiguous(), srcs)
    from ..tensordict import TensorDict

    return TensorDict(
        {k: o for (k, _), o in zip(items, outs)}, batch_size=[n],
        device=perm.device,
    )


def multi_shuffle_td(flat_td, keys: torch.Tensor):
    """Epoch shuffle via a keyed Feistel permutation computed inline in
    ONE kernel (csrc/loss_ops.hip), replacing randperm radix sort plus
    per-key gathers.  keys is an int32[4] cuda tensor of round keys
    (draw fresh per epoch so graph replays get a new permutation).
    Returns None when the leaves do not fit."""
    dbg = os.environ.get("RL_AMD_DEBUG_SHUFFLE") == "1"
    if not HAS_HIP_EXT:
        if dbg: print("shuffle: no ext", file=__import__("sys").stderr)
        return None
    try:
        items = list(flat_td.items())
    except Exception as e:
        if dbg: print("shuffle: items()", e, file=__import__("sys").stderr)
        return None
    if not (1 <= len(items) <= 8):
        if dbg: print("shuffle: n items", len(items), file=__import__("sys").stderr)
        return None
    n = flat_td.batch_size[0]
    srcs = []
    for k, v in items:
        if not (
            isinstance(v, torch.Tensor)
            and v.is_cuda
            and v.dtype == torch.float32
            and v.is_contiguous()
            and v.dim() >= 1
            and v.shape[0] == n
        ):
            if dbg: print("shuffle: leaf", k, type(v), getattr(v, "dtype", None), getattr(v, "shape", None), v.is_contiguous() if isinstance(v, torch.Tensor) else None, file=__import__("sys").stderr)
            return None
        srcs.append(v)
    outs = _C.multi_shuffle(keys, srcs)
    from ..tensordict import TensorDict

    return TensorDict(
        {k: o for (k, _), o in zip(items, outs)}, batch_size=[n],
        device=srcs[0].device,
    )


def ppo_head_loss(head, action, prev_log_prob, advantage, eps, *, sp_bias,
                  scale_lb, lo, hi, entropy_coeff, normalize, value=None,
                  value_target=None, critic_scale=1.0):
    """Fused ClipPPO losses straight from the raw policy-head output
    (``[N, 2A]`` = loc | pre-softplus scale).  Returns
    ``(loss_objective, loss_entropy, entropy_mean, ESS_per_sample,
    clip_fraction, loss_actor, loss_critic, loss_total)``.  When
    ``value``/``value_target`` are given, the scaled smooth-L1 critic
    loss and the whole-minibatch total ride in the SAME kernels (the
    backward emits d(head) and d(value) in one pass); otherwise the
    last two outputs are undefined scalars."""
    _require_ext()
    return _PPOHeadLossFn.apply(
        head, action, prev_log_prob, advantage, eps, value, value_target,
        float(sp_bias), float(scale_lb), float(lo), float(hi),
        float(entropy_coeff), float(critic_scale), bool(normalize),
    )


class _SmoothL1MeanFn(torch.autograd.Function):
    """Fused mean smooth-L1 (beta=1) critic loss (csrc/loss_ops.hip):
    one partials + one finalize launch forward, one analytic backward
    ``dv = g/N * clamp(v - t, -1, 1)``.  The value may be bf16 (under
    autocast); the target is fp32 data."""

    @staticmethod
    def forward(ctx, value, target, scale):
        ctx.set_materialize_grads(False)
        vf = value.contiguous().reshape(-1)
        tf = target.contiguous().reshape(-1)
        ctx.save_for_backward(vf, tf)
        ctx.v_shape = value.shape
        ctx.scale = scale
        return _C.smooth_l1_fwd(vf, tf, scale)

    @staticmethod
    def backward(ctx, gout):
        vf, tf = ctx.saved_tensors
        if gout is None:
            return None, None, None
        dv = _C.smooth_l1_bwd(vf, tf, gout.contiguous(), ctx.scale)
        return dv.reshape(ctx.v_shape), None, None


def smooth_l1_mean(value, target, scale: float = 1.0):
    """Fused ``scale * F.smooth_l1_loss(value, target, reduction=
    "mean")`` on GPU; gradient flows to ``value`` only (the target is
    data).  ``scale`` folds a loss coefficient into the kernel."""
    _require_ext()
    return _SmoothL1MeanFn.apply(value, target.detach(), float(scale))


class _FusedMLP3Fn(torch.autograd.Function):
    """Whole 3-layer tanh MLP forward + backward on HIP
    (csrc/fused_mlp.hip + the MFMA split-K wgrad): one forward launch
    (saving h1/h2), one dgrad-chain launch, three wgrad launches —
    replacing ~45 eager launches per network per minibatch.  The input
    is the rollout observation (a leaf), so dX is skipped."""

    @staticmethod
    def forward(ctx, x, w1b, b1b, w2b, b2b, w3b, b3b, w1, b1, w2, b2, w3, b3):
        x = x.contiguous()
        O, H, A2 = x.shape[-1], w1b.shape[0], w3b.shape[0]
        if _C.mlp3_mfma_ok(O, H, A2):
            # MFMA path: fp32 input converted during staging (the bf16
            # copy for the wgrad comes back as xb)
            out, h1, h2, xb = _C.mlp3_mfma_fwd(x, w1b, b1b, w2b, b2b, w3b, b3b)
            ctx.mfma = True
            ctx.save_for_backward(xb, h1, h2, w2b, w3b)
            return out
        ctx.mfma = False
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        out, h1, h2 = _C.mlp3_fwd(x, w1b, b1b, w2b, b2b, w3b, b3b)
        ctx.save_for_backward(x, h1, h2, w2b, w3b)
        return out

    @staticmethod
    def backward(ctx, dout):
        x, h1, h2, w2b, w3b = ctx.saved_tensors
        dout = dout.contiguous()
        if ctx.mfma:
            dh1, dh2 = _C.mlp3_mfma_bwd(dout, h1, h2, w2b, w3b)
            if dout.dtype != torch.bfloat16:
                dout = dout.to(torch.bfloat16)
        else:
            if dout.dtype != torch.bfloat16:
                dout = dout.to(torch.bfloat16)
            dh1, dh2 = _C.mlp3_bwd(dout, h1, h2, w2b, w3b)
        if max(dout.shape[1], h2.shape[1], x.shape[1]) <= 64:
            # all three layer wgrads in one mfma + one reduce launch
            dw3, db3, dw2, db2, dw1, db1 = _C.wgrad_splitk3(
                dout, h2, dh2, h1, dh1, x
            )
        else:
            dw3, db3 = _C.wgrad_splitk(dout, h2, True)
            dw2, db2 = _C.wgrad_splitk(dh2, h1, True)
            dw1, db1 = _C.wgrad_splitk(dh1, x, True)
        return (None, None, None, None, None, None, None,
                dw1, db1, dw2, db2, dw3, db3)


class FusedMLP3(torch.nn.Module):
    """Routes a Linear-Tanh-Linear-Tanh-Linear stack through the fused
    kernels on GPU (bf16 weight caches shared with SplitKLinear);
    eager fallback elsewhere.  Built from three weight-sharing
    :class:`SplitKLinear` layers with ``enable_bf16_cache`` on.

    .. note:: Two kernel generations: the original VALU dot kernels
       (small/odd shapes) and the MFMA v2 kernels (H % 16 == 0) that
       run the three GEMMs on the matrix cores — the v2 path beats the
       eager hipBLASLt chain at the PPO bench's 16k-row minibatches
       because it folds the tanh/bias/cast launches into the GEMM
       launch (the earlier VALU version lost there, measured r37:
       T=64 7.49 ms vs 5.18)."""

    def __init__(self, lin1, lin2, lin3, eager: torch.nn.Module):
        super().__init__()
        self.lin1, self.lin2, self.lin3 = lin1, lin2, lin3
        self.eager = eager

    def forward(self, x):
        if (
            x.is_cuda
            and HAS_HIP_EXT
            and getattr(self.lin1, "_bf16_cache", False)
            and x.shape[-1] == self.lin1.in_features
        ):
            lead = x.shape[:-1]
            flat = x.reshape(-1, x.shape[-1])
            out = _FusedMLP3Fn.apply(
                flat,
                self.lin1.weight_bf16, self.lin1.bias_bf16,
                self.lin2.weight_bf16, self.lin2.bias_bf16,
                self.lin3.weight_bf16, self.lin3.bias_bf16,
                self.lin1.weight, self.lin1.bias,
                self.lin2.weight, self.lin2.bias,
                self.lin3.weight, self.lin3.bias,
            )
            return out.reshape(*lead, out.shape[-1])
        return self.eager(x)


class _FusedACFn(torch.autograd.Function):
    """Actor + critic whole-MLP forward/backward fused across networks
    (csrc/fused_mlp.hip fwd2/bwd2 + the 6-layer batched wgrad): both
    nets read the SAME input rows, so one launch each way covers the
    entire minibatch model compute of a PPO update."""

    @staticmethod
    def forward(ctx, x, aw1b, ab1b, aw2b, ab2b, aw3b, ab3b,
                cw1b, cb1b, cw2b, cb2b, cw3b, cb3b,
                aw1, ab1, aw2, ab2, aw3, ab3,
                cw1, cb1, cw2, cb2, cw3, cb3):
        x = x.contiguous()
        head, a_h1, a_h2, value, c_h1, c_h2, xb = _C.mlp3_mfma_fwd2(
            x,
            [aw1b, ab1b, aw2b, ab2b, aw3b, ab3b],
            [cw1b, cb1b, cw2b, cb2b, cw3b, cb3b],
        )
        ctx.save_for_backward(xb, a_h1, a_h2, aw2b, aw3b, c_h1, c_h2, cw2b,
                              cw3b)
        return head, value

    @staticmethod
    def backward(ctx, dhead, dvalue):
        (xb, a_h1, a_h2, aw2b, aw3b, c_h1, c_h2, cw2b,
         cw3b) = ctx.saved_tensors
        dhead = dhead.contiguous()
        dvalue = dvalue.contiguous()
        if dhead.dtype != torch.bfloat16:
            dhead = dhead.to(torch.bfloat16)
        if dvalue.dtype != torch.bfloat16:
            dvalue = dvalue.to(torch.bfloat16)
        a_dh1, a_dh2, c_dh1, c_dh2 = _C.mlp3_mfma_bwd2(
            dhead, a_h1, a_h2, aw2b, aw3b, dvalue, c_h1, c_h2, cw2b, cw3b
        )
        (adw3, adb3, adw2, adb2, adw1, adb1, cdw3, cdb3, cdw2, cdb2, cdw1,
         cdb1) = _C.wgrad_splitk_batch(
            [dhead, a_dh2, a_dh1, dvalue, c_dh2, c_dh1],
            [a_h2, a_h1, xb, c_h2, c_h1, xb],
        )
        return (None,) * 13 + (adw1, adb1, adw2, adb2, adw3, adb3,
                               cdw1, cdb1, cdw2, cdb2, cdw3, cdb3)


def actor_critic_mlp3(x, actor_fused, critic_fused):
    """Run two :class:`FusedMLP3` networks (same input) as ONE launch
    each way.  Returns ``(actor_head, critic_out)``."""
    _require_ext()
    a, c = actor_fused, critic_fused
    return _FusedACFn.apply(
        x,
        a.lin1.weight_bf16, a.lin1.bias_bf16, a.lin2.weight_bf16,
        a.lin2.bias_bf16, a.lin3.weight_bf16, a.lin3.bias_bf16,
        c.lin1.weight_bf16, c.lin1.bias_bf16, c.lin2.weight_bf16,
        c.lin2.bias_bf16, c.lin3.weight_bf16, c.lin3.bias_bf16,
        a.lin1.weight, a.lin1.bias, a.lin2.weight, a.lin2.bias,
        a.lin3.weight, a.lin3.bias,
        c.lin1.weight, c.lin1.bias, c.lin2.weight, c.lin2.bias,
        c.lin3.weight, c.lin3.bias,
    )


def value_pair_eval(fused, x0, x1):
    """No-grad evaluation of one :class:`FusedMLP3` on TWO equal-shaped
    inputs in a single launch (csrc/fused_mlp.hip fwdpair) — GAE's
    value/next_value critic calls.  Returns ``(y0, y1)``."""
    _require_ext()
    w = [fused.lin1.weight_bf16, fused.lin1.bias_bf16,
         fused.lin2.weight_bf16, fused.lin2.bias_bf16,
         fused.lin3.weight_bf16, fused.lin3.bias_bf16]
    return _C.mlp3_mfma_fwdpair(x0.contiguous(), x1.contiguous(), w)


class _ACLossFn(torch.autograd.Function):
    """THE whole-minibatch Function: actor+critic MFMA MLPs and every
    ClipPPO loss scalar in one forward launch (+1-WG finalize), and one
    backward launch computing d(head)/d(value) inline before the dgrad
    chains, followed by the 6-layer batched wgrad
    (csrc/fused_mlp.hip acloss kernels)."""

    @staticmethod
    def forward(ctx, x, aw1b, ab1b, aw2b, ab2b, aw3b, ab3b,
                cw1b, cb1b, cw2b, cb2b, cw3b, cb3b,
                aw1, ab1, aw2, ab2, aw3, ab3,
                cw1, cb1, cw2, cb2, cw3, cb3,
                action, prev_lp, adv, vtarget, eps, stats_in, gradsq_out,
                sp_bias, lb, lo, hi, ent_coeff, crit_scale, normalize):
        ctx.set_materialize_grads(False)
        x = x.contiguous()
        action = action.contiguous().detach()
        prev_lp = prev_lp.contiguous().detach()
        adv = adv.contiguous().detach()
        vtarget = vtarget.contiguous().detach()
        eps = eps.contiguous()
        if stats_in is None:
            stats_in = x.new_empty(0, dtype=torch.float32)
        (head, a_h1, a_h2, value, c_h1, c_h2, xb,
         loss_obj, ess, cf, ent, lent, lact, lcrit, ltotal,
         stats) = _C.acloss_fwd(
            x, [aw1b, ab1b, aw2b, ab2b, aw3b, ab3b],
            [cw1b, cb1b, cw2b, cb2b, cw3b, cb3b],
            action, eps, prev_lp, adv, vtarget, stats_in.contiguous(),
            sp_bias, lb, lo, hi, ent_coeff, crit_scale, normalize,
        )
        ctx.save_for_backward(head, value, a_h1, a_h2, aw2b, aw3b, c_h1,
                              c_h2, cw2b, cw3b, action, eps, prev_lp, adv,
                              stats, vtarget, xb)
        ctx.gradsq_out = gradsq_out
        ctx.cfg = (sp_bias, lb, lo, hi, ent_coeff, crit_scale)
        ctx.mark_non_differentiable(ess, cf, ent)
        return loss_obj, lent, ent, ess, cf, lact, lcrit, ltotal

    @staticmethod
    def backward(ctx, g_obj, g_ent, g_em, g_ess, g_cf, g_act, g_crit, g_tot):
        (head, value, a_h1, a_h2, aw2b, aw3b, c_h1, c_h2, cw2b, cw3b,
         action, eps, prev_lp, adv, stats, vtarget, xb) = ctx.saved_tensors
        sp_bias, lb, lo, hi, ent_coeff, crit_scale = ctx.cfg
        empty = head.new_empty(0, dtype=torch.float32)
        g = lambda t: t.contiguous() if t is not None else empty
        dhead, dvalue, a_dh1, a_dh2, c_dh1, c_dh2 = _C.acloss_bwd(
            head, value, a_h1, a_h2, aw2b, aw3b, c_h1, c_h2, cw2b, cw3b,
            action, eps, prev_lp, adv, stats, vtarget,
            g(g_obj), g(g_ent), g(g_act), g(g_crit), g(g_tot),
            sp_bias, lb, lo, hi, ent_coeff, crit_scale,
        )
        dys = [dhead, a_dh2, a_dh1, dvalue, c_dh2, c_dh1]
        xs = [a_h2, a_h1, xb, c_h2, c_h1, xb]
        if ctx.gradsq_out is not None:
            # grad-sumsq partials ride in the reduce: the trainer turns
            # them into the clip coefficient with one tiny kernel
            (adw3, adb3, adw2, adb2, adw1, adb1, cdw3, cdb3, cdw2, cdb2,
             cdw1, cdb1) = _C.wgrad_splitk_batch_sq(dys, xs, ctx.gradsq_out)
        else:
            (adw3, adb3, adw2, adb2, adw1, adb1, cdw3, cdb3, cdw2, cdb2,
             cdw1, cdb1) = _C.wgrad_splitk_batch(dys, xs)
        return ((None,) * 13
                + (adw1, adb1, adw2, adb2, adw3, adb3,
                   cdw1, cdb1, cdw2, cdb2, cdw3, cdb3)
                + (None,) * 14)


def actor_critic_loss(x, actor_fused, critic_fused, action, prev_log_prob,
                      advantage, value_target, eps, *, sp_bias, scale_lb,
                      lo, hi, entropy_coeff, critic_scale, normalize,
                      stats_in=None, gradsq_out=None):
    """Run the fully-merged actor+critic+loss Function.  Returns
    ``(loss_objective, loss_entropy, entropy, ESS, clip_fraction,
    loss_actor, loss_critic, loss_total)``."""
    _require_ext()
    a, c = actor_fused, critic_fused
    return _ACLossFn.apply(
        x,
        a.lin1.weight_bf16, a.lin1.bias_bf16, a.lin2.weight_bf16,
        a.lin2.bias_bf16, a.lin3.weight_bf16, a.lin3.bias_bf16,
        c.lin1.weight_bf16, c.lin1.bias_bf16, c.lin2.weight_bf16,
        c.lin2.bias_bf16, c.lin3.weight_bf16, c.lin3.bias_bf16,
        a.lin1.weight, a.lin1.bias, a.lin2.weight, a.lin2.bias,
        a.lin3.weight, a.lin3.bias,
        c.lin1.weight, c.lin1.bias, c.lin2.weight, c.lin2.bias,
        c.lin3.weight, c.lin3.bias,
        action, prev_log_prob, advantage, value_target, eps, stats_in,
        gradsq_out,
        float(sp_bias), float(scale_lb), float(lo), float(hi),
        float(entropy_coeff), float(critic_scale), bool(normalize),
    )


def adv_stats_batch(adv_flat: torch.Tensor, n_mb: int) -> torch.Tensor:
    """Per-minibatch advantage (mean, 1/std) for ``n_mb`` contiguous
    slices of ``adv_flat`` in ONE launch pair.  Returns [n_mb, 2]."""
    _require_ext()
    return _C.adv_stats_batch(adv_flat.contiguous(), int(n_mb))


def actor_critic_mlp3_ok(actor_fused, critic_fused, in_features: int) -> bool:
    """Shape eligibility for the dual-network kernels."""
    if not HAS_HIP_EXT:
        return False
    try:
        return bool(
            _C.mlp3_mfma_ok(in_features, actor_fused.lin1.out_features,
                            actor_fused.lin3.out_features)
            and _C.mlp3_mfma_ok(in_features, critic_fused.lin1.out_features,
                                critic_fused.lin3.out_features)
            and actor_fused.lin1.in_features == in_features
            and critic_fused.lin1.in_features == in_features
        )
    except Exception:
        return False


def fuse_mlp3(module: torch.nn.Module) -> torch.nn.Module:
    """Wrap a Sequential [SplitKLinear, Tanh, SplitKLinear, Tanh,
    SplitKLinear] (e.g. rl_amd MLP internals after
    ``convert_linears_to_splitk`` + ``enable_splitk_bf16_cache``) in a
    :class:`FusedMLP3`.  Returns the module unchanged if the pattern
    does not match."""
    import torch.nn as nn

    linears = [m for m in module.modules() if isinstance(m, SplitKLinear)]
    tanhs = [m for m in module.modules() if isinstance(m, nn.Tanh)]
    if len(linears) == 3 and len(tanhs) == 2 and all(
        getattr(l, "_bf16_cache", False) for l in linears
    ):
        return FusedMLP3(*linears, eager=module)
    return module
