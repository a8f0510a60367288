"""Recurrent policies: LSTM/GRU with per-step reset handling.

Reference: pytorch/rl torchrl/modules/tensordict_module/rnn.py (LSTM:363,
LSTMModule:650, GRU:1818, GRUModule:2090, set_recurrent_mode:3004) and the
Triton fused-scan backends (_rnn_triton.py, 2,214 LoC) — re-designed for
MI355X:

* ``backend="scan"`` — a python-level time loop with reset masking (the
  numerics oracle; differentiable; runs everywhere).
* ``backend="fused"`` — the CDNA4 HIP kernel (rl_amd/csrc/rnn_scan.hip):
  the x@W_ih GEMM for ALL timesteps runs as ONE hipBLASLt GEMM outside,
  then a single kernel scans time with W_hh staged in LDS and h kept in
  registers (MFMA tiles for the h@W_hh product), handling `is_init`
  resets per step.  Forward-only (collector rollouts are no_grad);
  training falls back to scan.

Single-step mode (inside a collector) uses the plain cell update —
recurrent_mode switches between step and sequence processing
(reference set_recurrent_mode:3004).
"""
from __future__ import annotations

import contextlib
import threading
from typing import Optional, Sequence, Tuple

import torch
from torch import nn

from ...tensordict import TensorDict, TensorDictBase, TensorDictModuleBase, unravel_key

__all__ = [
    "LSTMModule",
    "GRUModule",
    "LSTMCell",
    "GRUCell",
    "set_recurrent_mode",
    "recurrent_mode",
    "lstm_scan",
    "gru_scan",
]

_RECURRENT = threading.local()


def recurrent_mode() -> bool:
    return getattr(_RECURRENT, "on", False)


@contextlib.contextmanager
def set_recurrent_mode(mode: bool = True):
    prev = getattr(_RECURRENT, "on", False)
    _RECURRENT.on = mode
    try:
        yield
    finally:
        _RECURRENT.on = prev


class LSTMCell(nn.Module):
    """Plain python LSTM cell (reference rnn.py:250) — used for the
    single-step path and as the scan body."""

    def __init__(self, input_size: int, hidden_size: int, device=None, dtype=None):
        super().__init__()
        self.input_size = input_size
        self.hidden_size = hidden_size
        factory = {"device": device, "dtype": dtype}
        self.weight_ih = nn.Parameter(torch.empty(4 * hidden_size, input_size, **factory))
        self.weight_hh = nn.Parameter(torch.empty(4 * hidden_size, hidden_size, **factory))
        self.bias_ih = nn.Parameter(torch.empty(4 * hidden_size, **factory))
        self.bias_hh = nn.Parameter(torch.empty(4 * hidden_size, **factory))
        self.reset_parameters()

    def reset_parameters(self):
        std = 1.0 / (self.hidden_size**0.5)
        for p in self.parameters():
            nn.init.uniform_(p, -std, std)

    def forward(self, x, hc: Optional[Tuple[torch.Tensor, torch.Tensor]] = None):
        if hc is None:
            z = torch.zeros(
                *x.shape[:-1], self.hidden_size, device=x.device, dtype=x.dtype
            )
            hc = (z, z.clone())
        h, c = hc
        gates = x @ self.weight_ih.T + self.bias_ih + h @ self.weight_hh.T + self.bias_hh
        i, f, g, o = gates.chunk(4, -1)
        i, f, o = i.sigmoid(), f.sigmoid(), o.sigmoid()
        g = g.tanh()
        c = f * c + i * g
        h = o * c.tanh()
        return h, c


class GRUCell(nn.Module):
    """Plain python GRU cell (reference rnn.py:1713)."""

    def __init__(self, input_size: int, hidden_size: int, device=None, dtype=None):
        super().__init__()
        self.input_size = input_size
        self.hidden_size = hidden_size
        factory = {"device": device, "dtype": dtype}
        self.weight_ih = nn.Parameter(torch.empty(3 * hidden_size, input_size, **factory))
        self.weight_hh = nn.Parameter(torch.empty(3 * hidden_size, hidden_size, **factory))
        self.bias_ih = nn.Parameter(torch.empty(3 * hidden_size, **factory))
        self.bias_hh = nn.Parameter(torch.empty(3 * hidden_size, **factory))
        self.reset_parameters()

    def reset_parameters(self):
        std = 1.0 / (self.hidden_size**0.5)
        for p in self.parameters():
            nn.init.uniform_(p, -std, std)

    def forward(self, x, h: Optional[torch.Tensor] = None):
        if h is None:
            h = torch.zeros(
                *x.shape[:-1], self.hidden_size, device=x.device, dtype=x.dtype
            )
        gx = x @ self.weight_ih.T + self.bias_ih
        gh = h @ self.weight_hh.T + self.bias_hh
        rx, zx, nx = gx.chunk(3, -1)
        rh, zh, nh = gh.chunk(3, -1)
        r = (rx + rh).sigmoid()
        z = (zx + zh).sigmoid()
        n = (nx + r * nh).tanh()
        return (1 - z) * n + z * h


def lstm_scan(
    cell: LSTMCell,
    x: torch.Tensor,
    is_init: torch.Tensor,
    h0: Optional[torch.Tensor] = None,
    c0: Optional[torch.Tensor] = None,
):
    """Sequence scan with per-step reset: where ``is_init[:, t]`` the state
    is zeroed before step t (reference fused-scan semantics,
    _rnn_triton.py:191).  x: [B, T, F]; is_init: [B, T] or [B, T, 1]."""
    B, T = x.shape[0], x.shape[1]
    H = cell.hidden_size
    if h0 is None:
        h0 = torch.zeros(B, H, device=x.device, dtype=x.dtype)
    if c0 is None:
        c0 = torch.zeros(B, H, device=x.device, dtype=x.dtype)
    if is_init.dim() == 3:
        is_init = is_init.squeeze(-1)
    # precompute the input GEMM for all timesteps at once (one big GEMM)
    gx_all = x @ cell.weight_ih.T + cell.bias_ih
    h, c = h0, c0
    hs = []
    cs = []
    for t in range(T):
        mask = is_init[:, t].unsqueeze(-1).to(x.dtype)
        h = h * (1 - mask)
        c = c * (1 - mask)
        gates = gx_all[:, t] + h @ cell.weight_hh.T + cell.bias_hh
        i, f, g, o = gates.chunk(4, -1)
        i, f, o = i.sigmoid(), f.sigmoid(), o.sigmoid()
        g = g.tanh()
        c = f * c + i * g
        h = o * c.tanh()
        hs.append(h)
        cs.append(c)
    return torch.stack(hs, 1), h, torch.stack(cs, 1)


def gru_scan(
    cell: GRUCell,
    x: torch.Tensor,
    is_init: torch.Tensor,
    h0: Optional[torch.Tensor] = None,
):
    B, T = x.shape[0], x.shape[1]
    H = cell.hidden_size
    if h0 is None:
        h0 = torch.zeros(B, H, device=x.device, dtype=x.dtype)
    if is_init.dim() == 3:
        is_init = is_init.squeeze(-1)
    gx_all = x @ cell.weight_ih.T + cell.bias_ih
    h = h0
    hs = []
    for t in range(T):
        mask = is_init[:, t].unsqueeze(-1).to(x.dtype)
        h = h * (1 - mask)
        gh = h @ cell.weight_hh.T + cell.bias_hh
        rx, zx, nx = gx_all[:, t].chunk(3, -1)
        rh, zh, nh = gh.chunk(3, -1)
        r = (rx + rh).sigmoid()
        z = (zx + zh).sigmoid()
        n = (nx + r * nh).tanh()
        h = (1 - z) * n + z * h
        hs.append(h)
    return torch.stack(hs, 1), h


def _fused_available(x: torch.Tensor, hidden: int) -> bool:
    if not x.is_cuda or torch.is_grad_enabled():
        return False
    try:
        from ... import ops

        return ops.HAS_HIP_EXT and hasattr(ops.ext_module(), "gru_fused")
    except Exception:
        return False


def _fused_train_available(x: torch.Tensor, hidden: int, kind: str) -> bool:
    """Differentiable fused scan (HIP fwd + reverse-time recompute bwd)."""
    if not x.is_cuda:
        return False
    try:
        from ... import ops

        # any H is supported: W_hh^T is LDS-resident when it fits the
        # 160 KB budget, L2-streamed otherwise (csrc/rnn_scan.hip)
        return ops.HAS_HIP_EXT and hasattr(ops.ext_module(), f"{kind}_bwd")
    except Exception:
        return False


class _RNNModuleBase(TensorDictModuleBase):
    cell_cls = None

    def __init__(
        self,
        input_size: int,
        hidden_size: int,
        *,
        in_key: str = "observation",
        out_key: str = "embed",
        device=None,
        backend: str = "scan",
        python_based: bool = True,
    ):
        super().__init__()
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.cell = self.cell_cls(input_size, hidden_size, device=device)
        self.in_key = unravel_key(in_key)
        self.out_key = unravel_key(out_key)
        self.backend = backend

    def make_tensordict_primer(self):
        """Primer that seeds recurrent state keys at reset
        (reference LSTMModule.make_tensordict_primer)."""
        from ...data.tensor_specs import Unbounded
        from ...envs.transforms import TensorDictPrimer

        specs = {
            k: Unbounded(shape=(self.hidden_size,))
            for k in self.state_keys
        }
        return TensorDictPrimer(specs)


class LSTMModule(_RNNModuleBase):
    """TensorDict LSTM with reset handling (reference rnn.py:650).

    Step mode (default, inside collectors): reads recurrent_state_{h,c},
    applies one cell step (zeroing state where ``is_init``), writes the
    new state under ``("next", ...)`` so the collector carries it.
    Sequence mode (``set_recurrent_mode(True)``, inside losses): scans the
    whole [B, T] batch using per-step ``is_init`` resets.
    """

    cell_cls = LSTMCell

    def __init__(self, input_size: int, hidden_size: int, *, in_key="observation", out_key="embed", device=None, backend: str = "scan", **kwargs):
        super().__init__(
            input_size, hidden_size, in_key=in_key, out_key=out_key, device=device, backend=backend
        )
        self.state_keys = ["recurrent_state_h", "recurrent_state_c"]
        self.in_keys = [self.in_key, "is_init", "recurrent_state_h", "recurrent_state_c"]
        self.out_keys = [
            self.out_key,
            ("next", "recurrent_state_h"),
            ("next", "recurrent_state_c"),
        ]

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        x = td.get(self.in_key)
        is_init = td.get("is_init", None)
        if recurrent_mode() and td.batch_dims >= 2:
            B = x.shape[0]
            if is_init is None:
                is_init = torch.zeros(*x.shape[:2], dtype=torch.bool, device=x.device)
            h0 = td.get("recurrent_state_h", None)
            c0 = td.get("recurrent_state_c", None)
            if h0 is not None and h0.dim() == 3:
                h0 = h0[:, 0]
                c0 = c0[:, 0]
            if self.backend == "fused" and _fused_train_available(
                x, self.hidden_size, "lstm"
            ):
                # one HIP launch fwd (+ the reverse-time recompute scan
                # in backward when grads are on)
                from ... import ops

                ys, h, cs = ops.lstm_train(self.cell, x, is_init, h0, c0)
            else:
                ys, h, cs = lstm_scan(self.cell, x, is_init, h0, c0)
            td.set(self.out_key, ys)
            td.set(("next", "recurrent_state_h"), ys)
            td.set(("next", "recurrent_state_c"), cs)
            return td
        # single-step
        h = td.get("recurrent_state_h", None)
        c = td.get("recurrent_state_c", None)
        if h is None:
            h = torch.zeros(*x.shape[:-1], self.hidden_size, device=x.device, dtype=x.dtype)
            c = torch.zeros_like(h)
        if is_init is not None:
            mask = is_init.to(x.dtype)
            if mask.shape != h.shape:
                mask = mask.reshape(*h.shape[:-1], 1).expand_as(h)
            h = h * (1 - mask)
            c = c * (1 - mask)
        h, c = self.cell(x, (h, c))
        td.set(self.out_key, h)
        td.set(("next", "recurrent_state_h"), h)
        td.set(("next", "recurrent_state_c"), c)
        td.set("recurrent_state_h", td.get("recurrent_state_h", h))
        td.set("recurrent_state_c", td.get("recurrent_state_c", c))
        return td


class GRUModule(_RNNModuleBase):
    """TensorDict GRU with reset handling (reference rnn.py:2090)."""

    cell_cls = GRUCell

    def __init__(self, input_size: int, hidden_size: int, *, in_key="observation", out_key="embed", device=None, backend: str = "scan", **kwargs):
        super().__init__(
            input_size, hidden_size, in_key=in_key, out_key=out_key, device=device, backend=backend
        )
        self.state_keys = ["recurrent_state"]
        self.in_keys = [self.in_key, "is_init", "recurrent_state"]
        self.out_keys = [self.out_key, ("next", "recurrent_state")]

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        x = td.get(self.in_key)
        is_init = td.get("is_init", None)
        if recurrent_mode() and td.batch_dims >= 2:
            if is_init is None:
                is_init = torch.zeros(*x.shape[:2], dtype=torch.bool, device=x.device)
            h0 = td.get("recurrent_state", None)
            if h0 is not None and h0.dim() == 3:
                h0 = h0[:, 0]
            if self.backend == "fused" and torch.is_grad_enabled() and _fused_train_available(
                x, self.hidden_size, "gru"
            ):
                from ... import ops

                ys, h = ops.gru_train(self.cell, x, is_init, h0)
            elif self.backend == "fused" and _fused_available(x, self.hidden_size):
                from ... import ops

                ys, h = ops.gru_fused(self.cell, x, is_init, h0)
            else:
                ys, h = gru_scan(self.cell, x, is_init, h0)
            td.set(self.out_key, ys)
            td.set(("next", "recurrent_state"), ys)
            return td
        h = td.get("recurrent_state", None)
        if h is None:
            h = torch.zeros(*x.shape[:-1], self.hidden_size, device=x.device, dtype=x.dtype)
        if is_init is not None:
            mask = is_init.to(x.dtype)
            if mask.shape != h.shape:
                mask = mask.reshape(*h.shape[:-1], 1).expand_as(h)
            h = h * (1 - mask)
        h = self.cell(x, h)
        td.set(self.out_key, h)
        td.set(("next", "recurrent_state"), h)
        td.set("recurrent_state", td.get("recurrent_state", h))
        return td
