"""RNN training-path benchmark: fused HIP scan (fwd + reverse-time
recompute bwd) vs the python autograd scan.

Reference analog: pytorch/rl benchmarks/bench_rnn_backward.py and
test_rnn_reset_backends_benchmark.py (pad vs scan vs triton).  The
reference's Triton comment (_rnn_triton.py:160) quotes the chunked path
"~3-5x slower than fused" at H=256, B=8000 on H200 — this measures the
same comparison for the MI355X HIP scans.

Usage: python benchmarks/bench_rnn_train.py [--reps 20]
Prints one JSON line per (kind, H) cell.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rl_amd import ops
from rl_amd.modules import GRUCell, LSTMCell, gru_scan, lstm_scan


def time_fn(fn, reps, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1000.0


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--reps", type=int, default=20)
    p.add_argument("--batch", type=int, default=256)
    p.add_argument("--horizon", type=int, default=64)
    p.add_argument("--input", type=int, default=32)
    args = p.parse_args()
    assert torch.cuda.is_available()
    device = "cuda"
    B, T, F = args.batch, args.horizon, args.input

    for kind in ("gru", "lstm"):
        for H in (128, 256):
            torch.manual_seed(0)
            cell_cls = GRUCell if kind == "gru" else LSTMCell
            cell = cell_cls(F, H, device=device)
            x = torch.randn(B, T, F, device=device)
            is_init = torch.rand(B, T, device=device) < 0.05
            is_init[:, 0] = True
            g = torch.randn(B, T, H, device=device)

            def run(path):
                def body():
                    cell.zero_grad(set_to_none=True)
                    xx = x.clone().requires_grad_()
                    if kind == "gru":
                        ys = (
                            ops.gru_train(cell, xx, is_init)[0]
                            if path == "fused"
                            else gru_scan(cell, xx, is_init)[0]
                        )
                    else:
                        ys = (
                            ops.lstm_train(cell, xx, is_init)[0]
                            if path == "fused"
                            else lstm_scan(cell, xx, is_init)[0]
                        )
                    (ys * g).sum().backward()

                return body

            ms_scan = time_fn(run("scan"), args.reps)
            ms_fused = time_fn(run("fused"), args.reps)
            print(
                json.dumps(
                    {
                        "metric": f"rnn_train_ms_{kind}_h{H}",
                        "kind": kind,
                        "H": H,
                        "B": B,
                        "T": T,
                        "ms_python_scan": round(ms_scan, 3),
                        "ms_fused": round(ms_fused, 3),
                        "speedup": round(ms_scan / ms_fused, 2),
                        "lds_resident_w": (
                            ops.gru_train_lds_ok(H)
                            if kind == "gru"
                            else ops.lstm_train_lds_ok(H)
                        ),
                    }
                )
            )


if __name__ == "__main__":
    main()
