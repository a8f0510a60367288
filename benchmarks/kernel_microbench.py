"""Micro-bench: fused HIP value-scan kernels vs torch doubling-scan vs
sequential loop, and device tree ops — writes JSON to gpurun_out/."""
import json, time, torch, sys
sys.path.insert(0, ".")
from rl_amd import ops
from rl_amd.objectives.value import functional as F

def timeit_gpu(fn, n=50):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3  # ms

results = {}
for B, T in [(4096, 16), (4096, 128), (512, 1024), (64, 8192)]:
    val = torch.randn(B, T, 1, device="cuda")
    nval = torch.randn(B, T, 1, device="cuda")
    r = torch.randn(B, T, 1, device="cuda")
    done = torch.rand(B, T, 1, device="cuda") < 0.02
    term = done.clone()
    results[f"gae_hip_B{B}_T{T}_ms"] = timeit_gpu(lambda: ops.gae(0.99, 0.95, val, nval, r, done, term))
    results[f"gae_torchscan_B{B}_T{T}_ms"] = timeit_gpu(lambda: F.vec_generalized_advantage_estimate(0.99, 0.95, val, nval, r, done, term))
    if T <= 128:
        results[f"gae_seqloop_B{B}_T{T}_ms"] = timeit_gpu(lambda: F.generalized_advantage_estimate(0.99, 0.95, val, nval, r, done, term), n=5)

# device tree ops at 1M capacity
from rl_amd.ops import DeviceSumTree
tree = DeviceSumTree(1_000_000, device="cuda")
idx = torch.randint(0, 1_000_000, (100_000,), device="cuda")
vals = torch.rand(100_000, device="cuda").double() + 1e-3
tree.update(idx, vals)
bidx = torch.randint(0, 1_000_000, (256,), device="cuda")
bvals = torch.rand(256, device="cuda").double() + 1e-3
results["tree_update256_ms"] = timeit_gpu(lambda: tree.update(bidx, bvals))
mass = torch.rand(256, device="cuda").double() * tree.total()
results["tree_sample256_ms"] = timeit_gpu(lambda: tree.scan_lower_bound(torch.rand(256, device="cuda").double() * tree.total()))

# fused GRU vs scan
from rl_amd.modules import GRUCell
from rl_amd.modules.tensordict_module.rnn import gru_scan
cell = GRUCell(64, 128, device="cuda")
x = torch.randn(256, 200, 64, device="cuda")
ii = torch.rand(256, 200, device="cuda") < 0.02
with torch.no_grad():
    results["gru_fused_B256_T200_H128_ms"] = timeit_gpu(lambda: ops.gru_fused(cell, x, ii), n=20)
    results["gru_scan_B256_T200_H128_ms"] = timeit_gpu(lambda: gru_scan(cell, x, ii), n=20)

x4 = torch.randn(4096, 64, 64, device="cuda")
ii4 = torch.rand(4096, 64, device="cuda") < 0.02
with torch.no_grad():
    results["gru_fused_B4096_T64_H128_ms"] = timeit_gpu(lambda: ops.gru_fused(cell, x4, ii4), n=20)
    results["gru_scan_B4096_T64_H128_ms"] = timeit_gpu(lambda: gru_scan(cell, x4, ii4), n=20)
from rl_amd.modules import LSTMCell
from rl_amd.modules.tensordict_module.rnn import lstm_scan
lcell = LSTMCell(64, 128, device="cuda")
with torch.no_grad():
    results["lstm_fused_B4096_T64_H128_ms"] = timeit_gpu(lambda: ops.lstm_fused(lcell, x4, ii4), n=20)
    results["lstm_scan_B4096_T64_H128_ms"] = timeit_gpu(lambda: lstm_scan(lcell, x4, ii4), n=20)

print(json.dumps(results, indent=1))
open("gpurun_out/kernel_microbench.json", "w").write(json.dumps(results, indent=1))
