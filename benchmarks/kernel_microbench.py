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

# wgrad: MFMA split-K vs torch mm (bf16, fp32 accum)
from rl_amd import _C
for K, N, M in [(16384, 64, 64), (65536, 64, 64)]:
    dy = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(K, M, device="cuda", dtype=torch.bfloat16)
    results[f"wgrad_mfma_K{K}_ms"] = timeit_gpu(lambda: _C.wgrad_splitk(dy, x, True), n=50)
    results[f"wgrad_torch_mm_K{K}_ms"] = timeit_gpu(
        lambda: torch.mm(dy.t().float(), x.float()), n=50
    )

# fused actor vs eager policy chain (rollout shape)
from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal
from rl_amd.ops import FusedTanhNormalActor
from rl_amd.tensordict import TensorDict, TensorDictModule
from rl_amd.envs.utils import ExplorationType, set_exploration_type
net = torch.nn.Sequential(
    MLP(in_features=17, out_features=12, num_cells=[64, 64],
        activation_class=torch.nn.Tanh, device="cuda"),
    NormalParamExtractor(),
)
actor = ProbabilisticActor(
    TensorDictModule(net, in_keys=["observation"], out_keys=["loc", "scale"]),
    in_keys=["loc", "scale"], distribution_class=TanhNormal, return_log_prob=True,
)
fa = FusedTanhNormalActor(actor)
td = TensorDict({"observation": torch.randn(4096, 17, device="cuda")}, batch_size=[4096], device="cuda")
with torch.no_grad(), set_exploration_type(ExplorationType.RANDOM):
    results["actor_fused_B4096_ms"] = timeit_gpu(lambda: fa(td.clone(False)), n=50)
    results["actor_eager_B4096_ms"] = timeit_gpu(lambda: actor(td.clone(False)), n=50)

# fused env transition vs eager step
from rl_amd.envs.custom.synthetic import HalfCheetahVec
env = HalfCheetahVec(batch_size=[4096], device="cuda", dtype=torch.float32)
env.reset()
env.enable_capture_mode(True)
act = torch.rand(4096, 6, device="cuda") * 2 - 1
etd = TensorDict({"action": act}, batch_size=[4096], device="cuda")
results["env_step_fused_B4096_ms"] = timeit_gpu(lambda: env._step(etd), n=50)
env._fused_step_ok = False
results["env_step_eager_B4096_ms"] = timeit_gpu(lambda: env._step(etd), n=50)

# fused TanhNormal log-prob / entropy vs eager distribution
from rl_amd import ops as _ops
loc = torch.randn(16384, 6, device="cuda")
scale = torch.rand(16384, 6, device="cuda") * 0.9 + 0.1
with torch.no_grad():
    a = TanhNormal(loc, scale).sample()
results["logprob_fused_N16k_ms"] = timeit_gpu(lambda: _ops.tanh_normal_logprob(loc, scale, a), n=50)
results["logprob_eager_N16k_ms"] = timeit_gpu(lambda: TanhNormal(loc, scale).log_prob(a), n=50)

print(json.dumps(results, indent=1))
open("gpurun_out/kernel_microbench.json", "w").write(json.dumps(results, indent=1))
