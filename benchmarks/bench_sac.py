"""Secondary benchmark — SAC with a 1M-transition prioritized replay
buffer resident in HBM (BASELINE.json config 3: "SAC Humanoid-v4,
1M-transition PrioritizedReplayBuffer resident in HBM on 1 MI355X").

Metric: SAC optimizer samples/sec.  One step = sample 256 from the PER
(device sum-tree inverse-CDF), SAC fwd+bwd over twin Q nets, optimizer
step, priority update (device tree scatter+recompute), plus collecting
256 fresh frames from a Humanoid-shaped vec env into the buffer.

Usage: python benchmarks/bench_sac.py [--steps K] [--warmup W]
Prints one JSON line like bench.py.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rl_amd.data import LazyTensorStorage, TensorDictPrioritizedReplayBuffer
from rl_amd.envs.custom.synthetic import HumanoidVec
from rl_amd.envs.utils import ExplorationType, set_exploration_type
from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal, ValueOperator
from rl_amd.objectives import SACLoss, SoftUpdate
from rl_amd.tensordict import TensorDict, TensorDictModule, stack as td_stack


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch", type=int, default=256)
    p.add_argument("--buffer", type=int, default=1_000_000)
    p.add_argument("--envs", type=int, default=256)
    p.add_argument("--utd", type=int, default=4, help="optim steps per env step batch")
    p.add_argument("--graph", dest="graph", action="store_true",
                   help="hipGraph-capture the SAC update (loss fwd/bwd + Adam + soft-update)")
    p.add_argument("--no-graph", dest="graph", action="store_false")
    p.set_defaults(graph=True)
    args = p.parse_args()

    cuda = torch.cuda.is_available()
    device = torch.device("cuda:0" if cuda else "cpu")
    torch.manual_seed(0)
    # bf16 compute: CDNA4 has no fp32 MFMA — fp32 GEMMs of the 256x256
    # MLPs ran single-workgroup at 80 us each (46% of the step in the
    # SAC rocprof); autocast moves them to the matrix cores
    import os as _os

    _ac_on = cuda and _os.environ.get("RL_AMD_SAC_AUTOCAST", "1") != "0"
    autocast = torch.autocast("cuda", dtype=torch.bfloat16, enabled=_ac_on,
                              cache_enabled=False)

    env = HumanoidVec(batch_size=[args.envs], device=device)
    obs_dim, act_dim = env.obs_dim, env.act_dim
    actor_net = torch.nn.Sequential(
        MLP(in_features=obs_dim, out_features=2 * act_dim, num_cells=[256, 256], device=device),
        NormalParamExtractor(),
    )
    from rl_amd.data import Bounded

    actor = ProbabilisticActor(
        TensorDictModule(actor_net, in_keys=["observation"], out_keys=["loc", "scale"]),
        in_keys=["loc", "scale"],
        distribution_class=TanhNormal,
        return_log_prob=True,
        spec=Bounded(-1.0, 1.0, shape=(act_dim,), device=device),
    )
    qnet = ValueOperator(
        MLP(in_features=obs_dim + act_dim, out_features=1, num_cells=[256, 256], device=device),
        in_keys=["observation", "action"],
    )
    if cuda:
        from rl_amd.ops import HAS_HIP_EXT, convert_linears_to_splitk

        if HAS_HIP_EXT:
            # split-K wgrad Linears BEFORE the loss builds its ensemble
            # and target deep-copies, so every copy shares the class
            convert_linears_to_splitk(actor)
            convert_linears_to_splitk(qnet)
    loss = SACLoss(actor, qnet, num_qvalue_nets=2)
    loss.make_value_estimator()
    loss = loss.to(device)
    if cuda:
        from rl_amd.ops import HAS_HIP_EXT, enable_splitk_bf16_cache

        if HAS_HIP_EXT:
            # one bf16 weight cast per STEP for online, ensemble and
            # target nets (autocast re-casts every call under capture)
            enable_splitk_bf16_cache(loss)
    use_graph = bool(args.graph and cuda)
    try:
        optim = torch.optim.Adam(loss.parameters(), lr=3e-4,
                                 capturable=use_graph, fused=cuda)
    except Exception:
        optim = torch.optim.Adam(loss.parameters(), lr=3e-4, capturable=use_graph)
    updater = SoftUpdate(loss, tau=0.005)

    rb = TensorDictPrioritizedReplayBuffer(
        storage=LazyTensorStorage(args.buffer, device=device),
        batch_size=args.batch,
        alpha=0.7,
        beta=0.5,
    )

    # collect through the PUBLIC Collector API: its GPU fast path
    # captures the policy+env step as a hipGraph (GraphedRollout) and the
    # iterator extends the replay buffer (only the writer index stays on
    # the host).  AutocastPolicy puts the 256x256 actor GEMMs on the
    # matrix cores with float32 outputs into the buffer.
    from rl_amd.collectors import Collector
    from rl_amd.modules import AutocastPolicy

    collector = Collector(
        env,
        AutocastPolicy(actor),
        frames_per_batch=args.envs,
        total_frames=-1,
        replay_buffer=rb,
        use_graph="auto" if (args.graph and cuda) else False,
    )
    col_iter = iter(collector)

    def collect():
        next(col_iter)

    def update_body(batch):
        with autocast:
            out = loss(batch)
        total = out.get("loss_actor") + out.get("loss_qvalue") + out.get("loss_alpha")
        # set_to_none even under capture: grads live in the graph pool
        # (stable across replays) and the zero-fill + accumulate-add per
        # parameter disappears (measured on the PPO step, r2)
        optim.zero_grad(set_to_none=True)
        total.backward()
        optim.step()
        updater.step()
        if cuda:
            from rl_amd.ops import refresh_splitk_caches

            refresh_splitk_caches(loss)

    # hipGraph-captured update: the SAC update at batch 256 is pure launch
    # overhead (hundreds of ~5us kernels); a single graph replay removes
    # it.  PER sample / priority write stay eager (buffer length grows).
    static_batch = None
    static_keys = None
    graph = None

    def capture_update():
        nonlocal static_batch, static_keys, graph
        static_batch = rb.sample().clone()
        # the loss writes td_error into the static buffer during capture;
        # replays only refresh the keys a fresh sample actually carries
        static_keys = list(static_batch.keys(True, True))
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                update_body(static_batch)
        torch.cuda.current_stream().wait_stream(side)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            update_body(static_batch)
        graph = g

    def _sample_into_static():
        # gather straight into the static graph buffers (index_select
        # with out=) instead of materializing a fresh batch + copying
        idx, info = rb._sampler.sample(rb._storage, args.batch)
        idx = idx.to(device)
        src = rb._storage._storage
        for k in static_keys:
            if k == "index":
                static_batch.get(k).copy_(idx)
            elif k == "_weight":
                static_batch.get(k).copy_(info["_weight"].to(device))
            else:
                torch.index_select(src.get(k), 0, idx, out=static_batch.get(k))

    def train_step():
        for _ in range(args.utd):
            if graph is not None:
                _sample_into_static()
                graph.replay()
                rb.update_tensordict_priority(static_batch)
            else:
                batch = rb.sample()
                update_body(batch)
                rb.update_tensordict_priority(batch)

    # prefill
    while len(rb) < 4 * args.batch:
        collect()

    if use_graph:
        try:
            capture_update()
        except Exception:
            import traceback

            traceback.print_exc()
            graph = None
            use_graph = False

    for _ in range(args.warmup):
        collect()
        train_step()
    if cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        collect()
        train_step()
    if cuda:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    samples = args.steps * args.utd * args.batch
    print(
        json.dumps(
            {
                "metric": "sac_per_samples_per_sec",
                "value": samples / dt,
                "unit": "samples/s",
                "n_gpus": 1,
                "steps": args.steps,
                "warmup": args.warmup,
                "ms_per_step": dt / args.steps * 1000,
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": None,
                "dtype": "bf16" if cuda else "fp32",
                "data": "synthetic",
                "config": {
                    "model": "sac_humanoid_mlp256x256",
                    "global_batch": args.batch,
                    "buffer_size": args.buffer,
                    "utd": args.utd,
                    "parallelism": "dp1",
                    "buffer_device": str(device),
                    "update_graph": bool(graph is not None),
                    "collect_graph": bool(
                        collector._graphed is not None
                        and (collector._graphed.captured or collector._graphed.mega)
                    ),
                    "library_api_collection": True,
                },
            }
        )
    )


if __name__ == "__main__":
    main()
