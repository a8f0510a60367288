"""rl_amd flagship benchmark — PPO over 4096 GPU-resident vectorized envs.

Metric (BASELINE.json): frames/sec (whole job) for PPO with 4096 vec-envs,
HalfCheetah-v4 shapes (obs 17 / act 6), synthetic dynamics + random-init
weights (no network for datasets), bf16 compute, GAE on the fused HIP scan.

The loop is constructed ONLY from public library classes:
  HalfCheetahVec env → Collector (GPU fast path: mega-kernel / hipGraph
  rollout) → GAE (fused HIP scan) → ClipPPOLoss → Adam, orchestrated by
  trainers.GraphedPPO which captures the ENTIRE iteration as one hipGraph
  at world=1 and, when distributed, captures the minibatch fwd+bwd and
  overlaps the bucketed RCCL gradient all-reduce (parallel.GradAllReducer)
  on a second HIP stream.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
Weak scaling: each rank owns its own 4096 envs; gradients all-reduce.

Secondary configs (same JSON contract): --config sac|impala|rlhf|dqn
dispatch to benchmarks/bench_<config>.py.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from rl_amd.collectors import Collector
from rl_amd.envs.custom.synthetic import HalfCheetahVec
from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal, ValueOperator
from rl_amd.objectives import ClipPPOLoss
from rl_amd.objectives.value.advantages import GAE
from rl_amd.tensordict import TensorDict, TensorDictModule
from rl_amd.trainers import GraphedPPO


def parse_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--config", choices=["ppo", "sac", "impala", "rlhf", "dqn"],
                   default="ppo")
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--envs", type=int, default=4096)
    p.add_argument("--horizon", type=int, default=16, help="env steps per PPO iter")
    p.add_argument("--minibatches", type=int, default=4)
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--hidden", type=int, default=64)
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--backend", type=str, default=None,
                   help="process-group backend override (e.g. gloo for a "
                        "2-rank smoke sharing one GPU)")
    p.add_argument("--seed", type=int, default=1234)
    p.add_argument("--same-seed", action="store_true",
                   help="identical seed on every rank (numerics tests: "
                        "world=N must then reproduce world=1 bit-for-bit)")
    p.add_argument("--dump-params", type=str, default=None,
                   help="save actor+critic state_dict after the timed run")
    p.add_argument("--graph", action="store_true", help="hipGraph capture")
    p.add_argument("--no-graph", dest="graph", action="store_false")
    p.add_argument("--fused-actor", action="store_true",
                   help="(kept for compat; the Collector fast path fuses automatically)")
    p.add_argument("--no-fused-actor", dest="fused_actor", action="store_false")
    p.add_argument("--full-graph", dest="full_graph", action="store_true")
    p.add_argument("--no-full-graph", dest="full_graph", action="store_false")
    p.add_argument("--splitk", dest="splitk", action="store_true",
                   help="split-K HIP wgrad kernel in the update backward")
    p.add_argument("--no-splitk", dest="splitk", action="store_false")
    p.add_argument("--track-reward", action="store_true",
                   help="print mean store reward every 20 steps (learning sanity)")
    p.set_defaults(graph=True, fused_actor=True, full_graph=True, splitk=True)
    return p.parse_args(argv)


def build(args, device):
    env = HalfCheetahVec(batch_size=[args.envs], device=device, dtype=torch.float32)
    obs_dim, act_dim = env.obs_dim, env.act_dim
    actor_net = torch.nn.Sequential(
        MLP(
            in_features=obs_dim,
            out_features=2 * act_dim,
            num_cells=[args.hidden, args.hidden],
            activation_class=torch.nn.Tanh,
            device=device,
        ),
        NormalParamExtractor(),
    )
    actor_mod = TensorDictModule(actor_net, in_keys=["observation"], out_keys=["loc", "scale"])
    actor = ProbabilisticActor(
        actor_mod,
        in_keys=["loc", "scale"],
        distribution_class=TanhNormal,
        return_log_prob=True,
    )
    critic = ValueOperator(
        MLP(
            in_features=obs_dim,
            out_features=1,
            num_cells=[args.hidden, args.hidden],
            activation_class=torch.nn.Tanh,
            device=device,
        ),
        in_keys=["observation"],
    )
    return env, actor, critic


def run_ppo(args):
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))

    cuda = torch.cuda.is_available()
    if args.device is not None:
        device = torch.device(args.device)
    elif cuda:
        device = torch.device(f"cuda:{local_rank}")
    else:
        device = torch.device("cpu")
    if cuda:
        torch.cuda.set_device(device)

    distributed = world > 1
    if distributed and not torch.distributed.is_initialized():
        backend = args.backend or ("nccl" if cuda else "gloo")
        torch.distributed.init_process_group(backend=backend)

    seed = args.seed if args.same_seed else args.seed + rank
    torch.manual_seed(seed)
    env, actor, critic = build(args, device)
    env.set_seed(seed)

    refresh_hook = None
    if args.splitk and cuda:
        # minibatch backward wgrad via the split-K kernel (csrc/wgrad.hip):
        # hipBLASLt runs these skinny [H,16k]x[16k,H] reductions on one
        # workgroup (measured 101us, 12.9% of the step in the r11 profile)
        from rl_amd.ops import (
            HAS_HIP_EXT,
            convert_linears_to_splitk,
            enable_splitk_bf16_cache,
            refresh_splitk_caches,
        )

        if HAS_HIP_EXT:
            convert_linears_to_splitk(actor)
            convert_linears_to_splitk(critic)
            # one weight cast per layer per optimizer step instead of a
            # re-cast on every minibatch call (autocast cache is off
            # under graph capture; r29 profile: ~130 cast kernels/step)
            enable_splitk_bf16_cache(actor)
            enable_splitk_bf16_cache(critic)
            # whole-MLP MFMA kernels for the update phase: 3 GEMMs +
            # bias + tanh in one launch each way (ops.FusedMLP3)
            from rl_amd.ops import fuse_mlp3

            actor.module[0].module[0] = fuse_mlp3(actor.module[0].module[0])
            critic.module = fuse_mlp3(critic.module)

            def refresh_hook():
                refresh_splitk_caches(actor, critic)
        else:
            args.splitk = False

    params = list(actor.parameters()) + list(critic.parameters())
    # fused multi-tensor Adam, capturable for full-step graph capture
    optim_kwargs = dict(lr=3e-4, capturable=bool(args.full_graph and cuda and world == 1))

    def _fused_adam_ok():
        try:
            p = torch.zeros(4, device=device, requires_grad=True)
            p.grad = torch.zeros_like(p)
            torch.optim.Adam([p], fused=True, **optim_kwargs).step()
            return True
        except Exception:
            return False

    if cuda and _fused_adam_ok():
        optim = torch.optim.Adam(params, fused=True, **optim_kwargs)
    else:
        optim = torch.optim.Adam(params, **optim_kwargs)

    loss_mod = ClipPPOLoss(actor, critic, clip_epsilon=0.2, entropy_coeff=0.01,
                           critic_coeff=0.5, normalize_advantage=True)
    gae = GAE(gamma=0.99, lmbda=0.95, value_network=critic, vectorized=True)

    T, B = args.horizon, args.envs
    frames_per_step = T * B

    collector = Collector(
        env,
        actor,
        frames_per_batch=frames_per_step,
        total_frames=-1,
        use_graph="auto" if (args.graph and cuda) else False,
    )

    runner = GraphedPPO(
        collector,
        gae,
        loss_mod,
        optim,
        minibatches=args.minibatches,
        epochs=args.epochs,
        capture="auto" if (args.full_graph and cuda) else False,
        post_optim_hook=refresh_hook,
    )
    runner.initialize()

    for _ in range(args.warmup):
        runner.step()

    if distributed:
        torch.distributed.barrier()
    if cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for it in range(args.steps):
        runner.step()
        if args.track_reward and it % 20 == 0:
            store = collector._graphed.store if collector._graphed is not None else None
            if store is not None:
                r = store.get(("next", "reward")).float().mean().item()
                print(f"[reward] step {it}: {r:.4f}", file=sys.stderr)
    if cuda:
        torch.cuda.synchronize()
    if distributed:
        torch.distributed.barrier()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    if distributed:
        tmax = torch.tensor([elapsed], device=device if cuda else "cpu")
        torch.distributed.all_reduce(tmax, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(tmax.item())

    fps = world * frames_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if args.dump_params:
        sd = {
            "actor": {k: v.detach().cpu() for k, v in actor.state_dict().items()},
            "critic": {k: v.detach().cpu() for k, v in critic.state_dict().items()},
        }
        torch.save(sd, args.dump_params + (f".rank{rank}" if world > 1 else ""))

    graphed = collector._graphed
    if rank == 0:
        result = {
            "metric": "frames_per_sec_ppo_4096envs",
            "value": fps,
            "unit": "frames/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "ppo_halfcheetah_mlp64x64",
                "global_batch": world * frames_per_step,
                "seq_len": T,
                "parallelism": f"dp{world}",
                "n_envs_per_gpu": B,
                "ppo_epochs": args.epochs,
                "minibatches": args.minibatches,
                "library_api": True,
                "hip_graph": bool(graphed is not None and (graphed.captured or graphed.mega)),
                "store_direct_rollout": bool(graphed is not None and graphed.mega),
                "splitk_wgrad": bool(args.splitk and cuda),
                "full_step_graph": runner.full_graph,
                "minibatch_graph": runner.minibatch_graph,
                "overlapped_comm": bool(runner.reducer is not None),
                "merged_loss_kernels": bool(
                    cuda
                    and os.environ.get("RL_AMD_MERGED_LOSS", "1") != "0"
                ),
                "rollout_mfma": bool(
                    cuda
                    and os.environ.get("RL_AMD_ROLLOUT_MFMA", "1") != "0"
                ),
            },
        }
        print(json.dumps(result))
    if distributed:
        torch.distributed.destroy_process_group()


def _dispatch_secondary(config: str, argv):
    """Run benchmarks/bench_<config>.py's main() with the remaining args."""
    import importlib.util

    name = {"sac": "bench_sac", "impala": "bench_impala",
            "rlhf": "bench_rlhf", "dqn": "bench_dqn_cpu"}[config]
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "benchmarks", f"{name}.py")
    spec = importlib.util.spec_from_file_location(name, path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    sys.argv = [path] + argv
    mod.main()


def main():
    argv = sys.argv[1:]
    if "--config" in argv:
        i = argv.index("--config")
        config = argv[i + 1]
        rest = argv[:i] + argv[i + 2 :]
        if config != "ppo":
            _dispatch_secondary(config, rest)
            return
        argv = rest
    args = parse_args(argv)
    run_ppo(args)


if __name__ == "__main__":
    main()
