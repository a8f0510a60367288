"""IMPALA benchmark — Atari-Pong-shaped actor-learner on MI355X
(BASELINE.json config 4: "IMPALA Atari Pong, MultiSyncDataCollector
sharded across 8xMI355X (RCCL all-gather rollouts + V-trace HIP
kernel)").

One step = collect a T-step unroll from B synthetic Atari envs
(uint8 [4, 84, 84] frames, 6 actions — Pong's shape) with the behavior
policy, then one learner update: V-trace advantages (fused HIP kernel,
csrc/value_scan.hip) + policy-gradient + value + entropy losses, Adam.

Single-process benchmark; the driver's multi-GPU run shards ranks via
torch.distributed (each rank collects its own unroll, gradients
all-reduce over RCCL) when launched with torchrun — same contract as
bench.py.  Metric: env frames/s (whole job).
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rl_amd import ops


class SyntheticAtariEnv:
    """GPU-resident Pong-shaped env: uint8 [4, 84, 84] frames evolve by a
    cheap device-side rule; reward depends on the action matching a
    hidden per-env target that drifts."""

    def __init__(self, batch_size: int, device, frame_shape=(4, 84, 84), n_actions: int = 6):
        self.B = batch_size
        self.device = device
        self.frame_shape = frame_shape
        self.n_actions = n_actions
        self.state = torch.randint(
            0, 255, (batch_size, *frame_shape), dtype=torch.uint8, device=device
        )
        self.target = torch.randint(0, n_actions, (batch_size,), device=device)
        self.t = torch.zeros(batch_size, device=device)

    def reset(self):
        self.state.random_(0, 255)
        self.t.zero_()
        return self.state

    def step(self, action: torch.Tensor):
        # scroll frames and inject noise — keeps the conv net honest.
        # All state updates are IN-PLACE so the step loop is
        # hipGraph-capturable (replays rewrite the same buffers).
        self.state.copy_(torch.roll(self.state, shifts=1, dims=1))
        self.state[:, 0].random_(0, 255)
        reward = (action == self.target).float().unsqueeze(-1) - 0.05
        self.t += 1
        done = (self.t >= 200).unsqueeze(-1)
        dflat = done.squeeze(-1)
        self.t.copy_(torch.where(dflat, torch.zeros_like(self.t), self.t))
        self.target.copy_(torch.where(
            dflat, torch.randint_like(self.target, 0, self.n_actions), self.target
        ))
        return self.state, reward, done


class ImpalaNet(torch.nn.Module):
    """Shallow IMPALA conv torso (Espeholt et al. 2018) + policy/value."""

    def __init__(self, n_actions: int, device=None):
        super().__init__()
        self.conv = torch.nn.Sequential(
            torch.nn.Conv2d(4, 32, 8, stride=4, device=device),
            torch.nn.ReLU(),
            torch.nn.Conv2d(32, 64, 4, stride=2, device=device),
            torch.nn.ReLU(),
            torch.nn.Conv2d(64, 64, 3, stride=1, device=device),
            torch.nn.ReLU(),
            torch.nn.Flatten(),
        )
        self.fc = torch.nn.Sequential(
            torch.nn.Linear(64 * 7 * 7, 512, device=device), torch.nn.ReLU()
        )
        self.policy = torch.nn.Linear(512, n_actions, device=device)
        self.value = torch.nn.Linear(512, 1, device=device)

    def forward(self, frames: torch.Tensor):
        x = frames.float() / 255.0
        h = self.fc(self.conv(x))
        return self.policy(h), self.value(h)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--envs", type=int, default=512)
    p.add_argument("--unroll", type=int, default=20)
    p.add_argument("--graph", dest="graph", action="store_true",
                   help="hipGraph-capture the T-step behavior rollout")
    p.add_argument("--no-graph", dest="graph", action="store_false")
    p.set_defaults(graph=True)
    args = p.parse_args()

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    cuda = torch.cuda.is_available()
    device = torch.device(f"cuda:{local_rank}" if cuda else "cpu")
    if cuda:
        torch.cuda.set_device(device)
    distributed = world > 1
    if distributed:
        torch.distributed.init_process_group(backend="nccl" if cuda else "gloo")
    torch.manual_seed(7 + rank)

    B, T, A = args.envs, args.unroll, 6
    # bf16 compute (no fp32 MFMA on CDNA4; see bench_sac r32 finding)
    autocast = torch.autocast("cuda", dtype=torch.bfloat16, enabled=cuda,
                              cache_enabled=False)
    env = SyntheticAtariEnv(B, device)
    net = ImpalaNet(A, device=device)
    # behavior policy: slightly stale copy of the learner (IMPALA lag)
    behavior = ImpalaNet(A, device=device)
    behavior.load_state_dict(net.state_dict())
    try:
        optim = torch.optim.Adam(net.parameters(), lr=6e-4, fused=cuda,
                                 capturable=bool(args.graph and cuda and not distributed))
    except Exception:
        optim = torch.optim.Adam(net.parameters(), lr=6e-4)
    gamma = 0.99

    _b_params = [p.detach() for p in behavior.parameters()] + list(behavior.buffers())
    _n_params = [p.detach() for p in net.parameters()] + list(net.buffers())

    obs = env.reset()
    # static rollout buffers: reused across iterations so the whole
    # T-step behavior rollout can be hipGraph-captured and replayed
    frames = torch.empty(B, T, *env.frame_shape, dtype=torch.uint8, device=device)
    actions = torch.empty(B, T, dtype=torch.long, device=device)
    log_mu = torch.empty(B, T, device=device)
    rewards = torch.empty(B, T, 1, device=device)
    dones = torch.empty(B, T, 1, dtype=torch.bool, device=device)

    def rollout_body():
        nonlocal obs
        with torch.no_grad(), autocast:
            for t in range(T):
                logits, _ = behavior(obs)
                dist = torch.distributions.Categorical(logits=logits, validate_args=False)
                a = dist.sample()
                frames[:, t].copy_(obs)
                actions[:, t].copy_(a)
                log_mu[:, t].copy_(dist.log_prob(a).float())
                _, r, d = env.step(a)  # obs aliases env.state (in-place)
                rewards[:, t].copy_(r)
                dones[:, t].copy_(d)

    rollout = rollout_body
    if args.graph and cuda:
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    rollout_body()
            torch.cuda.current_stream().wait_stream(side)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                rollout_body()
            rollout = g.replay
        except Exception:
            import traceback

            traceback.print_exc()
            args.graph = False

    def learner_body():
        # learner: recompute pi under current weights over the unroll
        flat = frames.reshape(B * T, *env.frame_shape)
        with autocast:
            logits, values = net(flat)
        logits = logits.float()
        values = values.float()
        logits = logits.reshape(B, T, A)
        values = values.reshape(B, T, 1)
        dist = torch.distributions.Categorical(logits=logits, validate_args=False)
        log_pi = dist.log_prob(actions)
        with torch.no_grad(), autocast:
            _, next_v_last = net(obs)
        next_v_last = next_v_last.float()
        next_values = torch.cat([values[:, 1:], next_v_last.reshape(B, 1, 1)], 1)
        adv, vs = ops.vtrace(
            gamma,
            log_pi.detach().unsqueeze(-1),
            log_mu.unsqueeze(-1),
            values.detach(),
            next_values.detach(),
            rewards,
            dones,
            dones,
        )
        pg_loss = -(log_pi.unsqueeze(-1) * adv).mean()
        v_loss = 0.5 * (values - vs).pow(2).mean()
        ent_loss = -0.01 * dist.entropy().mean()
        loss = pg_loss + v_loss + ent_loss
        optim.zero_grad(set_to_none=True)
        loss.backward()
        if distributed:
            grads = [p.grad for p in net.parameters() if p.grad is not None]
            flat_g = torch.cat([g.reshape(-1) for g in grads])
            torch.distributed.all_reduce(flat_g)
            flat_g /= world
            i = 0
            for g in grads:
                g.copy_(flat_g[i : i + g.numel()].reshape(g.shape))
                i += g.numel()
        optim.step()
        # one foreach copy instead of load_state_dict's per-tensor loop
        torch._foreach_copy_(_b_params, _n_params)

    learner = learner_body
    if args.graph and cuda and not distributed:
        # the learner reads only the static rollout buffers + carried
        # obs: capture it as a second graph (the update itself is ~100s
        # of small conv/glue launches at B*T = 10240)
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    learner_body()
            torch.cuda.current_stream().wait_stream(side)
            gl = torch.cuda.CUDAGraph()
            with torch.cuda.graph(gl):
                learner_body()
            learner = gl.replay
        except Exception:
            import traceback

            traceback.print_exc()
            learner = learner_body

    def one_iteration():
        rollout()
        learner()

    for _ in range(args.warmup):
        one_iteration()
    if distributed:
        torch.distributed.barrier()
    if cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_iteration()
    if cuda:
        torch.cuda.synchronize()
    if distributed:
        torch.distributed.barrier()
    dt = time.perf_counter() - t0
    tmax = torch.tensor([dt], device=device if cuda else "cpu")
    if distributed:
        torch.distributed.all_reduce(tmax, op=torch.distributed.ReduceOp.MAX)
    dt = float(tmax.item())
    frames_total = args.steps * B * T * world
    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "impala_frames_per_sec",
                    "value": frames_total / dt,
                    "unit": "frames/s",
                    "n_gpus": world,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": dt / args.steps * 1000,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bf16" if cuda else "fp32",
                    "data": "synthetic",
                    "config": {
                        "model": "impala_shallow_pong",
                        "global_batch": B * T * world,
                        "unroll": T,
                        "n_envs_per_gpu": B,
                        "parallelism": f"dp{world}",
                        "vtrace": "hip_kernel" if cuda and ops.HAS_HIP_EXT else "torch",
                    },
                }
            )
        )
    if distributed:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
