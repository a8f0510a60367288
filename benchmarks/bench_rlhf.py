"""RLHF PPO benchmark — GPT-2-small policy on MI355X
(BASELINE.json config 5: "RLHF PPO on GPT-2-small, 8-GPU distributed
collectors over xGMI feeding a DDP learner").

One step = the full RLHF PPO cycle on synthetic prompts:
  1. rollout: generate G new tokens for B prompts with the policy
     (sampling, per-token behavior log-probs recorded),
  2. score: synthetic sequence rewards (no reward model download),
  3. learn: recompute token log-probs under current weights, PPO
     clipped policy-gradient + value-head loss + k3 KL penalty to a
     frozen reference copy, one Adam step (DDP all-reduce when
     world > 1).

GPT-2-small architecture (124M params) from transformers with
random-init weights (no network); prompts are random token ids.
Metric: generated tokens/s through the WHOLE cycle (whole job).
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


def build_gpt2_small(device, attn: str = "sdpa"):
    from transformers import GPT2Config, GPT2LMHeadModel

    cfg = GPT2Config()  # gpt2-small: 12 layers, 768 hidden, 12 heads
    # attn="eager": torch SDPA with a StaticCache segfaults on
    # ROCm 7.2 + transformers 5.15 (single-token decode path)
    cfg._attn_implementation = attn
    model = GPT2LMHeadModel(cfg).to(device)
    model.config.pad_token_id = cfg.eos_token_id
    return model, cfg


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=8)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--batch", type=int, default=256, help="prompts per rank")
    p.add_argument("--prompt-len", type=int, default=64)
    p.add_argument("--gen-len", type=int, default=32)
    p.add_argument("--decode-mode",
                   choices=["generate", "manual", "graphdec", "static",
                            "graph"],
                   default="graphdec",
                   help="HF generate (default) / eager static-KV loop / "
                        "hipGraph-captured loop.  static and graph SEGFAULT on "
                        "this stack (ROCm 7.2 + transformers 5.15 GPT2 "
                        "StaticCache decode, with both sdpa and eager "
                        "attention) — kept opt-in for newer stacks")
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
    torch.manual_seed(11 + rank)

    attn = "eager" if args.decode_mode in ("static", "graph") else "sdpa"
    policy, cfg = build_gpt2_small(device, attn)
    ref = build_gpt2_small(device, attn)[0]
    ref.eval()
    for prm in ref.parameters():
        prm.requires_grad_(False)
    value_head = torch.nn.Linear(cfg.n_embd, 1, device=device)
    optim = torch.optim.AdamW(
        list(policy.parameters()) + list(value_head.parameters()), lr=1e-5
    )
    V = cfg.vocab_size
    B, P, G = args.batch, args.prompt_len, args.gen_len
    autocast = torch.autocast("cuda", dtype=torch.bfloat16, enabled=cuda, cache_enabled=False)
    kl_coef, clip_eps = 0.05, 0.2

    def token_log_probs(model, ids, mask):
        out = model.transformer(input_ids=ids, attention_mask=mask)
        hidden = out.last_hidden_state
        logits = model.lm_head(hidden)
        lp = logits[:, P - 1 : -1].log_softmax(-1)
        resp = ids[:, P:]
        return lp.gather(-1, resp.unsqueeze(-1)).squeeze(-1), hidden

    # ---- graph-captured static-KV decode -------------------------------- #
    # HF generate runs ~6 ms of host-side logic per token; a captured
    # decode step (single-token forward + multinomial + buffer writes)
    # replays in a fraction of that.
    decode_graph = None
    use_static = cuda and args.decode_mode in ("static", "graph")
    if use_static:
        from transformers import StaticCache

        cache = StaticCache(
            config=cfg, max_batch_size=B, max_cache_len=P + G, device=device,
            dtype=torch.bfloat16,
        )
        cur = torch.zeros(B, 1, dtype=torch.long, device=device)
        pos = torch.zeros(1, dtype=torch.long, device=device)
        pos_out = torch.zeros(1, dtype=torch.long, device=device)
        gen_buf = torch.zeros(B, G, dtype=torch.long, device=device)
        prompt_buf = torch.zeros(B, P, dtype=torch.long, device=device)

        def decode_body():
            with autocast:
                out = policy(
                    input_ids=cur, past_key_values=cache, use_cache=True,
                    cache_position=pos,
                )
            logits = out.logits[:, -1].float()
            nxt = torch.multinomial(torch.softmax(logits, -1), 1)
            cur.copy_(nxt)
            gen_buf.index_copy_(1, pos_out, nxt)
            pos.add_(1)
            pos_out.add_(1)

        def prefill(prompts):
            prompt_buf.copy_(prompts)
            with torch.no_grad(), autocast:
                out = policy(
                    input_ids=prompt_buf, past_key_values=cache, use_cache=True,
                    cache_position=torch.arange(P, device=device),
                )
            logits = out.logits[:, -1].float()
            first = torch.multinomial(torch.softmax(logits, -1), 1)
            cur.copy_(first)
            gen_buf[:, 0:1].copy_(first)
            pos.fill_(P)
            pos_out.fill_(1)

        if args.decode_mode == "graph":
            try:
                with torch.no_grad():
                    prefill(torch.randint(0, V, (B, P), device=device))
                    side = torch.cuda.Stream()
                    side.wait_stream(torch.cuda.current_stream())
                    with torch.cuda.stream(side):
                        for _ in range(3):
                            decode_body()
                    torch.cuda.current_stream().wait_stream(side)
                    g = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(g):
                        decode_body()
                    decode_graph = g
            except Exception:
                import traceback

                traceback.print_exc()
                decode_graph = None

    graphdec = None
    if args.decode_mode == "graphdec" and cuda:
        from rl_amd.modules.llm.decode import GraphedGPT2Decoder

        graphdec = GraphedGPT2Decoder(policy, B, P + G, device)
        # warm + capture once (needs a prefilled state)
        graphdec.prefill(torch.randint(0, V, (B, P), device=device))
        if not graphdec.capture():
            print("graphdec capture failed; eager decode body stays",
                  file=sys.stderr)

    def manual_rollout(prompts):
        """Hand-rolled sampling loop over the ordinary dynamic KV cache
        (the same cache `generate` uses — StaticCache segfaults on this
        stack): one prefill + G-1 single-token forwards with none of
        generate's ~6 ms/token host-side logic.  Identical sampling
        distribution (do_sample, top_k=0)."""
        with torch.no_grad(), autocast:
            out = policy(input_ids=prompts, use_cache=True)
            past = out.past_key_values
            logits = out.logits[:, -1].float()
            toks = []
            for i in range(G):
                nxt = torch.multinomial(torch.softmax(logits, -1), 1)
                toks.append(nxt)
                if i + 1 < G:
                    out = policy(input_ids=nxt, past_key_values=past,
                                 use_cache=True)
                    past = out.past_key_values
                    logits = out.logits[:, -1].float()
        return torch.cat([prompts] + toks, 1)

    def one_iteration():
        prompts = torch.randint(0, V, (B, P), device=device)
        mask = torch.ones_like(prompts)
        if graphdec is not None:
            gen = graphdec.rollout(prompts, G)
        elif args.decode_mode == "manual":
            gen = manual_rollout(prompts)
        elif use_static:
            with torch.no_grad():
                prefill(prompts)
                if decode_graph is not None:
                    for _ in range(G - 1):
                        decode_graph.replay()
                else:
                    for _ in range(G - 1):
                        decode_body()
            gen = torch.cat([prompt_buf, gen_buf], 1)
        else:
            with torch.no_grad(), autocast:
                gen = policy.generate(
                    input_ids=prompts,
                    attention_mask=mask,
                    max_new_tokens=G,
                    min_new_tokens=G,
                    do_sample=True,
                    top_k=0,
                    pad_token_id=cfg.eos_token_id,
                )
        full_mask = torch.ones_like(gen)
        with torch.no_grad(), autocast:
            old_lp, _ = token_log_probs(policy, gen, full_mask)
            ref_lp, _ = token_log_probs(ref, gen, full_mask)
        # synthetic scalar reward per sequence (stand-in for a reward model)
        reward = (gen[:, P:] % 97).float().mean(-1, keepdim=True) / 97.0
        with autocast:
            new_lp, hidden = token_log_probs(policy, gen, full_mask)
            values = value_head(hidden[:, P - 1 : -1].float()).squeeze(-1)
            # sequence-level advantage broadcast over tokens (GRPO-style
            # whitening across the batch)
            adv = (reward - reward.mean()) / reward.std().clamp_min(1e-4)
            ratio = (new_lp - old_lp).exp()
            g1 = ratio * adv
            g2 = ratio.clamp(1 - clip_eps, 1 + clip_eps) * adv
            pg_loss = -torch.minimum(g1, g2).mean()
            v_loss = 0.5 * (values - reward).pow(2).mean()
            lr_ = ref_lp - new_lp
            kl = (lr_.exp() - 1 - lr_).mean()  # k3 estimator
            loss = pg_loss + v_loss + kl_coef * kl
        optim.zero_grad(set_to_none=True)
        loss.backward()
        if distributed:
            grads = [p.grad for p in policy.parameters() if p.grad is not None]
            grads += [p.grad for p in value_head.parameters() if p.grad is not None]
            flat = torch.cat([g.reshape(-1) for g in grads])
            torch.distributed.all_reduce(flat)
            flat /= world
            i = 0
            for g in grads:
                g.copy_(flat[i : i + g.numel()].reshape(g.shape))
                i += g.numel()
        optim.step()
        if graphdec is not None:
            graphdec.refresh_weights()

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
    tokens = args.steps * B * G * world
    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "rlhf_ppo_gen_tokens_per_sec",
                    "value": tokens / dt,
                    "unit": "tokens/s",
                    "n_gpus": world,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": dt / args.steps * 1000,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bf16",
                    "data": "synthetic",
                    "config": {
                        "model": "gpt2_small_124M_random_init",
                        "global_batch": B * world,
                        "prompt_len": P,
                        "gen_len": G,
                        "parallelism": f"dp{world}",
                        "kl_to_ref": "k3",
                        "decode": ("graph" if decode_graph is not None
                                   else "static" if use_static else "generate"),
                    },
                }
            )
        )
    if distributed:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
