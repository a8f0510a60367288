"""GRPO RLHF loop on a tiny random-init LM (reference
sota-implementations/grpo shape): PromptDataset → ChatEnv →
LLMCollector (sampling generation) → MCAdvantage (per-group whitening)
→ GRPOLoss (+ optional KL-to-reference) → Adam.

Runs anywhere (CPU ok; tiny model).  For production GPT-2 decode on
MI355X see rl_amd.modules.llm.decode.GraphedGPT2Decoder (one hipGraph
replay per generated token — the RLHF bench's rollout engine).

Run: python examples/grpo_llm.py [--iters 4]
"""
from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=4)
    args = p.parse_args()

    try:
        import transformers  # noqa: F401
    except ImportError:
        print("transformers not installed; skipping")
        return

    from rl_amd.data.llm import PromptDataset
    from rl_amd.envs.llm import ChatEnv
    from rl_amd.collectors.llm import LLMCollector
    from rl_amd.modules.llm import TransformersWrapper
    from rl_amd.objectives import GRPOLoss, MCAdvantage

    torch.manual_seed(0)
    # tiny random-init LM + byte-level tokenizer (no network needed);
    # swap in GPT2LMHeadModel + AutoTokenizer for the real thing
    from rl_amd.testing.llm_mocks import ByteTokenizer, make_tiny_lm

    lm = make_tiny_lm()
    tok = ByteTokenizer()
    prompts = PromptDataset(
        ["2+2?", "3+3?", "4+4?", "5+5?", "6+6?", "7+7?"], repeat=True
    )
    env = ChatEnv(iter(prompts),
                  reward_fn=lambda h: float(len(h.last_content)))
    policy = TransformersWrapper(lm, tokenizer=tok, generate=True,
                                 max_new_tokens=6)
    actor = TransformersWrapper(lm, tokenizer=tok, generate=False)
    col = LLMCollector(env, policy, dialog_turns_per_batch=4,
                       total_dialog_turns=4 * args.iters)
    adv = MCAdvantage(grpo_size=2)
    loss_mod = GRPOLoss(actor)
    optim = torch.optim.Adam(lm.parameters(), lr=1e-4)

    for i, batch in enumerate(col):
        adv(batch)
        out = loss_mod(batch)
        total = out.get("loss_objective")
        optim.zero_grad()
        total.backward()
        optim.step()
        r = batch.get(("next", "reward")).float().mean().item()
        print(f"iter {i}: loss {float(total):.4f} mean reward {r:.2f}")
    col.shutdown()
    print("done")


if __name__ == "__main__":
    main()
