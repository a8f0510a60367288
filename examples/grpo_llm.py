"""GRPO on a tiny random-init LM — the RLHF vertical end to end.

Reference analog: pytorch/rl sota-implementations/grpo.  Pipeline:
PromptDataset → ChatEnv (length reward) → LLMCollector (generation) →
MCAdvantage group baseline → GRPOLoss (+ KL to reference) → Adam.
Runs on CPU in under a minute; swap in a transformers checkpoint and a
real reward model for production use.
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from rl_amd.collectors import LLMCollector
from rl_amd.data.llm import PromptDataset
from rl_amd.envs import ChatEnv
from rl_amd.modules import TransformersWrapper
from rl_amd.objectives import GRPOLoss, MCAdvantage
from rl_amd.testing.llm_mocks import ByteTokenizer, make_tiny_lm


def main(total_turns: int = 16, group: int = 4):
    torch.manual_seed(0)
    lm, tok = make_tiny_lm(), ByteTokenizer()
    prompts = PromptDataset(["2+2?", "3+3?", "capital of France?", "9*9?"], repeat=True)
    env = ChatEnv(iter(prompts), reward_fn=lambda h: float(len(h.last_content)))
    policy = TransformersWrapper(lm, tokenizer=tok, generate=True, max_new_tokens=6)
    col = LLMCollector(env, policy, dialog_turns_per_batch=group, total_dialog_turns=total_turns)
    actor = TransformersWrapper(lm, tokenizer=tok, generate=False)
    loss_mod = GRPOLoss(actor)
    optim = torch.optim.Adam(lm.parameters(), lr=1e-4)
    for i, batch in enumerate(col):
        batch = batch.reshape(-1)
        MCAdvantage(grpo_size=group)(batch)
        out = loss_mod(batch)
        optim.zero_grad()
        out.get("loss_objective").backward()
        optim.step()
        r = batch.get(("next", "reward")).float().mean().item()
        print(f"update {i}: mean reward {r:.2f}, loss {out.get('loss_objective').item():.4f}")


if __name__ == "__main__":
    main()
