"""LLM reward models.

Reference: pytorch/rl torchrl/modules/models/llm.py:18 (GPT2RewardModel):
a causal LM with its LM head swapped for a scalar reward head, plus the
pairwise (chosen vs rejected) reward loss of InstructGPT-style RLHF.

rl_amd form: works with ANY transformers causal-LM (not just GPT2) or a
user-supplied backbone module; reward at every token plus the
end-of-sequence score.
"""
from __future__ import annotations

import importlib.util
from typing import Optional, Tuple

import torch
from torch import nn

__all__ = ["RewardModel", "GPT2RewardModel"]

_has_transformers = importlib.util.find_spec("transformers") is not None


class RewardModel(nn.Module):
    """Causal-LM backbone + scalar reward head.

    ``forward(input_ids, attention_mask)`` returns ``(rewards,
    end_scores)``: per-token rewards ``[B, T]`` and the reward at the
    last non-padding token ``[B, 1]``.
    """

    def __init__(self, model=None, model_path: Optional[str] = None, pad_token_id: int = 0):
        super().__init__()
        if model is None:
            if not _has_transformers:
                raise ImportError("transformers is required to build from model_path")
            from transformers import AutoModelForCausalLM

            model = AutoModelForCausalLM.from_pretrained(model_path)
        # backbone: the transformer body (strip the LM head)
        self.backbone = getattr(model, "transformer", None) or getattr(
            model, "model", model
        )
        hidden = getattr(model.config, "n_embd", None) or getattr(
            model.config, "hidden_size"
        )
        self.reward_head = nn.Linear(hidden, 1, bias=False)
        self.pad_token_id = pad_token_id

    def forward(
        self, input_ids: torch.Tensor, attention_mask: Optional[torch.Tensor] = None
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        out = self.backbone(input_ids=input_ids, attention_mask=attention_mask)
        hidden = out.last_hidden_state if hasattr(out, "last_hidden_state") else out[0]
        rewards = self.reward_head(hidden).squeeze(-1)  # [B, T]
        if attention_mask is not None:
            last = attention_mask.long().sum(-1).clamp_min(1) - 1
        else:
            pad = input_ids == self.pad_token_id
            last = (~pad).long().sum(-1).clamp_min(1) - 1
        end_scores = rewards.gather(-1, last.unsqueeze(-1))
        return rewards, end_scores

    @staticmethod
    def compute_reward_loss(chosen_scores: torch.Tensor, rejected_scores: torch.Tensor) -> torch.Tensor:
        """Pairwise preference loss: -log sigmoid(r_chosen - r_rejected)."""
        return -torch.nn.functional.logsigmoid(chosen_scores - rejected_scores).mean()


class GPT2RewardModel(RewardModel):
    """Name-compatible alias building from a GPT2 checkpoint path
    (reference llm.py:18)."""

    def __init__(self, model_path: Optional[str] = None, pad_token_id: int = 0, model=None):
        super().__init__(model=model, model_path=model_path, pad_token_id=pad_token_id)
