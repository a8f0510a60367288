"""TinyVLA — a dependency-free reference VLA policy.

Reference: pytorch/rl torchrl/modules/vla/models.py:31 (TinyVLA): small
conv image encoder + proprioceptive MLP + hashed language-instruction
embedding, fused into a trunk feeding a continuous action-chunk head or
a discrete action-token head.  Exercises the VLA pipeline
(ToyVLAEnv, ActionChunkTransform, ActionTokenizerTransform, ACT/BC
losses) end-to-end with no external model weights.
"""
from __future__ import annotations

import hashlib
from typing import List, Optional, Sequence, Union

import torch
from torch import nn

from ..tensordict import TensorDictBase

__all__ = ["TinyVLA", "LeRobotPolicyWrapper"]


def _hash_instruction(text: str, vocab: int) -> int:
    return int(hashlib.sha1(text.encode()).hexdigest(), 16) % vocab


class TinyVLA(nn.Module):
    """Language-conditioned chunk policy over the VLA schema.

    Reads ``("observation", "image")`` (uint8 CHW), optionally
    ``("observation", "state")``, and the non-tensor
    ``language_instruction``; writes ``("vla_action", "chunk")``
    ``[B, H, A]`` (continuous head) or ``action_tokens`` ``[B, H, A]``
    (token head), plus a flat ``action`` = first chunk step.
    """

    in_keys = [("observation", "image"), ("observation", "state"), "language_instruction"]
    out_keys = [("vla_action", "chunk"), "action"]

    def __init__(
        self,
        *,
        action_dim: int,
        chunk_size: int,
        action_head: str = "continuous",
        vocab_size: int = 256,
        use_state: bool = True,
        state_dim: Optional[int] = None,
        hidden_dim: int = 128,
        text_vocab: int = 256,
        text_dim: int = 32,
        device=None,
    ):
        super().__init__()
        if action_head not in ("continuous", "tokens"):
            raise ValueError("action_head must be 'continuous' or 'tokens'")
        self.action_dim = action_dim
        self.chunk_size = chunk_size
        self.action_head_kind = action_head
        self.vocab_size = vocab_size
        self.use_state = use_state
        self.text_vocab = text_vocab
        self.encoder = nn.Sequential(
            nn.Conv2d(3, 16, 3, stride=2, padding=1, device=device),
            nn.ReLU(),
            nn.Conv2d(16, 32, 3, stride=2, padding=1, device=device),
            nn.ReLU(),
            nn.AdaptiveAvgPool2d(4),
            nn.Flatten(),
        )
        self.text_embed = nn.Embedding(text_vocab, text_dim, device=device)
        self._img_feat = 32 * 4 * 4
        self.state_mlp = (
            nn.Sequential(nn.LazyLinear(hidden_dim // 2, device=device), nn.ReLU())
            if use_state
            else None
        )
        trunk_in = self._img_feat + text_dim + (hidden_dim // 2 if use_state else 0)
        self.trunk = nn.Sequential(
            nn.Linear(trunk_in, hidden_dim, device=device),
            nn.ReLU(),
            nn.Linear(hidden_dim, hidden_dim, device=device),
            nn.ReLU(),
        )
        out = chunk_size * action_dim * (vocab_size if action_head == "tokens" else 1)
        self.head = nn.Linear(hidden_dim, out, device=device)

    def _embed_text(self, instruction, batch: int, device) -> torch.Tensor:
        if isinstance(instruction, str):
            idx = [_hash_instruction(instruction, self.text_vocab)] * batch
        else:
            texts = list(instruction)
            if len(texts) == 1:
                texts = texts * batch
            idx = [_hash_instruction(t, self.text_vocab) for t in texts]
        return self.text_embed(torch.tensor(idx, device=device))

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        image = td.get(("observation", "image")).float() / 255.0
        B = image.shape[0]
        feats = [self.encoder(image)]
        instruction = td.get_non_tensor("language_instruction", "")
        feats.append(self._embed_text(instruction, B, image.device))
        if self.use_state:
            state = td.get(("observation", "state"))
            feats.append(self.state_mlp(state.float()))
        h = self.trunk(torch.cat(feats, dim=-1))
        out = self.head(h)
        if self.action_head_kind == "continuous":
            chunk = torch.tanh(out.reshape(B, self.chunk_size, self.action_dim))
            td.set(("vla_action", "chunk"), chunk)
            td.set("action", chunk[:, 0])
        else:
            logits = out.reshape(B, self.chunk_size, self.action_dim, self.vocab_size)
            tokens = logits.argmax(-1)
            td.set("action_tokens_logits", logits)
            td.set("action_tokens", tokens)
            td.set("action", tokens[:, 0])
        return td


class LeRobotPolicyWrapper(torch.nn.Module):
    """Adapt a LeRobot-style pretrained chunk policy to the rl_amd VLA
    schema (reference modules/vla/wrappers.py:24) — gated: `lerobot` is
    not installed in this image; TinyVLA covers the schema offline."""

    def __init__(self, *args, **kwargs):
        import importlib.util

        super().__init__()
        if importlib.util.find_spec("lerobot") is None:
            raise ImportError(
                "LeRobotPolicyWrapper requires the `lerobot` package, which "
                "is not installed in this image. Use TinyVLA for offline work."
            )
        raise NotImplementedError("lerobot adapter scaffolding")
