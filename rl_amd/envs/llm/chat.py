"""ChatEnv — conversation environment for LLM RL.

Reference: pytorch/rl torchrl/envs/llm/chat.py (ChatEnv): reset seeds a
conversation from a prompt dataloader, step appends the policy's response
and (optionally) an env/user turn; reward comes from transforms.
"""
from __future__ import annotations

from typing import Any, Callable, Iterable, List, Optional

import torch

from ...data.llm.history import History
from ...data.tensor_specs import Composite, NonTensor, Unbounded
from ...tensordict import NonTensorData, TensorDict, TensorDictBase
from ..common import EnvBase

__all__ = ["ChatEnv"]


class ChatEnv(EnvBase):
    """One conversation per env; batch_size conversations in parallel.

    ``dataloader`` yields prompt strings (or dicts with "prompt"); an
    optional ``reward_fn(history) -> float`` scores the assistant's last
    message at each step; ``max_turns`` truncates.
    """

    def __init__(
        self,
        dataloader: Optional[Iterable] = None,
        *,
        batch_size=(),
        system_prompt: Optional[str] = None,
        reward_fn: Optional[Callable[[History], float]] = None,
        max_turns: int = 1,
        device=None,
    ):
        if isinstance(batch_size, int):
            batch_size = (batch_size,)
        super().__init__(device=device, batch_size=batch_size)
        self.dataloader = iter(dataloader) if dataloader is not None else None
        self.system_prompt = system_prompt
        self.reward_fn = reward_fn
        self.max_turns = max_turns
        bs = self.batch_size
        self.observation_spec = Composite(
            {"history": NonTensor(example_data={})},
            shape=bs,
            device=self.device,
        )
        self.full_action_spec = Composite(
            {"text_response": NonTensor(example_data="")}, shape=bs, device=self.device
        )
        self.reward_spec = Unbounded(shape=(*bs, 1), device=self.device)
        self._turns: Optional[torch.Tensor] = None

    def _next_prompts(self, n: int) -> List[str]:
        prompts = []
        for _ in range(n):
            if self.dataloader is None:
                prompts.append("Hello")
                continue
            try:
                item = next(self.dataloader)
            except StopIteration:
                item = "Hello"
            if isinstance(item, dict):
                item = item.get("prompt", item.get("text", ""))
            prompts.append(str(item))
        return prompts

    def _reset(self, tensordict: Optional[TensorDictBase] = None, **kwargs) -> TensorDictBase:
        bs = self.batch_size
        n = max(1, int(torch.tensor(bs).prod())) if bs else 1
        prompts = self._next_prompts(n)
        h = History(batch_size=(n,) if bs else ())
        if bs:
            h.roles = [[] for _ in range(n)]
            h.contents = [[] for _ in range(n)]
        for i, p in enumerate(prompts):
            if self.system_prompt:
                h.append("system", self.system_prompt, index=i if bs else None)
                if not bs:
                    break
        for i, p in enumerate(prompts):
            h.append("user", p, index=i if bs else None)
            if not bs:
                break
        self._turns = torch.zeros((*bs, 1), device=self.device)
        td = TensorDict({}, batch_size=bs, device=self.device)
        td.set_non_tensor("history", {"roles": h.roles, "contents": h.contents})
        td.set("done", torch.zeros((*bs, 1), dtype=torch.bool, device=self.device))
        td.set("terminated", torch.zeros((*bs, 1), dtype=torch.bool, device=self.device))
        return td

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        bs = self.batch_size
        hist_data = tensordict.get_non_tensor("history")
        h = History(batch_size=bs)
        h.roles = [list(r) for r in hist_data["roles"]]
        h.contents = [list(c) for c in hist_data["contents"]]
        # if the policy already appended the assistant turn (wrapper does),
        # nothing to add; otherwise append text_response
        resp = tensordict.get_non_tensor("text_response", None)
        if resp is not None and (not h.roles[0] or h.roles[0][-1] != "assistant"):
            if isinstance(resp, str):
                resp = [resp]
            for i, r in enumerate(resp):
                h.append("assistant", r, index=i if bs else None)
                if not bs:
                    break
        self._turns = self._turns + 1
        if self.reward_fn is not None:
            convs = [h[i] for i in range(len(h.roles))] if bs else [h]
            rewards = torch.tensor(
                [[self.reward_fn(c)] for c in convs],
                dtype=torch.float32,
                device=self.device,
            ).reshape(*bs, 1)
        else:
            rewards = torch.zeros((*bs, 1), device=self.device)
        done = self._turns >= self.max_turns
        out = TensorDict({}, batch_size=bs, device=self.device)
        out.set_non_tensor("history", {"roles": h.roles, "contents": h.contents})
        out.set("reward", rewards)
        out.set("done", done)
        out.set("terminated", done)
        return out

    def _set_seed(self, seed):
        return seed
