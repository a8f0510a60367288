"""History — the chat-conversation data container.

Reference: pytorch/rl torchrl/data/llm/history.py:465 (History
TensorClass), :374 (ContentBase).  A batched container of chat messages
(role + content) with template rendering and tokenization hooks.
"""
from __future__ import annotations

from typing import Any, List, Optional, Sequence, Union

import torch

from ...tensordict import NonTensorData, TensorDict, TensorDictBase

__all__ = ["History", "ContentBase"]

_DEFAULT_TEMPLATE = "{role}: {content}\n"


class ContentBase:
    """Structured message content (reference history.py:374) — text plus
    optional non-text payloads."""

    def __init__(self, type: str = "text", text: str = "", **extra):
        self.type = type
        self.text = text
        self.extra = extra

    def __repr__(self):
        return f"ContentBase(type={self.type!r}, text={self.text!r})"


class History:
    """A (batched) list of chat turns.

    ``roles`` and ``contents`` are parallel python lists (batch) of lists
    (turns).  Single-conversation histories have batch_size == ().
    """

    def __init__(
        self,
        role: Union[str, Sequence] = None,
        content: Union[str, Sequence] = None,
        batch_size=(),
    ):
        self.batch_size = tuple(batch_size)
        if self.batch_size:
            self.roles: List[List[str]] = [list(r) for r in (role or [[]] * self.batch_size[0])]
            self.contents: List[List[str]] = [
                list(c) for c in (content or [[]] * self.batch_size[0])
            ]
        else:
            if role is None:
                self.roles = [[]]
                self.contents = [[]]
            elif isinstance(role, str):
                self.roles = [[role]]
                self.contents = [[content]]
            else:
                self.roles = [list(role)]
                self.contents = [list(content)]

    # -- construction ------------------------------------------------------ #
    @classmethod
    def from_chats(cls, chats: Sequence[Sequence[dict]]) -> "History":
        """[[{"role": ..., "content": ...}, ...], ...] → batched History."""
        roles = [[m["role"] for m in chat] for chat in chats]
        contents = [[m["content"] for m in chat] for chat in chats]
        h = cls(batch_size=(len(chats),))
        h.roles = roles
        h.contents = contents
        return h

    @classmethod
    def from_text(cls, text: Union[str, Sequence[str]], role: str = "user") -> "History":
        if isinstance(text, str):
            return cls(role=role, content=text)
        h = cls(batch_size=(len(text),))
        h.roles = [[role] for _ in text]
        h.contents = [[t] for t in text]
        return h

    # -- mutation ---------------------------------------------------------- #
    def append(self, role: str, content: str, index: Optional[int] = None) -> "History":
        if index is None:
            for i in range(len(self.roles)):
                self.roles[i].append(role)
                self.contents[i].append(content)
        else:
            self.roles[index].append(role)
            self.contents[index].append(content)
        return self

    def extend(self, other: "History") -> "History":
        for i in range(len(self.roles)):
            self.roles[i].extend(other.roles[i if len(other.roles) > 1 else 0])
            self.contents[i].extend(other.contents[i if len(other.contents) > 1 else 0])
        return self

    # -- views ------------------------------------------------------------- #
    def __len__(self):
        return self.batch_size[0] if self.batch_size else len(self.roles[0])

    def __getitem__(self, i):
        if self.batch_size:
            h = History(batch_size=())
            h.roles = [list(self.roles[i])]
            h.contents = [list(self.contents[i])]
            return h
        h = History(batch_size=())
        h.roles = [[self.roles[0][i]]]
        h.contents = [[self.contents[0][i]]]
        return h

    @property
    def last_role(self) -> str:
        return self.roles[0][-1] if self.roles[0] else ""

    @property
    def last_content(self) -> str:
        return self.contents[0][-1] if self.contents[0] else ""

    def apply_chat_template(
        self,
        tokenizer=None,
        template: str = _DEFAULT_TEMPLATE,
        add_generation_prompt: bool = False,
        chat_template_name: Optional[str] = None,
        **kwargs,
    ) -> Union[str, List[str]]:
        """Render to text — uses the tokenizer's template when available."""
        outs = []
        for roles, contents in zip(self.roles, self.contents):
            msgs = [
                {"role": r, "content": c} for r, c in zip(roles, contents)
            ]
            if tokenizer is not None and hasattr(tokenizer, "apply_chat_template"):
                try:
                    outs.append(
                        tokenizer.apply_chat_template(
                            msgs,
                            tokenize=False,
                            add_generation_prompt=add_generation_prompt,
                        )
                    )
                    continue
                except Exception:
                    pass
            text = "".join(template.format(role=r, content=c) for r, c in zip(roles, contents))
            if add_generation_prompt:
                text += "assistant: "
            outs.append(text)
        if not self.batch_size:
            return outs[0]
        return outs

    def to_tensordict(self) -> TensorDictBase:
        return TensorDict(
            {"history": NonTensorData({"roles": self.roles, "contents": self.contents})},
            batch_size=self.batch_size,
        )

    @classmethod
    def from_tensordict(cls, td: TensorDictBase) -> "History":
        data = td.get_non_tensor("history")
        h = cls(batch_size=td.batch_size)
        h.roles = data["roles"]
        h.contents = data["contents"]
        return h

    def clone(self) -> "History":
        h = History(batch_size=self.batch_size)
        h.roles = [list(r) for r in self.roles]
        h.contents = [list(c) for c in self.contents]
        return h

    def __repr__(self):
        lines = []
        for roles, contents in zip(self.roles[:2], self.contents[:2]):
            for r, c in zip(roles, contents):
                lines.append(f"  [{r}] {c[:60]}")
        return f"History(batch_size={self.batch_size},\n" + "\n".join(lines) + ")"
