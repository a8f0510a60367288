"""LLM policy wrappers over HuggingFace transformers.

Reference: pytorch/rl torchrl/modules/llm/policies/common.py:783
(LLMWrapperBase) and transformers_wrapper.py — generate / log-prob modes
over a causal LM.  vLLM/SGLang engine wrappers are out of image scope
(no engines installed); the transformers backend is the reference-parity
path and runs fine on ROCm.
"""
from __future__ import annotations

from typing import Any, List, Optional, Sequence, Union

import torch

from ...data.llm.history import History
from ...tensordict import NonTensorData, TensorDict, TensorDictBase, TensorDictModuleBase

__all__ = ["LLMWrapperBase", "TransformersWrapper"]


class LLMWrapperBase(TensorDictModuleBase):
    """ABC: generate(td) and log_probs(td) over conversation data
    (reference policies/common.py:783)."""

    generate: bool

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        raise NotImplementedError


class TransformersWrapper(LLMWrapperBase):
    """Wrap an HF causal LM + tokenizer.

    * ``generate=True``: reads ``history`` (History) or ``text``, samples a
      completion, writes ``text_response``, ``tokens_response``,
      ``log_probs`` and the updated ``history``.
    * ``generate=False``: computes per-token log-probs of the stored
      response under the current weights (the PPO/GRPO ratio path).
    """

    def __init__(
        self,
        model,
        *,
        tokenizer=None,
        generate: bool = True,
        max_new_tokens: int = 32,
        temperature: float = 1.0,
        do_sample: bool = True,
        device=None,
        input_mode: str = "history",
        pad_token_id: Optional[int] = None,
    ):
        super().__init__()
        self.model = model
        # RLHF-style losses need deterministic log-probs: dropout off.
        # (Re-enable with wrapper.model.train() if you want it.)
        self.model.eval()
        self.tokenizer = tokenizer
        self.generate = generate
        self.max_new_tokens = max_new_tokens
        self.temperature = temperature
        self.do_sample = do_sample
        self.device = torch.device(device) if device is not None else None
        self.input_mode = input_mode
        if pad_token_id is None and tokenizer is not None:
            pad_token_id = getattr(tokenizer, "pad_token_id", None) or getattr(
                tokenizer, "eos_token_id", 0
            )
        self.pad_token_id = pad_token_id if pad_token_id is not None else 0
        self.in_keys = ["history" if input_mode == "history" else "text"]
        self.out_keys = (
            ["text_response", "tokens_response", "log_probs", "history"]
            if generate
            else ["log_probs"]
        )

    # ------------------------------------------------------------------ #
    def _encode(self, texts: List[str]):
        enc = self.tokenizer(texts, return_tensors="pt", padding=True)
        ids = enc["input_ids"]
        mask = enc["attention_mask"]
        if self.device is not None:
            ids, mask = ids.to(self.device), mask.to(self.device)
        return ids, mask

    def _texts_from(self, td: TensorDictBase) -> List[str]:
        if self.input_mode == "history":
            hist = td.get_non_tensor("history")
            if isinstance(hist, dict):
                h = History(batch_size=td.batch_size or (1,))
                h.roles, h.contents = hist["roles"], hist["contents"]
                hist = h
            # always render with the generation prompt: in log-prob mode the
            # history holds the PROMPT only, and the stored response tokens
            # were generated conditioned on prompt + generation suffix
            out = hist.apply_chat_template(self.tokenizer, add_generation_prompt=True)
            return out if isinstance(out, list) else [out]
        text = td.get_non_tensor("text")
        return text if isinstance(text, list) else [text]

    @torch.no_grad()
    def _generate(self, td: TensorDictBase) -> TensorDictBase:
        texts = self._texts_from(td)
        ids, mask = self._encode(texts)
        prompt_len = ids.shape[1]
        out = self.model.generate(
            input_ids=ids,
            attention_mask=mask,
            max_new_tokens=self.max_new_tokens,
            do_sample=self.do_sample,
            temperature=self.temperature,
            pad_token_id=self.pad_token_id,
            return_dict_in_generate=True,
            output_scores=True,
        )
        seq = out.sequences
        resp_tokens = seq[:, prompt_len:]
        # per-step log-probs of the sampled tokens
        lps = []
        for t, scores in enumerate(out.scores):
            logp = scores.log_softmax(-1)
            tok = resp_tokens[:, t]
            lps.append(logp.gather(-1, tok.unsqueeze(-1)).squeeze(-1))
        log_probs = torch.stack(lps, 1) if lps else torch.zeros_like(resp_tokens, dtype=torch.float)
        resp_texts = self.tokenizer.batch_decode(resp_tokens, skip_special_tokens=True)
        td.set_non_tensor("text_response", resp_texts if len(td.batch_size) else resp_texts[0])
        if not td.batch_size:
            # scalar-batch dialog: drop the singleton generation batch dim
            resp_tokens = resp_tokens.squeeze(0)
            log_probs = log_probs.squeeze(0)
        td.set("tokens_response", resp_tokens.cpu() if td.device is None else resp_tokens)
        td.set("log_probs", log_probs.cpu() if td.device is None else log_probs)
        # update history
        if self.input_mode == "history":
            hist_data = td.get_non_tensor("history")
            h = History(batch_size=td.batch_size or ())
            h.roles, h.contents = (
                [list(r) for r in hist_data["roles"]],
                [list(c) for c in hist_data["contents"]],
            )
            for i, resp in enumerate(resp_texts):
                h.append("assistant", resp, index=i if len(h.roles) > 1 else None)
                if len(h.roles) == 1:
                    break
            td.set_non_tensor(
                "history", {"roles": h.roles, "contents": h.contents}
            )
        return td

    def _log_probs(self, td: TensorDictBase) -> TensorDictBase:
        """Log-probs of the stored response tokens under current weights."""
        texts = self._texts_from(td)
        ids, mask = self._encode(texts)
        resp = td.get("tokens_response")
        if self.device is not None:
            resp = resp.to(self.device)
        if resp.dim() == 1:  # scalar-batch dialog
            resp = resp.unsqueeze(0)
        full = torch.cat([ids, resp], 1)
        full_mask = torch.cat([mask, torch.ones_like(resp)], 1)
        logits = self.model(input_ids=full, attention_mask=full_mask).logits
        # logits at position t predict token t+1
        resp_logits = logits[:, ids.shape[1] - 1 : -1]
        logp = resp_logits.log_softmax(-1)
        lp = logp.gather(-1, resp.unsqueeze(-1)).squeeze(-1)
        td.set("log_probs", lp if td.device is None else lp.to(td.device or lp.device))
        return td

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        if self.generate:
            return self._generate(td)
        return self._log_probs(td)
