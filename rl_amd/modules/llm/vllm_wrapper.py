"""vLLM / SGLang-backed LLM policy wrappers (import-gated — neither
engine ships in this image, so construction raises ImportError exactly
like the reference does when the dependency is absent).

Reference: pytorch/rl torchrl/modules/llm/policies/vllm_wrapper.py:88
(vLLMWrapper) and sglang_wrapper.py:53.  The bodies below are complete
implementations against the engines' public offline APIs
(``vllm.LLM``/``SamplingParams``; ``sglang.Engine``) sharing the
TensorDict schema of :class:`TransformersWrapper` (history/text in;
text_response, tokens_response, log_probs, updated history out), so the
losses and collectors see one interface regardless of backend.  On an
image with vLLM-ROCm these run continuous batching + paged KV cache in
the 288 GB HBM3E; weight publication mid-training uses
``rl_amd.weight_update.LLMCollectiveWeightSyncScheme`` (packed RCCL
broadcast) via :meth:`update_weights`.
"""
from __future__ import annotations

import importlib.util
from typing import List, Optional

import torch

from ...data.llm.history import History
from ...tensordict import TensorDictBase
from .transformers_wrapper import LLMWrapperBase

__all__ = ["vLLMWrapper", "SGLangWrapper"]

_has_vllm = importlib.util.find_spec("vllm") is not None
_has_sglang = importlib.util.find_spec("sglang") is not None


class _EngineWrapperBase(LLMWrapperBase):
    """Shared TensorDict plumbing for engine-backed wrappers."""

    def __init__(self, *, tokenizer, generate, max_new_tokens, temperature,
                 input_mode):
        super().__init__()
        self.tokenizer = tokenizer
        self.generate = generate
        self.max_new_tokens = max_new_tokens
        self.temperature = temperature
        self.input_mode = input_mode
        self.in_keys = ["history" if input_mode == "history" else "text"]
        self.out_keys = (
            ["text_response", "tokens_response", "log_probs", "history"]
            if generate
            else ["log_probs"]
        )

    def _texts_from(self, td: TensorDictBase) -> List[str]:
        if self.input_mode == "history":
            hist = td.get_non_tensor("history")
            if isinstance(hist, dict):
                h = History(batch_size=td.batch_size or (1,))
                h.roles, h.contents = hist["roles"], hist["contents"]
                hist = h
            out = hist.apply_chat_template(self.tokenizer, add_generation_prompt=True)
            return out if isinstance(out, list) else [out]
        text = td.get_non_tensor("text")
        return text if isinstance(text, list) else [text]

    def _write_outputs(self, td, resp_texts, resp_tokens, log_probs):
        td.set_non_tensor(
            "text_response", resp_texts if len(td.batch_size) else resp_texts[0]
        )
        if not td.batch_size:
            resp_tokens = resp_tokens.squeeze(0)
            log_probs = log_probs.squeeze(0)
        td.set("tokens_response", resp_tokens)
        td.set("log_probs", log_probs)
        if self.input_mode == "history":
            hist_data = td.get_non_tensor("history")
            h = History(batch_size=td.batch_size or ())
            h.roles = [list(r) for r in hist_data["roles"]]
            h.contents = [list(c) for c in hist_data["contents"]]
            for i, resp in enumerate(resp_texts):
                h.append("assistant", resp, index=i if len(h.roles) > 1 else None)
                if len(h.roles) == 1:
                    break
            td.set_non_tensor("history", {"roles": h.roles, "contents": h.contents})
        return td


class vLLMWrapper(_EngineWrapperBase):
    """Generation/log-prob wrapper over an offline ``vllm.LLM`` engine
    (reference vllm_wrapper.py:88).  Pass a model name/path or an
    existing ``vllm.LLM``; ``tensor_parallel_size`` shards the engine
    over xGMI GPUs."""

    def __init__(
        self,
        model,
        *,
        tokenizer=None,
        generate: bool = True,
        max_new_tokens: int = 32,
        temperature: float = 1.0,
        input_mode: str = "history",
        tensor_parallel_size: int = 1,
        dtype: str = "bfloat16",
        **engine_kwargs,
    ):
        if not _has_vllm:
            raise ImportError(
                "vLLMWrapper requires the `vllm` package (vLLM-ROCm), which is "
                "not installed in this image. Use TransformersWrapper instead."
            )
        import vllm

        if isinstance(model, str):
            self_engine = vllm.LLM(
                model=model,
                tensor_parallel_size=tensor_parallel_size,
                dtype=dtype,
                **engine_kwargs,
            )
        else:
            self_engine = model
        if tokenizer is None:
            tokenizer = self_engine.get_tokenizer()
        super().__init__(
            tokenizer=tokenizer,
            generate=generate,
            max_new_tokens=max_new_tokens,
            temperature=temperature,
            input_mode=input_mode,
        )
        self.engine = self_engine

    def update_weights(self, weights_iter) -> None:
        """Push (name, tensor) pairs into the engine's model runner —
        the sink for LLMCollectiveWeightSyncScheme.receive."""
        runner = (
            self.engine.llm_engine.model_executor.driver_worker.model_runner
        )
        runner.model.load_weights(weights_iter)

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        import vllm

        texts = self._texts_from(td)
        if self.generate:
            sp = vllm.SamplingParams(
                max_tokens=self.max_new_tokens,
                temperature=self.temperature,
                logprobs=0,
            )
            outs = self.engine.generate(texts, sp, use_tqdm=False)
            resp_texts = [o.outputs[0].text for o in outs]
            tok_lists = [list(o.outputs[0].token_ids) for o in outs]
            width = max(len(t) for t in tok_lists)
            pad = self.tokenizer.pad_token_id or 0
            resp_tokens = torch.full((len(tok_lists), width), pad, dtype=torch.long)
            log_probs = torch.zeros(len(tok_lists), width)
            for i, o in enumerate(outs):
                toks = tok_lists[i]
                resp_tokens[i, : len(toks)] = torch.tensor(toks)
                lps = o.outputs[0].logprobs or []
                for t, (tok, lp_dict) in enumerate(zip(toks, lps)):
                    entry = lp_dict.get(tok)
                    if entry is not None:
                        log_probs[i, t] = float(getattr(entry, "logprob", entry))
            return self._write_outputs(td, resp_texts, resp_tokens, log_probs)
        # log-prob mode: score the stored response under current weights
        resp = td.get("tokens_response")
        if resp.dim() == 1:
            resp = resp.unsqueeze(0)
        sp = vllm.SamplingParams(max_tokens=1, prompt_logprobs=0, temperature=0)
        prompts = []
        for i, text in enumerate(texts):
            ids = self.tokenizer(text)["input_ids"]
            prompts.append(
                vllm.TokensPrompt(prompt_token_ids=ids + resp[i].tolist())
            )
        outs = self.engine.generate(prompts, sp, use_tqdm=False)
        T = resp.shape[1]
        lp = torch.zeros(resp.shape[0], T)
        for i, o in enumerate(outs):
            plps = o.prompt_logprobs or []
            tail = plps[-T:]
            for t, lp_dict in enumerate(tail):
                if lp_dict:
                    tok = int(resp[i, t])
                    entry = lp_dict.get(tok)
                    if entry is not None:
                        lp[i, t] = float(getattr(entry, "logprob", entry))
        td.set("log_probs", lp)
        return td


class SGLangWrapper(_EngineWrapperBase):
    """SGLang-backed LLM policy wrapper (reference sglang_wrapper.py:53)
    over an offline ``sglang.Engine``."""

    def __init__(
        self,
        model,
        *,
        tokenizer=None,
        generate: bool = True,
        max_new_tokens: int = 32,
        temperature: float = 1.0,
        input_mode: str = "history",
        **engine_kwargs,
    ):
        if not _has_sglang:
            raise ImportError(
                "SGLangWrapper requires the `sglang` package, which is not "
                "installed in this image. Use TransformersWrapper instead."
            )
        import sglang

        engine = (
            sglang.Engine(model_path=model, **engine_kwargs)
            if isinstance(model, str)
            else model
        )
        if tokenizer is None:
            from transformers import AutoTokenizer

            tokenizer = AutoTokenizer.from_pretrained(model)
        super().__init__(
            tokenizer=tokenizer,
            generate=generate,
            max_new_tokens=max_new_tokens,
            temperature=temperature,
            input_mode=input_mode,
        )
        self.engine = engine

    def update_weights(self, named_tensors) -> None:
        self.engine.update_weights_from_tensor(list(named_tensors))

    def forward(self, td: TensorDictBase) -> TensorDictBase:
        texts = self._texts_from(td)
        params = {
            "max_new_tokens": self.max_new_tokens,
            "temperature": self.temperature,
        }
        outs = self.engine.generate(texts, params, return_logprob=True)
        if isinstance(outs, dict):
            outs = [outs]
        resp_texts = [o["text"] for o in outs]
        tok_lp = [
            o.get("meta_info", {}).get("output_token_logprobs", []) for o in outs
        ]
        width = max((len(t) for t in tok_lp), default=1) or 1
        pad = getattr(self.tokenizer, "pad_token_id", 0) or 0
        resp_tokens = torch.full((len(outs), width), pad, dtype=torch.long)
        log_probs = torch.zeros(len(outs), width)
        for i, entries in enumerate(tok_lp):
            for t, entry in enumerate(entries):
                lp, tok = entry[0], entry[1]
                resp_tokens[i, t] = int(tok)
                log_probs[i, t] = float(lp)
        return self._write_outputs(td, resp_texts, resp_tokens, log_probs)
