"""vLLM-backed LLM policy wrapper (gated — vLLM not in this image).

Reference: pytorch/rl torchrl/modules/llm/policies/vllm_wrapper.py:88
(vLLMWrapper) and backends/vllm (AsyncVLLM).  The MI355X inference path
of choice when vLLM-ROCm is installed: continuous batching + paged KV
cache on the 288 GB HBM3E.  In this image vLLM is absent, so the class
raises a clear ImportError at construction; the transformers-backed
:class:`~rl_amd.modules.TransformersWrapper` covers generation/log-prob
duty with the same TensorDict interface.
"""
from __future__ import annotations

import importlib.util

__all__ = ["vLLMWrapper", "SGLangWrapper"]

_has_vllm = importlib.util.find_spec("vllm") is not None


class vLLMWrapper:
    """Generation/log-prob wrapper over a vLLM engine (same TensorDict
    schema as TransformersWrapper: history/text in, tokens_response +
    log_probs + updated history out)."""

    def __init__(self, *args, **kwargs):
        if not _has_vllm:
            raise ImportError(
                "vLLMWrapper requires the `vllm` package (vLLM-ROCm), which is "
                "not installed in this image. Use TransformersWrapper instead."
            )
        raise NotImplementedError(
            "vLLM backend scaffolding: install vllm and implement engine glue"
        )


class SGLangWrapper:
    """SGLang-backed LLM policy wrapper (reference
    policies/sglang_wrapper.py:53) — gated: sglang is not installed in
    this image; TransformersWrapper covers the TensorDict interface."""

    def __init__(self, *args, **kwargs):
        import importlib.util

        if importlib.util.find_spec("sglang") is None:
            raise ImportError(
                "SGLangWrapper requires the `sglang` package, which is not "
                "installed in this image. Use TransformersWrapper instead."
            )
        raise NotImplementedError("sglang backend scaffolding")
