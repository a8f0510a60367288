from .transformers_wrapper import LLMWrapperBase, TransformersWrapper
from .vllm_wrapper import SGLangWrapper, vLLMWrapper
