from .collector import LLMCollector
