from .engine import LLMEngine, SamplingParams  # noqa: F401
