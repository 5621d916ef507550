from .kv_cache import PagedKVCache  # noqa: F401
from .engine import LLMEngine, SamplingParams, StepOutput  # noqa: F401
