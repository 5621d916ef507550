from .llama import LlamaConfig, LlamaForCausalLM, MODEL_PRESETS, get_config  # noqa: F401
