from .llama import LlamaConfig, LlamaModel, llama3_8b, llama_tiny  # noqa: F401
