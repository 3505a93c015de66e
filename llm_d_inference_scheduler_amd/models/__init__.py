from .configs import LLAMA_3_8B, LLAVA_1_5_7B_TEXT, TINY_LLAMA, ModelConfig  # noqa: F401
