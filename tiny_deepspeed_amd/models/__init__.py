from .gpt2 import GPTConfig, GPT2Model

__all__ = ["GPTConfig", "GPT2Model"]
