from .config import ModelArchConfig
from .llama import LlamaForCausalLM

__all__ = ["ModelArchConfig", "LlamaForCausalLM"]
