from .gpt import GPT, Block, CausalSelfAttention, LayerNorm, MLP, model_getter

__all__ = ["GPT", "Block", "CausalSelfAttention", "LayerNorm", "MLP", "model_getter"]
