from .gpt2 import GPT2Config, GPT2Model, gpt2_small, gpt2_medium, gpt2_xl_1p3b
from .llama import LlamaConfig, LlamaModel, llama3_8b, llama_tiny
from .moe_model import MoEConfig, MoEModel, mixtral_style_8x
