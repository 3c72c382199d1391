from .gpt2 import GPT2Config, GPT2Model, gpt2_small, gpt2_medium, gpt2_xl_1p3b
