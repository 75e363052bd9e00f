from .gpt_model import GPTModel
