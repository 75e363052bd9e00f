from .t5_model import T5Model
