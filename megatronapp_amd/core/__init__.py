from . import parallel_state
from .enums import ModelType
from .transformer_config import MLATransformerConfig, ModelParallelConfig, TransformerConfig
