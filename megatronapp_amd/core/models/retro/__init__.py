from .config import RetroConfig
from .model import RetroModel
from .specs import (
    get_retro_decoder_block_spec,
    get_retro_decoder_layer_local_spec,
    get_retro_encoder_block_spec,
    get_retro_encoder_layer_local_spec,
)
