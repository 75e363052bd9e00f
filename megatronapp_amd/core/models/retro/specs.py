"""Retro decoder/encoder block specs (reference
core/models/retro/{decoder_spec,encoder_spec}.py).

Layout parity: retro decoder layers sit every 3 layers starting at
layer 6 (≤15 layers) or 9, the first one instantiating the neighbor
encoder; the encoder block is ``retro_encoder_num_layers`` deep with its
retro layer first.  Unlike the reference there is no TE variant — the
local HIP-fused modules are the only path.  Small test models
(num_layers < 6) place the single retro layer mid-stack.
"""

from __future__ import annotations

from ... import parallel_state
from ...enums import AttnMaskType
from ...fusions.fused_layer_norm import FusedLayerNorm
from ...tensor_parallel.layers import ColumnParallelLinear, RowParallelLinear
from ...transformer.cross_attention import CrossAttentionSubmodules
from ...transformer.dot_product_attention import DotProductAttention
from ...transformer.spec_utils import ModuleSpec
from ...transformer.transformer_block import (
    TransformerBlockSubmodules,
    get_num_layers_to_build,
)
from ..gpt.gpt_layer_specs import get_gpt_layer_local_spec
from .attention import (
    RetroDecoderBiasDropoutAdd,
    RetroDecoderCrossAttention,
    RetroEncoderBiasDropoutAdd,
    RetroEncoderCrossAttention,
    RetroEncoderLayerNorm,
)
from .config import RetroConfig

_CROSS_SUBMODULES = CrossAttentionSubmodules(
    linear_q=ColumnParallelLinear,
    linear_kv=ColumnParallelLinear,
    core_attention=DotProductAttention,
    linear_proj=RowParallelLinear,
)


def get_retro_decoder_layer_local_spec(
        encoder_block_spec=None) -> ModuleSpec:
    spec = get_gpt_layer_local_spec(use_flash=False)
    spec.submodules.pre_cross_attn_layernorm = FusedLayerNorm
    spec.submodules.cross_attention = ModuleSpec(
        module=RetroDecoderCrossAttention,
        params={"encoder_block_spec": encoder_block_spec},
        submodules=_CROSS_SUBMODULES)
    spec.submodules.cross_attn_bda = ModuleSpec(
        module=RetroDecoderBiasDropoutAdd)
    return spec


def get_retro_encoder_layer_local_spec() -> ModuleSpec:
    spec = get_gpt_layer_local_spec(
        use_flash=False, attn_mask_type=AttnMaskType.padding)
    spec.submodules.pre_cross_attn_layernorm = FusedLayerNorm
    spec.submodules.cross_attention = ModuleSpec(
        module=RetroEncoderCrossAttention,
        params={"attn_mask_type": AttnMaskType.padding},
        submodules=_CROSS_SUBMODULES)
    spec.submodules.cross_attn_bda = ModuleSpec(
        module=RetroEncoderBiasDropoutAdd)
    spec.submodules.pre_mlp_layernorm = ModuleSpec(
        module=RetroEncoderLayerNorm, submodules=FusedLayerNorm)
    return spec


def get_retro_encoder_block_spec(
        config: RetroConfig) -> TransformerBlockSubmodules:
    """Encoder: retro layer first, plain bidirectional GPT layers after
    (reference encoder_spec.py:119-168)."""
    num_layers = config.retro_encoder_num_layers
    gpt_spec = get_gpt_layer_local_spec(
        use_flash=False, attn_mask_type=AttnMaskType.padding)
    for spec in (gpt_spec,):
        spec.params["hidden_dropout"] = config.retro_encoder_hidden_dropout
    retro_spec = get_retro_encoder_layer_local_spec()
    retro_spec.params["hidden_dropout"] = config.retro_encoder_hidden_dropout
    layer_specs = [retro_spec] + [gpt_spec] * (num_layers - 1)
    # reference encoder block has no final norm (post_process=False when
    # instantiated inside the first retro decoder layer)
    return TransformerBlockSubmodules(layer_specs=layer_specs)


def get_retro_decoder_block_spec(
        config: RetroConfig) -> TransformerBlockSubmodules:
    """Decoder: GPT layers with retro layers interleaved every 3 from
    layer 6/9 (reference decoder_spec.py:123-185)."""
    assert parallel_state.get_pipeline_model_parallel_world_size() == 1, \
        "retro does not support pipeline parallelism"
    num_layers = get_num_layers_to_build(config)
    retro_layer_start = 6 if num_layers <= 15 else 9
    retro_layer_numbers = list(
        range(retro_layer_start, num_layers + 1, 3)) or \
        [max(1, (num_layers + 1) // 2)]    # small test models

    gpt_spec = get_gpt_layer_local_spec(use_flash=False)
    retro_spec = get_retro_decoder_layer_local_spec()
    retro_spec_with_retriever = get_retro_decoder_layer_local_spec(
        get_retro_encoder_block_spec(config))

    layer_specs = []
    for n in range(1, num_layers + 1):
        if n == retro_layer_numbers[0]:
            layer_specs.append(retro_spec_with_retriever)
        elif n in retro_layer_numbers:
            layer_specs.append(retro_spec)
        else:
            layer_specs.append(gpt_spec)
    return TransformerBlockSubmodules(layer_specs=layer_specs,
                                      layer_norm=FusedLayerNorm)
