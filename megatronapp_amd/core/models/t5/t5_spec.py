"""T5 layer specs (reference models/T5/t5_spec.py): padding-mask encoder
layers; causal decoder layers with a cross-attention slot."""

from ...enums import AttnMaskType
from ...fusions.fused_layer_norm import FusedLayerNorm, FusedRMSNorm
from ...tensor_parallel.layers import ColumnParallelLinear, RowParallelLinear
from ...transformer.attention import SelfAttention, SelfAttentionSubmodules
from ...transformer.cross_attention import (
    CrossAttention,
    CrossAttentionSubmodules,
)
from ...transformer.dot_product_attention import DotProductAttention
from ...transformer.mlp import MLP, MLPSubmodules
from ...transformer.spec_utils import ModuleSpec
from ...transformer.transformer_layer import (
    TransformerLayer,
    TransformerLayerSubmodules,
)
from ..gpt.gpt_layer_specs import get_gpt_layer_local_spec


def get_t5_encoder_layer_spec(normalization="LayerNorm"):
    return get_gpt_layer_local_spec(
        normalization=normalization, use_flash=False,
        attn_mask_type=AttnMaskType.padding)


def get_t5_decoder_layer_spec(normalization="LayerNorm"):
    norm = FusedRMSNorm if normalization == "RMSNorm" else FusedLayerNorm
    return ModuleSpec(
        module=TransformerLayer,
        submodules=TransformerLayerSubmodules(
            input_layernorm=norm,
            self_attention=ModuleSpec(
                module=SelfAttention,
                params={"attn_mask_type": AttnMaskType.causal},
                submodules=SelfAttentionSubmodules(
                    linear_qkv=ColumnParallelLinear,
                    core_attention=DotProductAttention,
                    linear_proj=RowParallelLinear)),
            pre_cross_attn_layernorm=norm,
            cross_attention=ModuleSpec(
                module=CrossAttention,
                submodules=CrossAttentionSubmodules(
                    linear_q=ColumnParallelLinear,
                    linear_kv=ColumnParallelLinear,
                    core_attention=DotProductAttention,
                    linear_proj=RowParallelLinear)),
            pre_mlp_layernorm=norm,
            mlp=ModuleSpec(module=MLP, submodules=MLPSubmodules(
                linear_fc1=ColumnParallelLinear,
                linear_fc2=RowParallelLinear)),
        ))
