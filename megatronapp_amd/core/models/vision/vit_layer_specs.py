"""ViT layer specs (reference: core/models/vision/vit_layer_specs.py).

The vision tower is bidirectional (``AttnMaskType.no_mask``) so the core
attention is the baddbmm+fused-softmax path — the flash MFMA kernel is
causal-only.  Norms/bias-gelu use the same fused HIP modules as the
language stack.
"""

from __future__ import annotations

from ...enums import AttnMaskType
from ...fusions.fused_layer_norm import FusedLayerNorm
from ...tensor_parallel.layers import ColumnParallelLinear, RowParallelLinear
from ...transformer.attention import SelfAttention, SelfAttentionSubmodules
from ...transformer.dot_product_attention import DotProductAttention
from ...transformer.mlp import MLP, MLPSubmodules
from ...transformer.spec_utils import ModuleSpec
from ...transformer.transformer_layer import (
    TransformerLayer,
    TransformerLayerSubmodules,
)


def get_vit_layer_local_spec() -> ModuleSpec:
    """Non-causal transformer layer for vision towers."""
    return ModuleSpec(
        module=TransformerLayer,
        submodules=TransformerLayerSubmodules(
            input_layernorm=FusedLayerNorm,
            self_attention=ModuleSpec(
                module=SelfAttention,
                params={"attn_mask_type": AttnMaskType.no_mask},
                submodules=SelfAttentionSubmodules(
                    linear_qkv=ColumnParallelLinear,
                    core_attention=DotProductAttention,
                    linear_proj=RowParallelLinear,
                ),
            ),
            pre_mlp_layernorm=FusedLayerNorm,
            mlp=ModuleSpec(
                module=MLP,
                submodules=MLPSubmodules(
                    linear_fc1=ColumnParallelLinear,
                    linear_fc2=RowParallelLinear,
                ),
            ),
        ),
    )
