"""GPT layer specs (reference gpt_layer_specs.py:173).

``get_gpt_layer_local_spec``: the fused-kernel path — ColumnParallel QKV,
core attention (flash MFMA kernel by default, baddbmm+fused-softmax when
``use_flash=False``), fused norms + bias-activation.  There is no
"transformer_engine" variant on MI355X — the HIP kernels *are* the fast
path.
"""

from __future__ import annotations

from ...fusions.fused_layer_norm import FusedLayerNorm, FusedRMSNorm
from ...tensor_parallel.layers import ColumnParallelLinear, RowParallelLinear
from ...transformer.attention import SelfAttention, SelfAttentionSubmodules
from ...transformer.dot_product_attention import DotProductAttention, FlashAttention
from ...transformer.mlp import MLP, MLPSubmodules
from ...transformer.spec_utils import ModuleSpec
from ...transformer.transformer_layer import (
    TransformerLayer,
    TransformerLayerSubmodules,
)
from ...enums import AttnMaskType


def _norm_spec(normalization: str):
    return FusedRMSNorm if normalization == "RMSNorm" else FusedLayerNorm


def get_gpt_layer_local_spec(num_experts: int = None, moe_grouped_gemm: bool = False,
                             qk_layernorm: bool = False, *, normalization: str = "LayerNorm",
                             use_flash: bool = True,
                             multi_latent_attention: bool = False,
                             attn_mask_type: AttnMaskType = AttnMaskType.causal
                             ) -> ModuleSpec:
    norm = _norm_spec(normalization)
    core_attn = FlashAttention if use_flash else DotProductAttention
    if multi_latent_attention:
        # MLA's concatenated qk dim (192) never matches the flash kernel's
        # supported head sizes; the fused baddbmm+softmax path handles the
        # asymmetric v_head_dim
        core_attn = DotProductAttention
    if num_experts is None:
        mlp = ModuleSpec(module=MLP, submodules=MLPSubmodules(
            linear_fc1=ColumnParallelLinear, linear_fc2=RowParallelLinear))
    else:
        from ...transformer.moe.moe_layer import MoELayer, MoESubmodules
        from ...transformer.moe.experts import GroupedMLP, SequentialMLP
        from ...transformer.moe.shared_experts import SharedExpertMLP
        experts = GroupedMLP if moe_grouped_gemm else SequentialMLP
        mlp = ModuleSpec(module=MoELayer, submodules=MoESubmodules(
            experts=experts,
            shared_experts=ModuleSpec(module=SharedExpertMLP, params={"gate": False},
                                      submodules=MLPSubmodules(
                                          linear_fc1=ColumnParallelLinear,
                                          linear_fc2=RowParallelLinear))))
    if multi_latent_attention:
        from ...transformer.multi_latent_attention import (
            MLASelfAttention, MLASelfAttentionSubmodules)
        attn_spec = ModuleSpec(
            module=MLASelfAttention,
            params={"attn_mask_type": attn_mask_type},
            submodules=MLASelfAttentionSubmodules(
                linear_q_proj=ColumnParallelLinear,
                linear_q_up_proj=ColumnParallelLinear,
                linear_kv_up_proj=ColumnParallelLinear,
                core_attention=core_attn,
                linear_proj=RowParallelLinear,
                q_layernorm=norm,
                kv_layernorm=norm,
            ),
        )
    else:
        attn_spec = ModuleSpec(
            module=SelfAttention,
            params={"attn_mask_type": attn_mask_type},
            submodules=SelfAttentionSubmodules(
                linear_qkv=ColumnParallelLinear,
                core_attention=core_attn,
                linear_proj=RowParallelLinear,
                q_layernorm=norm if qk_layernorm else None,
                k_layernorm=norm if qk_layernorm else None,
            ),
        )
    return ModuleSpec(
        module=TransformerLayer,
        submodules=TransformerLayerSubmodules(
            input_layernorm=norm,
            self_attention=attn_spec,
            self_attn_bda=None,
            pre_mlp_layernorm=norm,
            mlp=mlp,
            mlp_bda=None,
        ),
    )


# Alias for reference-API compatibility: on MI355X there is no TE — the
# "fast spec" is the same local HIP-fused spec.
def get_gpt_layer_with_transformer_engine_spec(*args, **kwargs) -> ModuleSpec:
    return get_gpt_layer_local_spec(*args, **kwargs)


def get_gpt_decoder_block_spec(config, use_transformer_engine: bool = False,
                               normalization: str = None, qk_layernorm: bool = False):
    """Heterogeneous stacks (dense/MoE pattern); reference
    gpt_layer_specs.py get_gpt_decoder_block_spec."""
    from ...transformer.transformer_block import TransformerBlockSubmodules
    from ...fusions.fused_layer_norm import get_norm_cls

    normalization = normalization or config.normalization
    num_layers = config.num_layers // config.pipeline_model_parallel_size
    if config.virtual_pipeline_model_parallel_size:
        num_layers //= config.virtual_pipeline_model_parallel_size
    specs = []
    for i in range(num_layers):
        is_moe = (config.num_moe_experts is not None and
                  (i % config.moe_layer_freq == config.moe_layer_freq - 1
                   if config.moe_layer_freq > 1 else True))
        specs.append(get_gpt_layer_local_spec(
            num_experts=config.num_moe_experts if is_moe else None,
            moe_grouped_gemm=config.moe_grouped_gemm,
            qk_layernorm=qk_layernorm, normalization=normalization))
    return TransformerBlockSubmodules(
        layer_specs=specs, layer_norm=get_norm_cls(normalization))
