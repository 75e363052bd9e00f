"""One transformer layer (reference transformer_layer.py, 821 LoC).

Structure: input_norm -> self-attention -> bias-dropout-add residual ->
pre-mlp norm -> MLP (or MoE) -> bias-dropout-add residual.  MegaScan
scopes wrap attention/MLP; MegaScope disturbance injects at the MLP
output (reference :563-565).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Union

import torch

from ..enums import AttnMaskType
from ..fusions.fused_bias_dropout import get_bias_dropout_add
from ..tensor_disturbance import get_disturbance
from ..transformer_config import TransformerConfig
from ..trace_hooks import trace_scope
from .module import MegatronModule
from .spec_utils import ModuleSpec, build_module
from ..utils import make_viewless_tensor


@dataclass
class TransformerLayerSubmodules:
    input_layernorm: Union[ModuleSpec, type] = None
    self_attention: Union[ModuleSpec, type] = None
    self_attn_bda: Union[ModuleSpec, type] = None
    pre_cross_attn_layernorm: Union[ModuleSpec, type] = None
    cross_attention: Union[ModuleSpec, type] = None
    cross_attn_bda: Union[ModuleSpec, type] = None
    pre_mlp_layernorm: Union[ModuleSpec, type] = None
    mlp: Union[ModuleSpec, type] = None
    mlp_bda: Union[ModuleSpec, type] = None
    sharded_state_dict_keys_map: dict = field(default_factory=dict)


class TransformerLayer(MegatronModule):
    def __init__(self, config: TransformerConfig,
                 submodules: TransformerLayerSubmodules,
                 layer_number: int = 1, hidden_dropout: float = None):
        super().__init__(config)
        self.layer_number = layer_number
        self.hidden_dropout = (config.hidden_dropout if hidden_dropout is None
                               else hidden_dropout)

        self.input_layernorm = build_module(
            submodules.input_layernorm, config=config,
            hidden_size=config.hidden_size, eps=config.layernorm_epsilon)
        self.self_attention = build_module(
            submodules.self_attention, config=config, layer_number=layer_number)
        if submodules.cross_attention is not None:
            self.pre_cross_attn_layernorm = build_module(
                submodules.pre_cross_attn_layernorm, config=config,
                hidden_size=config.hidden_size, eps=config.layernorm_epsilon)
            self.cross_attention = build_module(
                submodules.cross_attention, config=config,
                layer_number=layer_number)
            # custom bias-dropout-add operator (e.g. retro's chunked
            # re-permute); None -> the standard fused bda
            self.cross_attn_bda = (
                build_module(submodules.cross_attn_bda, config=config)
                if submodules.cross_attn_bda is not None else None)
        else:
            self.cross_attention = None
            self.cross_attn_bda = None
        self.pre_mlp_layernorm = build_module(
            submodules.pre_mlp_layernorm, config=config,
            hidden_size=config.hidden_size, eps=config.layernorm_epsilon)
        self.mlp = build_module(submodules.mlp, config=config)
        if hasattr(self.mlp, "__class__"):
            setattr(self.mlp, "layer_number", layer_number)
        self.bias_dropout_add_exec_handler = torch.enable_grad

    def _bda(self):
        return get_bias_dropout_add(self.training, self.config.bias_dropout_fusion)

    def forward(self, hidden_states, attention_mask=None, context=None,
                context_mask=None, rotary_pos_emb=None, rotary_pos_cos=None,
                rotary_pos_sin=None, attention_bias=None, inference_context=None,
                packed_seq_params=None, sequence_len_offset=None):
        with trace_scope("transformer_layer"):
            # --- attention block ---
            with trace_scope("_forward_attention"):
                if hasattr(self.input_layernorm, "forward_with_residual"):
                    input_layernorm_output, residual = \
                        self.input_layernorm.forward_with_residual(
                            hidden_states)
                else:
                    residual = hidden_states
                    input_layernorm_output = self.input_layernorm(
                        hidden_states)
                attention_output_with_bias = self.self_attention(
                    input_layernorm_output, attention_mask=attention_mask,
                    inference_context=inference_context,
                    rotary_pos_emb=rotary_pos_emb,
                    rotary_pos_cos=rotary_pos_cos,
                    rotary_pos_sin=rotary_pos_sin,
                    attention_bias=attention_bias,
                    packed_seq_params=packed_seq_params,
                    sequence_len_offset=sequence_len_offset)
                with self.bias_dropout_add_exec_handler():
                    hidden_states = self._bda()(
                        attention_output_with_bias, residual, self.hidden_dropout)

            # --- cross-attention block (encoder-decoder models) ---
            if self.cross_attention is not None and context is not None:
                residual = hidden_states
                normed = self.pre_cross_attn_layernorm(hidden_states)
                cross_out_with_bias = self.cross_attention(
                    normed, attention_mask=context_mask,
                    key_value_states=context)
                if isinstance(cross_out_with_bias, dict) and \
                        "context" in cross_out_with_bias:
                    context = cross_out_with_bias["context"]
                bda = (self.cross_attn_bda(
                           self.training, self.config.bias_dropout_fusion)
                       if self.cross_attn_bda is not None else self._bda())
                with self.bias_dropout_add_exec_handler():
                    hidden_states = bda(
                        cross_out_with_bias, residual, self.hidden_dropout)

            # --- MLP block ---
            with trace_scope("_forward_mlp"):
                if hasattr(self.pre_mlp_layernorm, "forward_with_residual"):
                    pre_mlp_layernorm_output, residual = \
                        self.pre_mlp_layernorm.forward_with_residual(
                            hidden_states)
                else:
                    residual = hidden_states
                    pre_mlp_layernorm_output = self.pre_mlp_layernorm(
                        hidden_states)
                mlp_output_with_bias = self.mlp(pre_mlp_layernorm_output)

                disturbance = get_disturbance()
                if disturbance.calculation_perturbation:
                    out, bias = mlp_output_with_bias
                    mlp_output_with_bias = (
                        disturbance.perturb_calculation(out), bias)

                with self.bias_dropout_add_exec_handler():
                    hidden_states = self._bda()(
                        mlp_output_with_bias, residual, self.hidden_dropout)

        output = make_viewless_tensor(hidden_states, requires_grad=hidden_states.requires_grad,
                                      keep_graph=True)
        return output, context
