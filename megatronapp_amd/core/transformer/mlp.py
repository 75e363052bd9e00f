"""MLP block (reference transformer/mlp.py:371 file).

linear_fc1 (column-parallel, optionally gated 2F) -> fused bias+activation
HIP kernel -> linear_fc2 (row-parallel).  Bias of fc2 is deferred
(skip_bias_add) into the layer's bias-dropout-add epilogue.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Union

import torch

from ..fusions.fused_bias_act import (
    bias_gelu_impl,
    bias_squared_relu_impl,
    bias_swiglu_impl,
)
from ..tensor_tracer import FlagType, get_tensor_tracers
from ..transformer_config import TransformerConfig
from ..trace_hooks import trace_scope
from .module import MegatronModule
from .spec_utils import ModuleSpec, build_module


@dataclass
class MLPSubmodules:
    linear_fc1: Union[ModuleSpec, type] = None
    linear_fc2: Union[ModuleSpec, type] = None


class MLP(MegatronModule):
    def __init__(self, config: TransformerConfig, submodules: MLPSubmodules,
                 is_expert: bool = False, input_size: int = None,
                 ffn_hidden_size: int = None):
        super().__init__(config)
        self.input_size = input_size if input_size is not None else config.hidden_size
        ffn = ffn_hidden_size if ffn_hidden_size is not None else config.ffn_hidden_size
        fc1_out = ffn * 2 if config.gated_linear_unit else ffn

        self.linear_fc1 = build_module(
            submodules.linear_fc1, self.input_size, fc1_out, config=config,
            init_method=config.init_method, bias=config.add_bias_linear,
            skip_bias_add=True, gather_output=False, is_expert=is_expert)
        self.linear_fc2 = build_module(
            submodules.linear_fc2, ffn, config.hidden_size, config=config,
            init_method=config.output_layer_init_method,
            bias=config.add_bias_linear, input_is_parallel=True,
            skip_bias_add=True, is_expert=is_expert)

        if config.gated_linear_unit and config.activation_func == "gelu":
            from ..fusions.fused_bias_act import bias_geglu_impl
            self.activation = bias_geglu_impl     # GeGLU
        elif config.gated_linear_unit:
            self.activation = bias_swiglu_impl
        elif config.activation_func == "squared_relu":
            self.activation = bias_squared_relu_impl
        else:
            self.activation = bias_gelu_impl

    def forward(self, hidden_states, per_token_scale=None):
        with trace_scope("mlp"):
            intermediate, bias = self.linear_fc1(hidden_states)
            tt = get_tensor_tracers()
            layer_number = getattr(self, "layer_number", 0)
            if tt is not None and tt.enabled(FlagType.MLP1, layer_number):
                tt.report(FlagType.MLP1, layer_number, intermediate)
            if self.config.bias_activation_fusion or not self.config.add_bias_linear:
                intermediate = self.activation(intermediate, bias)
            else:
                if bias is not None:
                    intermediate = intermediate + bias
                intermediate = self.activation(intermediate, None)
            output, output_bias = self.linear_fc2(intermediate)
            if tt is not None and tt.enabled(FlagType.MLP2, layer_number):
                tt.report(FlagType.MLP2, layer_number, output)
            if tt is not None and tt.enabled(FlagType.MLP2_Plot, layer_number):
                tt.report(FlagType.MLP2_Plot, layer_number, output)
        return output, output_bias
