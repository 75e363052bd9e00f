"""Cross-attention (reference transformer/attention.py CrossAttention):
queries from the decoder stream, keys/values projected from the encoder
output, same core-attention backends as self-attention."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Union

import torch

from .. import parallel_state
from ..enums import AttnMaskType
from ..tensor_parallel.layers import ColumnParallelLinear, RowParallelLinear
from ..tensor_parallel.utils import divide
from ..transformer_config import TransformerConfig
from .module import MegatronModule
from .spec_utils import ModuleSpec, build_module


@dataclass
class CrossAttentionSubmodules:
    linear_q: Union[ModuleSpec, type] = None
    linear_kv: Union[ModuleSpec, type] = None
    core_attention: Union[ModuleSpec, type] = None
    linear_proj: Union[ModuleSpec, type] = None


class CrossAttention(MegatronModule):
    def __init__(self, config: TransformerConfig,
                 submodules: CrossAttentionSubmodules, layer_number: int,
                 attn_mask_type=AttnMaskType.padding, cp_comm_type=None):
        super().__init__(config)
        self.layer_number = layer_number
        self.attn_mask_type = attn_mask_type
        world = parallel_state.get_tensor_model_parallel_world_size()
        self.hn = config.kv_channels
        self.np_ = divide(config.num_attention_heads, world)
        proj = config.kv_channels * config.num_attention_heads

        self.linear_q = build_module(
            submodules.linear_q, config.hidden_size, proj, config=config,
            init_method=config.init_method, bias=config.add_bias_linear,
            skip_bias_add=False, gather_output=False)
        self.linear_kv = build_module(
            submodules.linear_kv, config.hidden_size, 2 * proj, config=config,
            init_method=config.init_method, bias=config.add_bias_linear,
            skip_bias_add=False, gather_output=False)
        self.core_attention = build_module(
            submodules.core_attention, config=config,
            layer_number=layer_number, attn_mask_type=attn_mask_type,
            attention_type="cross")
        self.linear_proj = build_module(
            submodules.linear_proj, proj, config.hidden_size, config=config,
            init_method=config.output_layer_init_method,
            bias=config.add_bias_linear, input_is_parallel=True,
            skip_bias_add=True)

    def forward(self, hidden_states, attention_mask=None,
                key_value_states=None, **kwargs):
        sq, b, _ = hidden_states.shape
        sk = key_value_states.shape[0]
        q, _ = self.linear_q(hidden_states)
        kv, _ = self.linear_kv(key_value_states)
        query = q.view(sq, b, self.np_, self.hn)
        kv = kv.view(sk, b, self.np_, 2 * self.hn)
        key, value = torch.split(kv, [self.hn, self.hn], dim=3)
        context = self.core_attention(
            query.contiguous(), key.contiguous(), value.contiguous(),
            attention_mask=attention_mask,
            attn_mask_type=self.attn_mask_type)
        output, bias = self.linear_proj(context)
        return output, bias
