"""Transformer block: the per-pipeline-stage stack of layers.

Reference: transformer_block.py:220 (layer offsets for PP+VPP via
get_num_layers_to_build:52, full-block recompute, final norm).  MegaScope
system perturbation injects between layers (reference :542-544).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Union

import torch
from torch import nn

from .. import parallel_state
from ..tensor_disturbance import get_disturbance
from ..tensor_parallel.random import checkpoint as tp_checkpoint
from ..transformer_config import TransformerConfig
from ..utils import make_viewless_tensor
from .module import MegatronModule
from .spec_utils import ModuleSpec, build_module
from .transformer_layer import TransformerLayer


def get_num_layers_to_build(config: TransformerConfig, vp_stage=None) -> int:
    pp = config.pipeline_model_parallel_size
    vpp = config.virtual_pipeline_model_parallel_size
    assert config.num_layers % pp == 0, (
        f"num_layers {config.num_layers} not divisible by pp {pp}")
    per_stage = config.num_layers // pp
    if vpp is not None:
        assert per_stage % vpp == 0
        return per_stage // vpp
    return per_stage


def _layer_offset(config: TransformerConfig, vp_stage=None) -> int:
    pp_rank = parallel_state.get_pipeline_model_parallel_rank()
    vpp = config.virtual_pipeline_model_parallel_size
    pp = config.pipeline_model_parallel_size
    if vpp is not None:
        vp = vp_stage if vp_stage is not None else (
            parallel_state.get_virtual_pipeline_model_parallel_rank() or 0)
        layers_per_chunk = config.num_layers // pp // vpp
        return (vp * pp + pp_rank) * layers_per_chunk
    return pp_rank * (config.num_layers // pp)


@dataclass
class TransformerBlockSubmodules:
    layer_specs: List[ModuleSpec] = None
    layer_norm: Optional[Union[ModuleSpec, type]] = None


def _get_block_submodules(config, spec) -> TransformerBlockSubmodules:
    if isinstance(spec, TransformerBlockSubmodules):
        return spec
    if isinstance(spec, ModuleSpec):
        if issubclass(spec.module, TransformerBlock):
            return spec.submodules
        if issubclass(spec.module, TransformerLayer):
            num_layers = get_num_layers_to_build(config)
            from ..fusions.fused_layer_norm import get_norm_cls
            return TransformerBlockSubmodules(
                layer_specs=[spec] * num_layers,
                layer_norm=get_norm_cls(config.normalization))
    raise ValueError(f"invalid block spec: {spec}")


class TransformerBlock(MegatronModule):
    def __init__(self, config: TransformerConfig,
                 spec: Union[TransformerBlockSubmodules, ModuleSpec],
                 post_layer_norm: bool = True, pre_process: bool = True,
                 post_process: bool = True, vp_stage: Optional[int] = None):
        super().__init__(config)
        self.submodules = _get_block_submodules(config, spec)
        self.post_layer_norm = post_layer_norm
        self.pre_process = pre_process
        self.post_process = post_process
        self.vp_stage = vp_stage
        self.input_tensor = None

        offset = _layer_offset(config, vp_stage)
        self.layers = nn.ModuleList([
            build_module(layer_spec, config=config, layer_number=offset + i + 1)
            for i, layer_spec in enumerate(self.submodules.layer_specs)])

        if self.post_process and self.post_layer_norm and \
                self.submodules.layer_norm is not None:
            self.final_layernorm = build_module(
                self.submodules.layer_norm, config=config,
                hidden_size=config.hidden_size, eps=config.layernorm_epsilon)
        else:
            self.final_layernorm = None

    def set_input_tensor(self, input_tensor):
        self.input_tensor = input_tensor

    def _checkpointed_forward(self, hidden_states, attention_mask,
                              rotary_pos_emb):
        """Full-granularity activation recompute (reference :220 area)."""

        def custom(start, end):
            def custom_forward(hs, am, rope):
                for idx in range(start, end):
                    hs, _ = self.layers[idx](hs, attention_mask=am,
                                             rotary_pos_emb=rope)
                return hs
            return custom_forward

        chunk = self.config.recompute_num_layers or 1
        n = len(self.layers)
        i = 0
        while i < n:
            hidden_states = tp_checkpoint(
                custom(i, min(i + chunk, n)),
                self.config.distribute_saved_activations,
                hidden_states, attention_mask, rotary_pos_emb)
            i += chunk
        return hidden_states

    def forward(self, hidden_states, attention_mask=None, context=None,
                context_mask=None, rotary_pos_emb=None, rotary_pos_cos=None,
                rotary_pos_sin=None, attention_bias=None, inference_context=None,
                packed_seq_params=None, sequence_len_offset=None):
        if not self.pre_process and self.input_tensor is not None:
            hidden_states = self.input_tensor

        hidden_states = make_viewless_tensor(
            hidden_states, requires_grad=True, keep_graph=True)

        if (self.config.recompute_granularity == "full" and self.training
                and torch.is_grad_enabled()):
            hidden_states = self._checkpointed_forward(
                hidden_states, attention_mask, rotary_pos_emb)
        else:
            disturbance = get_disturbance()
            for layer in self.layers:
                hidden_states, context = layer(
                    hidden_states, attention_mask=attention_mask,
                    context=context, context_mask=context_mask,
                    rotary_pos_emb=rotary_pos_emb,
                    rotary_pos_cos=rotary_pos_cos,
                    rotary_pos_sin=rotary_pos_sin,
                    attention_bias=attention_bias,
                    inference_context=inference_context,
                    packed_seq_params=packed_seq_params,
                    sequence_len_offset=sequence_len_offset)
                if disturbance.system_perturbation:
                    hidden_states = disturbance.perturb_system(hidden_states)

        if self.final_layernorm is not None:
            hidden_states = self.final_layernorm(hidden_states)
        return hidden_states
