"""Multi-token prediction (reference transformer/multi_token_prediction.py,
DeepSeek-V3 style).

MTP depth k predicts token t+k+1: the depth's input interleaves the main
stream's hidden state for token i with the embedding of token i+k
(enorm/hnorm -> concat -> eh_proj 2h->h), runs one extra transformer
layer, and scores through the SHARED output head.  Each depth's
cross-entropy is scaled by mtp_loss_scaling_factor/num_layers and attached
to the main stream's autograd graph (the reported lm loss stays the main
head's loss; MTP losses are tracked separately for logging).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Union

import torch

from ..transformer_config import TransformerConfig
from .module import MegatronModule
from .spec_utils import ModuleSpec, build_module

_MTP_LOSS_TRACKER = {}


def roll_tensor(tensor, shifts=-1, dims=-1):
    """Shift left along dims, zero-filling the vacated tail."""
    rolled = torch.roll(tensor, shifts=shifts, dims=dims)
    rolled.select(dims, shifts).fill_(0)
    return rolled


class MTPLossAutoScaler(torch.autograd.Function):
    """Attach an auxiliary scalar loss to the main graph (same trick as
    MoEAuxLossAutoScaler): forward passes hidden through; backward emits
    grad 1.0 * scale for the attached loss."""

    main_loss_backward_scale = 1.0

    @staticmethod
    def forward(ctx, output, mtp_loss):
        ctx.save_for_backward(mtp_loss)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        (mtp_loss,) = ctx.saved_tensors
        scale = MTPLossAutoScaler.main_loss_backward_scale
        return grad_output, torch.full_like(mtp_loss, scale)


class MTPLossLoggingHelper:
    @staticmethod
    def save_loss_to_tracker(loss, layer_number, num_layers):
        vals = _MTP_LOSS_TRACKER.setdefault(
            "values", torch.zeros(num_layers, device=loss.device))
        if vals.device != loss.device:
            vals = vals.to(loss.device)
            _MTP_LOSS_TRACKER["values"] = vals
        vals[layer_number] += loss.detach()

    @staticmethod
    def get_and_clear():
        vals = _MTP_LOSS_TRACKER.pop("values", None)
        return vals


@dataclass
class MultiTokenPredictionLayerSubmodules:
    enorm: Union[ModuleSpec, type] = None
    hnorm: Union[ModuleSpec, type] = None
    eh_proj: Union[ModuleSpec, type] = None
    transformer_layer: Union[ModuleSpec, type] = None
    layer_norm: Union[ModuleSpec, type] = None


class MultiTokenPredictionLayer(MegatronModule):
    def __init__(self, config: TransformerConfig,
                 submodules: MultiTokenPredictionLayerSubmodules,
                 layer_number: int = 1):
        super().__init__(config)
        self.layer_number = layer_number
        self.enorm = build_module(submodules.enorm, config=config,
                                  hidden_size=config.hidden_size,
                                  eps=config.layernorm_epsilon)
        self.hnorm = build_module(submodules.hnorm, config=config,
                                  hidden_size=config.hidden_size,
                                  eps=config.layernorm_epsilon)
        # concat(embedding, hidden) -> h; gather_output so the transformer
        # layer sees the full hidden dim under TP
        self.eh_proj = build_module(
            submodules.eh_proj, 2 * config.hidden_size, config.hidden_size,
            config=config, init_method=config.init_method, bias=False,
            skip_bias_add=False, gather_output=True)
        self.transformer_layer = build_module(
            submodules.transformer_layer, config=config)
        self.final_layernorm = build_module(
            submodules.layer_norm, config=config,
            hidden_size=config.hidden_size, eps=config.layernorm_epsilon)

    def forward(self, decoder_input, hidden_states, attention_mask=None,
                rotary_pos_emb=None, **kwargs):
        decoder_input = self.enorm(decoder_input)
        hidden_states = self.hnorm(hidden_states)
        hidden_states = torch.cat((decoder_input, hidden_states), -1)
        hidden_states, _ = self.eh_proj(hidden_states)
        hidden_states, _ = self.transformer_layer(
            hidden_states, attention_mask=attention_mask,
            rotary_pos_emb=rotary_pos_emb)
        return self.final_layernorm(hidden_states)


class MultiTokenPredictionBlock(MegatronModule):
    """The stack of MTP depths plus their loss plumbing."""

    def __init__(self, config: TransformerConfig, spec: ModuleSpec):
        super().__init__(config)
        self.num_layers = config.mtp_num_layers
        self.layers = torch.nn.ModuleList(
            [build_module(spec, config=config, layer_number=k + 1)
             for k in range(self.num_layers)])

    def forward(self, input_ids, position_ids, hidden_states,
                attention_mask=None, rotary_pos_emb=None, *, embedding,
                output_layer, output_weight, compute_loss, labels,
                loss_mask=None):
        """Returns hidden_states with every depth's scaled CE attached."""
        cfg = self.config
        ids_k, labels_k = input_ids, labels
        mask_k = loss_mask
        h = hidden_states
        for k, layer in enumerate(self.layers):
            # depth k trains on tokens shifted left by k+1
            ids_k = roll_tensor(ids_k)
            labels_k = roll_tensor(labels_k)
            if mask_k is not None:
                mask_k = roll_tensor(mask_k)
            emb_k = embedding(ids_k, position_ids)
            h = layer(emb_k, h, attention_mask=attention_mask,
                      rotary_pos_emb=rotary_pos_emb)
            logits_k, _ = output_layer(h, weight=output_weight)
            losses_k = compute_loss(labels_k, logits_k)  # [b, s]
            if mask_k is not None:
                denom = mask_k.sum().clamp(min=1)
                loss_k = (losses_k * mask_k).sum() / denom
            else:
                loss_k = losses_k.mean()
            scale = cfg.mtp_loss_scaling_factor / self.num_layers
            MTPLossLoggingHelper.save_loss_to_tracker(
                loss_k * scale, k, self.num_layers)
            hidden_states = MTPLossAutoScaler.apply(
                hidden_states, loss_k * scale)
        return hidden_states


def get_gpt_mtp_block_spec(config, transformer_layer_spec,
                           normalization: str = None) -> ModuleSpec:
    """Build the MTP layer spec from the model's transformer layer spec
    (reference gpt_layer_specs.py get_gpt_mtp_block_spec)."""
    from ..fusions.fused_layer_norm import FusedLayerNorm, FusedRMSNorm
    from ..tensor_parallel.layers import ColumnParallelLinear
    norm = (FusedRMSNorm if (normalization or config.normalization)
            == "RMSNorm" else FusedLayerNorm)
    return ModuleSpec(
        module=MultiTokenPredictionLayer,
        submodules=MultiTokenPredictionLayerSubmodules(
            enorm=norm, hnorm=norm, eh_proj=ColumnParallelLinear,
            transformer_layer=transformer_layer_spec, layer_norm=norm))
