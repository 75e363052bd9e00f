"""Retro cross-attention operators (reference
core/models/retro/{base_attention,decoder_attention,encoder_attention}.py).

Chunked cross-attention (CCA), RETRO paper arXiv:2112.04426:

* the decoder sequence is split into chunks of ``retro_chunk_length``;
  chunk i attends to the encoded retrieval neighbors of chunk i-1
  (implemented as the reference does, by shifting the attending window
  to start at the last token of the first chunk)
* the neighbor encoder is a small transformer block instantiated inside
  the FIRST retro decoder layer; its retro layer cross-attends each
  neighbor back to its decoder chunk, and its output replaces the
  threaded ``context`` for all later retro decoder layers
* the custom bias-dropout-add operators undo the chunk permutes and are
  installed via ``TransformerLayerSubmodules.cross_attn_bda``
"""

from __future__ import annotations

import math
from functools import partial
from typing import Callable, List, Optional, Tuple, Type

import torch
from torch import Tensor

from ...enums import AttnMaskType
from ...fusions.fused_bias_dropout import get_bias_dropout_add
from ...transformer.cross_attention import (
    CrossAttention,
    CrossAttentionSubmodules,
)
from ...transformer.module import MegatronModule
from ...transformer.spec_utils import ModuleSpec
from ...transformer.transformer_block import TransformerBlock
from .config import RetroConfig


class BaseRetroCrossAttention(MegatronModule):
    """Shared wrapper around the plain cross-attention module
    (reference base_attention.py:14-43)."""

    def __init__(self, config: RetroConfig,
                 submodules: CrossAttentionSubmodules,
                 layer_number: int = 1,
                 attn_mask_type: AttnMaskType = AttnMaskType.padding):
        super().__init__(config=config)
        self.attn = CrossAttention(
            config=config, submodules=submodules, layer_number=layer_number,
            attn_mask_type=attn_mask_type)
        self.retro_num_neighbors = config.retro_num_neighbors
        self.retro_chunk_length = config.retro_chunk_length
        self.retro_retrieved_length = config.retro_retrieved_length


class RetroDecoderCrossAttention(BaseRetroCrossAttention):
    """Decoder CCA (reference decoder_attention.py:25-216).

    Returns a dict consumed by RetroDecoderBiasDropoutAdd; when this is
    the first retro layer (``encoder_block_spec`` given) it also encodes
    the raw neighbor embeddings and exposes them as ``context`` so the
    transformer block threads them to later retro layers.
    """

    def __init__(self, config: RetroConfig,
                 submodules: CrossAttentionSubmodules,
                 layer_number: int = 1,
                 attn_mask_type: AttnMaskType = AttnMaskType.padding,
                 encoder_block_spec: ModuleSpec = None):
        super().__init__(config=config, submodules=submodules,
                         layer_number=layer_number,
                         attn_mask_type=attn_mask_type)
        if encoder_block_spec is not None:
            self.encoder = TransformerBlock(
                config=config, spec=encoder_block_spec,
                pre_process=True, post_process=False)
        else:
            self.encoder = None

    def forward(self, hidden_states: Tensor, attention_mask: Tensor = None,
                key_value_states: Tensor = None, **kwargs) -> dict:
        # hidden_states: [ns, bs, d]; key_value_states: [r, k*bs*l, d]
        ns, bs, d = hidden_states.shape
        m = self.retro_chunk_length
        l = int(math.ceil(ns / m))

        if self.encoder is not None:
            # chunk the decoder hidden states (pad a leading partial
            # chunk) to [m, bs*l, d] for the encoder's cross-attention
            first_ns = ns % m
            if first_ns > 0:
                first = torch.nn.functional.pad(
                    hidden_states[:first_ns],
                    (0, 0, 0, 0, 0, m - first_ns))
                chunked = torch.cat([first, hidden_states[first_ns:]], dim=0)
            else:
                chunked = hidden_states
            chunked = (chunked.reshape(l, m, bs, d).permute(1, 2, 0, 3)
                       .reshape(m, bs * l, d).contiguous())
            key_value_states = self.encoder(
                key_value_states, attention_mask=None,
                context=chunked, context_mask=None)  # [r, k*bs*l, d]
            key_value_states = key_value_states.reshape(
                self.retro_retrieved_length * self.retro_num_neighbors,
                bs * l, d)  # [r*k, bs*l, d]

        # attend starting at the last token of the first chunk
        pad = (ns - 1) % m
        attending = torch.nn.functional.pad(
            hidden_states[pad:], (0, 0, 0, 0, 0, m - 1))
        attending = (attending.reshape(l, m, bs, d).permute(1, 2, 0, 3)
                     .reshape(m, bs * l, d).contiguous())

        attention_output, attention_bias = self.attn(
            attending, attention_mask=None,
            key_value_states=key_value_states)

        return {"ns": ns, "bs": bs, "d": d, "l": l, "pad": pad,
                "attention_output": attention_output,   # [m, bs*l, d]
                "attention_bias": attention_bias,       # [d]
                "context": key_value_states}            # [r*k, bs*l, d]


class RetroDecoderBiasDropoutAdd(MegatronModule):
    """Chunk-aligned bias-dropout-add + inverse permute
    (reference decoder_attention.py:219-314)."""

    def __init__(self, config: RetroConfig):
        super().__init__(config=config)
        self.retro_chunk_length = config.retro_chunk_length

    @classmethod
    def _forward(cls, x_with_bias: dict, residual: Tensor, prob: float,
                 retro_chunk_length: int,
                 bias_dropout_add: Callable) -> Tensor:
        ns = x_with_bias["ns"]
        bs = x_with_bias["bs"]
        d = x_with_bias["d"]
        l = x_with_bias["l"]
        pad = x_with_bias["pad"]
        out = x_with_bias["attention_output"]
        bias = x_with_bias["attention_bias"]
        with torch.enable_grad():
            x = bias_dropout_add(
                (out, None if bias is None else bias.expand_as(out)),
                torch.zeros_like(out), prob)
            # [m, bs*l, d] -> [l*m, bs, d], zero-prefix non-attending
            x = (x.reshape(retro_chunk_length, bs, l, d).permute(2, 0, 1, 3)
                 .reshape(retro_chunk_length * l, bs, d))
            x = torch.nn.functional.pad(
                x, (0, 0, 0, 0, pad, 0))[:ns]
            x = x + residual
        return x

    def forward(self, training: bool, fused: bool) -> partial:
        return partial(self._forward,
                       retro_chunk_length=self.retro_chunk_length,
                       bias_dropout_add=get_bias_dropout_add(training, fused))


class RetroEncoderCrossAttention(BaseRetroCrossAttention):
    """Encoder cross-attention: each retrieved neighbor attends back to
    its decoder chunk (reference encoder_attention.py:20-106)."""

    def forward(self, hidden_states: Tensor, attention_mask: Tensor = None,
                key_value_states: Tensor = None, **kwargs
                ) -> List[Tuple[Tensor, Optional[Tensor], Tensor]]:
        # hidden_states: [r, k*bs*l, d]; key_value_states: [m, bs*l, d]
        ns, bs, d = hidden_states.shape
        chunked = hidden_states.reshape(
            self.retro_retrieved_length, -1, self.retro_num_neighbors, d)
        outs = []
        for k in range(self.retro_num_neighbors):
            chunk = chunked[:, :, k].contiguous()  # [r, bs*l, d]
            attention_output, attention_bias = self.attn(
                chunk, attention_mask=None, key_value_states=key_value_states)
            # the per-neighbor residual is the (normed) neighbor chunk
            outs.append((attention_output, attention_bias, chunk))
        return outs


class RetroEncoderBiasDropoutAdd(MegatronModule):
    """Per-neighbor bias-dropout-add, concatenated back to
    [r, k*bs*l, d] (reference encoder_attention.py:109-186)."""

    def __init__(self, config: RetroConfig):
        super().__init__(config=config)
        self.retro_num_neighbors = config.retro_num_neighbors

    @classmethod
    def _forward(cls, x_with_bias: list, residual: Tensor, prob: float,
                 retro_num_neighbors: int,
                 bias_dropout_add: Callable) -> Tensor:
        with torch.enable_grad():
            outs = [bias_dropout_add(
                        (out, None if bias is None else bias.expand_as(res)),
                        res, prob)
                    for out, bias, res in x_with_bias]
        r, _, d = outs[0].shape
        return torch.stack(outs, dim=1).reshape(r, -1, d)

    def forward(self, training: bool, fused: bool) -> partial:
        return partial(self._forward,
                       retro_num_neighbors=self.retro_num_neighbors,
                       bias_dropout_add=get_bias_dropout_add(training, fused))


class RetroEncoderLayerNorm(MegatronModule):
    """Per-neighbor layer norm, concatenated back
    (reference encoder_attention.py:189-231)."""

    def __init__(self, config: RetroConfig, submodules: Type,
                 **kwargs):
        super().__init__(config=config)
        self.norm = submodules(config=config, **kwargs)
        self.retro_num_neighbors = config.retro_num_neighbors

    def forward(self, input: Tensor) -> Tensor:
        chunk = input.shape[1] // self.retro_num_neighbors
        inputs = torch.split(input, chunk, dim=1)
        outs = [self.norm(inp.contiguous()) for inp in inputs]
        r, _, d = inputs[0].shape
        return torch.stack(outs, dim=1).reshape(r, -1, d)
