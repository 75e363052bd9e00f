"""Multi-latent attention (reference transformer/multi_latent_attention.py,
DeepSeek-V2 style).

Queries and keys/values are projected through low-rank "latent"
compressions: hidden -> kv_lora_rank (+ a shared rotary slice) -> per-head
qk_head_dim + v_head_dim.  The rotary part of K is computed once from the
compressed stream and broadcast to all heads; the non-rotary parts carry no
positional information.  On MI355X this keeps the KV cache at
kv_lora_rank + qk_pos_emb_head_dim per token (vs 2*n*hn), which is what
makes the large-batch decode path HBM-friendly.

Supported here: rope_type="rope" (yarn scaling is a config knob the rotary
embedding handles via rotary_scaling_factor), training and KV-cached
inference.  TP shards the up-projections per head like regular attention.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Union

import torch

from .. import parallel_state
from ..enums import AttnMaskType
from ..models.common.embeddings.rotary_pos_embedding import (
    RotaryEmbedding, apply_rotary_pos_emb)
from ..tensor_parallel.utils import divide
from ..transformer_config import MLATransformerConfig
from .module import MegatronModule
from .spec_utils import ModuleSpec, build_module


@dataclass
class MLASelfAttentionSubmodules:
    linear_q_proj: Union[ModuleSpec, type] = None
    linear_q_down_proj: Union[ModuleSpec, type] = None
    linear_q_up_proj: Union[ModuleSpec, type] = None
    linear_kv_down_proj: Union[ModuleSpec, type] = None
    linear_kv_up_proj: Union[ModuleSpec, type] = None
    core_attention: Union[ModuleSpec, type] = None
    linear_proj: Union[ModuleSpec, type] = None
    q_layernorm: Union[ModuleSpec, type] = None
    kv_layernorm: Union[ModuleSpec, type] = None


class MLASelfAttention(MegatronModule):
    def __init__(self, config: MLATransformerConfig,
                 submodules: MLASelfAttentionSubmodules, layer_number: int,
                 attn_mask_type=AttnMaskType.causal, cp_comm_type=None):
        super().__init__(config)
        self.layer_number = layer_number
        self.attn_mask_type = attn_mask_type
        cfg = config
        world = parallel_state.get_tensor_model_parallel_world_size()
        self.np_ = divide(cfg.num_attention_heads, world)
        self.q_head_dim = cfg.qk_head_dim + cfg.qk_pos_emb_head_dim

        if cfg.q_lora_rank is None:
            self.linear_q_proj = build_module(
                submodules.linear_q_proj, cfg.hidden_size,
                cfg.num_attention_heads * self.q_head_dim, config=cfg,
                init_method=cfg.init_method, bias=False, skip_bias_add=False,
                gather_output=False)
            self.linear_q_down_proj = None
        else:
            # down-projections are replicated (tiny), up-projections TP-shard
            self.linear_q_down_proj = torch.nn.Linear(
                cfg.hidden_size, cfg.q_lora_rank, bias=False,
                dtype=cfg.params_dtype)
            self.linear_q_up_proj = build_module(
                submodules.linear_q_up_proj, cfg.q_lora_rank,
                cfg.num_attention_heads * self.q_head_dim, config=cfg,
                init_method=cfg.init_method, bias=False, skip_bias_add=False,
                gather_output=False)
            self.q_layernorm = build_module(
                submodules.q_layernorm, config=cfg,
                hidden_size=cfg.q_lora_rank, eps=cfg.layernorm_epsilon)

        self.linear_kv_down_proj = torch.nn.Linear(
            cfg.hidden_size, cfg.kv_lora_rank + cfg.qk_pos_emb_head_dim,
            bias=False, dtype=cfg.params_dtype)
        self.linear_kv_up_proj = build_module(
            submodules.linear_kv_up_proj, cfg.kv_lora_rank,
            cfg.num_attention_heads * (cfg.qk_head_dim + cfg.v_head_dim),
            config=cfg, init_method=cfg.init_method, bias=False,
            skip_bias_add=False, gather_output=False)
        self.kv_layernorm = build_module(
            submodules.kv_layernorm, config=cfg,
            hidden_size=cfg.kv_lora_rank, eps=cfg.layernorm_epsilon)

        softmax_scale = 1.0 / math.sqrt(self.q_head_dim)
        self.core_attention = build_module(
            submodules.core_attention, config=cfg, layer_number=layer_number,
            attn_mask_type=attn_mask_type, attention_type="self",
            softmax_scale=softmax_scale)
        self.linear_proj = build_module(
            submodules.linear_proj, cfg.num_attention_heads * cfg.v_head_dim,
            cfg.hidden_size, config=cfg,
            init_method=cfg.output_layer_init_method,
            bias=cfg.add_bias_linear, input_is_parallel=True,
            skip_bias_add=True)

        self.rotary_pos_emb = RotaryEmbedding(
            cfg.qk_pos_emb_head_dim, rotary_percent=1.0,
            rotary_base=cfg.rotary_base)

    def forward(self, hidden_states, attention_mask=None,
                inference_context=None, rotary_pos_emb=None,
                rotary_pos_cos=None, rotary_pos_sin=None, attention_bias=None,
                packed_seq_params=None, sequence_len_offset=None, **kwargs):
        cfg = self.config
        sq, b, _ = hidden_states.shape

        # ---- Q path
        if self.linear_q_down_proj is not None:
            q_compressed = self.q_layernorm(
                self.linear_q_down_proj(hidden_states))
            q, _ = self.linear_q_up_proj(q_compressed)
        else:
            q, _ = self.linear_q_proj(hidden_states)
        q = q.view(sq, b, self.np_, self.q_head_dim)
        q_no_pe, q_pos = torch.split(
            q, [cfg.qk_head_dim, cfg.qk_pos_emb_head_dim], dim=-1)

        # ---- KV path: shared compressed stream + shared rotary slice
        kv_combined = self.linear_kv_down_proj(hidden_states)
        kv_compressed, k_pos = torch.split(
            kv_combined, [cfg.kv_lora_rank, cfg.qk_pos_emb_head_dim], dim=-1)
        kv_compressed = self.kv_layernorm(kv_compressed.contiguous())
        kv, _ = self.linear_kv_up_proj(kv_compressed)
        kv = kv.view(sq, b, self.np_, cfg.qk_head_dim + cfg.v_head_dim)
        k_no_pe, value = torch.split(
            kv, [cfg.qk_head_dim, cfg.v_head_dim], dim=-1)

        # ---- rotary on the positional slices
        offset = (inference_context.sequence_len_offset
                  if inference_context is not None else 0)
        freqs = self.rotary_pos_emb(offset + sq)
        q_pos = apply_rotary_pos_emb(q_pos.contiguous(),
                                     freqs[offset:offset + sq], config=cfg)
        k_pos = apply_rotary_pos_emb(
            k_pos.unsqueeze(2).contiguous(), freqs[offset:offset + sq],
            config=cfg)

        key = torch.cat(
            [k_no_pe, k_pos.expand(sq, b, self.np_, cfg.qk_pos_emb_head_dim)],
            dim=-1)
        query = torch.cat([q_no_pe, q_pos], dim=-1)

        if inference_context is not None:
            key, value = inference_context.update_kv_cache(
                self.layer_number, key, value)
            attn_mask_type = (AttnMaskType.no_mask if sq == 1
                              else self.attn_mask_type)
        else:
            attn_mask_type = self.attn_mask_type

        context = self.core_attention(
            query.contiguous(), key.contiguous(), value.contiguous(),
            attention_mask=attention_mask, attn_mask_type=attn_mask_type)
        output, bias = self.linear_proj(context)
        return output, bias
