"""Self-attention block: fused QKV projection, GQA, RoPE, KV cache.

Reference: transformer/attention.py (Attention:88, SelfAttention:845).
MegaScope taps (QKV / raw scores / context) hook via core.tensor_tracer.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional, Union

import torch

from .. import parallel_state
from ..enums import AttnMaskType
from ..models.common.embeddings.rotary_pos_embedding import apply_rotary_pos_emb
from ..tensor_parallel.layers import ColumnParallelLinear, RowParallelLinear
from ..tensor_parallel.utils import divide
from ..transformer_config import TransformerConfig
from ..trace_hooks import trace_scope
from ..tensor_tracer import get_tensor_tracers, FlagType
from .module import MegatronModule
from .spec_utils import ModuleSpec, build_module


@dataclass
class SelfAttentionSubmodules:
    linear_qkv: Union[ModuleSpec, type] = None
    core_attention: Union[ModuleSpec, type] = None
    linear_proj: Union[ModuleSpec, type] = None
    q_layernorm: Union[ModuleSpec, type] = None
    k_layernorm: Union[ModuleSpec, type] = None


class _SplitQKV(torch.autograd.Function):
    """Strided-view QKV split.

    torch.split + .contiguous() costs three 16 MB copies in forward and a
    batched-cat (~80 us/layer on MI355X) in backward.  The attention bmms
    consume strided views directly (hipBLASLt takes arbitrary lda/batch
    strides), so forward returns VIEWS into the fused-QKV GEMM output and
    backward assembles the grads with three slice copies into one buffer.
    The views are built on a detached alias (autograd treats them as new
    outputs); do not mutate them in place."""

    @staticmethod
    def forward(ctx, mixed, ng, rep, hn):
        sq, b, _ = mixed.shape
        ctx.dims = (sq, b, ng, rep, hn)
        m = mixed.detach().view(sq, b, ng, (rep + 2) * hn)
        # rep==1: all three are pure views.  rep>1 (GQA): the per-group
        # head interleave makes the merged q head-dim non-uniform-stride,
        # so q alone materializes; k/v stay views.
        q = m[..., :rep * hn]
        if rep > 1:
            q = q.reshape(sq, b, ng * rep, hn)
        k = m[..., rep * hn:(rep + 1) * hn]
        v = m[..., (rep + 1) * hn:]
        return q, k, v

    @staticmethod
    def backward(ctx, dq, dk, dv):
        sq, b, ng, rep, hn = ctx.dims
        # zero-copy path: the flash backward (attn_bwd_into) already wrote
        # the three grads into one fused buffer with exactly this layout
        if (rep == 1 and dq._base is not None and dq._base is dk._base
                and dk._base is dv._base):
            base = dq._base
            if (tuple(base.shape) == (sq, b, ng, 3 * hn)
                    and dq.storage_offset() == base.storage_offset()
                    and dk.storage_offset() == base.storage_offset() + hn
                    and dv.storage_offset() == base.storage_offset() + 2 * hn
                    and dq.stride() == dk.stride() == dv.stride()
                    and dq.stride() == base[..., :hn].stride()
                    and base.is_contiguous()):
                return base.view(sq, b, -1), None, None, None
        dm = torch.empty((sq, b, ng, (rep + 2) * hn), dtype=dq.dtype,
                         device=dq.device)
        dm[..., :rep * hn].copy_(dq.reshape(sq, b, ng, rep * hn))
        dm[..., rep * hn:(rep + 1) * hn].copy_(dk)
        dm[..., (rep + 1) * hn:].copy_(dv)
        return dm.view(sq, b, -1), None, None, None


class SelfAttention(MegatronModule):
    def __init__(self, config: TransformerConfig,
                 submodules: SelfAttentionSubmodules, layer_number: int,
                 attn_mask_type=AttnMaskType.causal, cp_comm_type: str = None):
        super().__init__(config)
        self.layer_number = layer_number
        self.attn_mask_type = attn_mask_type

        world_size = parallel_state.get_tensor_model_parallel_world_size()
        self.hidden_size_per_attention_head = config.kv_channels
        self.num_attention_heads_per_partition = divide(
            config.num_attention_heads, world_size)
        self.num_query_groups_per_partition = divide(
            config.num_query_groups, world_size)
        self.query_projection_size = config.kv_channels * config.num_attention_heads
        self.kv_projection_size = config.kv_channels * config.num_query_groups

        self.linear_qkv = build_module(
            submodules.linear_qkv, config.hidden_size,
            self.query_projection_size + 2 * self.kv_projection_size,
            config=config, init_method=config.init_method,
            bias=config.add_bias_linear or config.add_qkv_bias,
            skip_bias_add=False, gather_output=False)

        self.core_attention = build_module(
            submodules.core_attention, config=config,
            layer_number=layer_number, attn_mask_type=attn_mask_type,
            attention_type="self", cp_comm_type=cp_comm_type)
        if config.context_parallel_size > 1:
            from .cp_attention import ContextParallelAttention
            self.core_attention = ContextParallelAttention(
                self.core_attention, config)

        self.linear_proj = build_module(
            submodules.linear_proj, self.query_projection_size,
            config.hidden_size, config=config,
            init_method=config.output_layer_init_method,
            bias=config.add_bias_linear, input_is_parallel=True,
            skip_bias_add=True)

        if submodules.q_layernorm is not None and config.qk_layernorm:
            self.q_layernorm = build_module(
                submodules.q_layernorm, config=config,
                hidden_size=self.hidden_size_per_attention_head,
                eps=config.layernorm_epsilon)
        else:
            self.q_layernorm = None
        if submodules.k_layernorm is not None and config.qk_layernorm:
            self.k_layernorm = build_module(
                submodules.k_layernorm, config=config,
                hidden_size=self.hidden_size_per_attention_head,
                eps=config.layernorm_epsilon)
        else:
            self.k_layernorm = None

    def _split_qkv(self, mixed_qkv):
        """[sq, b, (np/g + 2) * g * hn] -> q [sq,b,np,hn], k/v [sq,b,ng,hn]."""
        ng = self.num_query_groups_per_partition
        np_ = self.num_attention_heads_per_partition
        hn = self.hidden_size_per_attention_head
        return _SplitQKV.apply(mixed_qkv, ng, np_ // ng, hn)

    def forward(self, hidden_states, attention_mask=None, key_value_states=None,
                inference_context=None, rotary_pos_emb=None, rotary_pos_cos=None,
                rotary_pos_sin=None, attention_bias=None, packed_seq_params=None,
                sequence_len_offset=None):
        # hidden_states: [sq, b, h]
        mixed_qkv, _ = self.linear_qkv(hidden_states)
        query, key, value = self._split_qkv(mixed_qkv)

        if self.q_layernorm is not None:
            query = self.q_layernorm(query)
        if self.k_layernorm is not None:
            key = self.k_layernorm(key)

        # MegaScope QKV tap (reference attention.py:979-981)
        tt = get_tensor_tracers()
        if tt is not None and tt.enabled(FlagType.QKV, self.layer_number):
            tt.report(FlagType.QKV, self.layer_number, (query, key, value))

        if rotary_pos_emb is not None:
            if isinstance(rotary_pos_emb, tuple):
                q_pos_emb, k_pos_emb = rotary_pos_emb
            else:
                q_pos_emb = k_pos_emb = rotary_pos_emb
            if parallel_state.get_context_parallel_world_size() > 1:
                # CP: local rows are global chunks (r, 2cp-1-r); take their
                # rows of the full-frequency table
                from .cp_attention import _local_global_positions
                cp = parallel_state.get_context_parallel_world_size()
                cp_rank = parallel_state.get_context_parallel_rank()
                pos = _local_global_positions(cp, cp_rank, query.shape[0],
                                              q_pos_emb.device)
                q_pos_emb = q_pos_emb.index_select(0, pos)
                k_pos_emb = k_pos_emb.index_select(0, pos)
            if inference_context is not None:
                if getattr(inference_context, "is_graph_context", False):
                    # hipGraph capture: position comes from a device tensor
                    idx = (inference_context.cur_len +
                           inference_context._arange[:query.shape[0]])
                    q_pos_emb = q_pos_emb.index_select(0, idx)
                    k_pos_emb = k_pos_emb.index_select(0, idx)
                elif getattr(inference_context, "is_dynamic", False) and \
                        query.shape[0] == 1:
                    # continuous batching: per-ROW positions
                    idx = inference_context.row_positions()
                    q_pos_emb = q_pos_emb.index_select(0, idx).permute(
                        1, 0, 2, 3)             # [1, b, 1, rot]
                    k_pos_emb = q_pos_emb
                else:
                    offset = inference_context.sequence_len_offset
                    q_pos_emb = q_pos_emb[offset:offset + query.shape[0]]
                    k_pos_emb = k_pos_emb[:inference_context.sequence_len_offset
                                          + key.shape[0]][-key.shape[0]:]
            query = apply_rotary_pos_emb(query, q_pos_emb, config=self.config)
            key = apply_rotary_pos_emb(key, k_pos_emb, config=self.config)

        attn_mask_type = self.attn_mask_type
        if inference_context is not None:
            key, value = inference_context.update_kv_cache(
                self.layer_number, key, value)
            if getattr(inference_context, "is_graph_context", False):
                # fixed-shape window + padding mask (hipGraph decode)
                attention_mask = inference_context.decode_padding_mask(
                    query.shape[0], query.shape[1])
                attn_mask_type = AttnMaskType.padding
            elif getattr(inference_context, "is_dynamic", False) and \
                    query.shape[0] == 1:
                # continuous batching: per-row window mask
                attention_mask = inference_context.decode_padding_mask()
                attn_mask_type = AttnMaskType.padding
            elif query.shape[0] == 1:
                attn_mask_type = AttnMaskType.no_mask

        with trace_scope("attention"):
            if (self.config.recompute_granularity == "selective"
                    and self.training and torch.is_grad_enabled()
                    and inference_context is None):
                # selective recompute (reference attention.py checkpointed
                # core attention): the [b,np,sq,sk] probs are recomputed in
                # backward instead of stored
                from ..tensor_parallel.random import checkpoint as tp_ckpt
                core_attn_out = tp_ckpt(
                    lambda q, k, v: self.core_attention(
                        q, k, v, attention_mask=attention_mask,
                        attn_mask_type=attn_mask_type,
                        attention_bias=attention_bias,
                        packed_seq_params=packed_seq_params),
                    False, query, key, value)
            else:
                core_attn_out = self.core_attention(
                    query, key, value, attention_mask=attention_mask,
                    attn_mask_type=attn_mask_type,
                    attention_bias=attention_bias,
                    packed_seq_params=packed_seq_params)

        if tt is not None and tt.enabled(FlagType.ContextLayer, self.layer_number):
            tt.report(FlagType.ContextLayer, self.layer_number, core_attn_out)

        output, bias = self.linear_proj(core_attn_out)
        return output, bias
