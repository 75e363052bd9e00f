"""Core attention, two MI355X backends.

* ``DotProductAttention`` — the "fused" path: QK^T via hipBLASLt baddbmm,
  hand-written scaled-masked-softmax HIP kernel, PV via bmm.  Mirrors the
  reference local path (dot_product_attention.py:20-210) and is the
  numerics oracle for the flash kernel.
* ``FlashAttention``    — the flash path: a single MFMA HIP kernel
  (ops/csrc/attention.hip) computing the whole tile pipeline with online
  softmax; O(s) memory.  Default on GPU.
"""

from __future__ import annotations

import math
import os

import torch
from torch import nn

from .. import parallel_state
from ..enums import AttnMaskType
from ..fusions.fused_softmax import FusedScaleMaskSoftmax
from ..tensor_parallel.utils import divide
from ..transformer_config import TransformerConfig
from ... import ops as _ops


def attention_mask_func(attention_scores, attention_mask):
    return attention_scores.masked_fill(attention_mask, -10000.0)


class DotProductAttention(nn.Module):
    def __init__(self, config: TransformerConfig, layer_number: int,
                 attn_mask_type: AttnMaskType, attention_type: str = "self",
                 attention_dropout: float = None, softmax_scale=None,
                 cp_comm_type: str = None):
        super().__init__()
        self.config = config
        self.layer_number = max(1, layer_number)
        self.attn_mask_type = attn_mask_type

        projection_size = config.kv_channels * config.num_attention_heads
        world_size = parallel_state.get_tensor_model_parallel_world_size()
        self.hidden_size_per_partition = divide(projection_size, world_size)
        self.hidden_size_per_attention_head = config.kv_channels
        self.num_attention_heads_per_partition = divide(
            config.num_attention_heads, world_size)
        self.num_query_groups_per_partition = divide(
            config.num_query_groups, world_size)

        coeff = None
        self.norm_factor = math.sqrt(self.hidden_size_per_attention_head)
        if config.apply_query_key_layer_scaling:
            coeff = self.layer_number
            self.norm_factor *= coeff
        if softmax_scale is not None:
            self.norm_factor = 1.0 / softmax_scale

        self.scale_mask_softmax = FusedScaleMaskSoftmax(
            config.fp16, config.bf16, attn_mask_type,
            config.masked_softmax_fusion, attention_mask_func,
            config.attention_softmax_in_fp32, coeff)
        self.attention_dropout = nn.Dropout(
            config.attention_dropout if attention_dropout is None
            else attention_dropout)

    def forward(self, query, key, value, attention_mask=None,
                attn_mask_type=None, attention_bias=None, packed_seq_params=None):
        # q: [sq, b, np, hn]; k/v: [sk, b, ng, hn]
        sq, b, np_, hn = query.shape
        sk, _, ng, _ = key.shape
        if np_ != ng:
            rep = np_ // ng
            key = key.repeat_interleave(rep, dim=2)
            value = value.repeat_interleave(rep, dim=2)

        # permute+reshape stays a VIEW even for the strided _SplitQKV
        # outputs (b,np adjacent with nested strides); hipBLASLt handles
        # the resulting lda/batch strides without materializing copies
        q = query.permute(1, 2, 0, 3).reshape(b * np_, sq, hn)      # [b*np, sq, hn]
        k = key.permute(1, 2, 0, 3).reshape(b * np_, sk, hn)        # [b*np, sk, hn]
        scores = torch.empty(b * np_, sq, sk, dtype=query.dtype,
                             device=query.device)
        scores = torch.baddbmm(scores, q, k.transpose(1, 2), beta=0.0,
                               alpha=1.0 / self.norm_factor)
        scores = scores.view(b, np_, sq, sk)

        mask_type = attn_mask_type if attn_mask_type is not None else self.attn_mask_type
        if packed_seq_params is not None and \
                packed_seq_params.cu_seqlens_q is not None:
            # packed (THD) stream: block-diagonal causal mask
            from ..packed_seq_params import packed_attention_mask
            from ..enums import AttnMaskType as _AMT
            attention_mask = packed_attention_mask(
                packed_seq_params.cu_seqlens_q, sk,
                causal=(mask_type == _AMT.causal))
            mask_type = _AMT.padding
        probs = self.scale_mask_softmax(scores, attention_mask, mask_type)

        # MegaScope raw-attention-score tap (reference
        # dot_product_attention.py:168-170)
        from ..tensor_tracer import FlagType, get_tensor_tracers
        tt = get_tensor_tracers()
        if tt is not None and tt.enabled(FlagType.RawAttentionScore,
                                         self.layer_number):
            tt.report(FlagType.RawAttentionScore, self.layer_number, probs)
        if self.training and self.attention_dropout.p > 0:
            # fork only when dropout actually randomizes: the RNG state
            # swap is illegal inside hipGraph capture and pointless in eval
            from ..tensor_parallel.random import get_cuda_rng_tracker
            if torch.cuda.is_available():
                with get_cuda_rng_tracker().fork():
                    probs = self.attention_dropout(probs)
            else:
                probs = self.attention_dropout(probs)

        hn_v = value.shape[-1]  # MLA: v_head_dim may differ from qk dim
        v = value.permute(1, 2, 0, 3).reshape(b * np_, sk, hn_v)    # [b*np, sk, hn_v]
        context = torch.bmm(probs.view(b * np_, sq, sk), v)         # [b*np, sq, hn_v]
        context = context.transpose(0, 1).reshape(sq, b, np_ * hn_v)
        return context


class _FlashAttnFn(torch.autograd.Function):
    """Autograd over the MFMA flash kernels (fwd: O + row LSE; bwd: dQ/dK/dV)."""

    @staticmethod
    def forward(ctx, q, k, v, scale, causal, dropout_p):
        o, lse = _ops.get_ops().attn_fwd(q, k, v, scale, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        ops = _ops.get_ops()
        nh, hn = q.shape[2], q.shape[3]
        ng = k.shape[2]
        # rep==1: write dQ/dK/dV straight into one fused [s, b, g, 3*hn]
        # buffer laid out like the QKV GEMM output — when the grads flow
        # back to _SplitQKV unchanged (no rope in between), its backward
        # recognizes the shared base and skips three slice copies
        if (nh == ng and q.shape[0] == k.shape[0]
                and hasattr(ops, "attn_bwd_into")
                and "qkv_fuse" not in os.environ.get(
                    "MEGATRONAPP_DISABLE_FUSED", "")):
            sq, b = q.shape[0], q.shape[1]
            dm = torch.empty(sq, b, ng, 3 * hn, dtype=q.dtype,
                             device=q.device)
            dq_v = dm[..., :hn]
            dk_v = dm[..., hn:2 * hn]
            dv_v = dm[..., 2 * hn:]
            if ops.attn_bwd_into(do.contiguous(), q, k, v, o, lse,
                                 ctx.scale, ctx.causal, dq_v, dk_v, dv_v):
                return dq_v, dk_v, dv_v, None, None, None
        dq, dk, dv = ops.attn_bwd(
            do.contiguous(), q, k, v, o, lse, ctx.scale, ctx.causal)
        return dq, dk, dv, None, None, None


class FlashAttention(nn.Module):
    """Flash-attention core (causal / GQA); same call signature as
    DotProductAttention so layer specs can swap them."""

    def __init__(self, config: TransformerConfig, layer_number: int,
                 attn_mask_type: AttnMaskType, attention_type: str = "self",
                 attention_dropout: float = None, softmax_scale=None,
                 cp_comm_type: str = None):
        super().__init__()
        self.config = config
        self.attn_mask_type = attn_mask_type
        self.softmax_scale = (softmax_scale if softmax_scale is not None
                              else 1.0 / math.sqrt(config.kv_channels))
        self.dropout_p = (config.attention_dropout if attention_dropout is None
                          else attention_dropout)
        # fall back to the math path off-GPU or for unsupported shapes
        self._fallback = DotProductAttention(
            config, layer_number, attn_mask_type, attention_type,
            attention_dropout, softmax_scale, cp_comm_type)

    def _supported(self, query, key):
        if not query.is_cuda or not _ops.have_ops():
            return False
        if not hasattr(_ops.get_ops(), "attn_fwd"):
            return False
        if self.attn_mask_type != AttnMaskType.causal:
            return False
        if self.dropout_p > 0 and self.training:
            return False
        hn = query.shape[-1]
        if query.shape[0] % 128 != 0 or key.shape[0] % 64 != 0:
            return False
        return hn in (64, 128) and query.dtype == torch.bfloat16

    def forward(self, query, key, value, attention_mask=None,
                attn_mask_type=None, attention_bias=None, packed_seq_params=None):
        if not self._supported(query, key):
            return self._fallback(query, key, value, attention_mask,
                                  attn_mask_type, attention_bias, packed_seq_params)
        sq, b, np_, hn = query.shape
        # the kernels take strided views (d contiguous) directly — no
        # .contiguous() copies of the QKV-split outputs
        o = _FlashAttnFn.apply(query, key, value, self.softmax_scale, True,
                               self.dropout_p)
        return o.reshape(sq, b, np_ * hn)
