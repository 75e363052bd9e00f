"""Rotary position embeddings (reference rotary_pos_embedding.py:36 +
rope_utils.py fused apply).

The cos/sin table is precomputed once per (seq_len, rotary_dim) on device
(Appendix B of the CDNA4 guide: never evaluate trig per element on the
VALU — that turns an HBM-bound op into a VALU-bound one).  The fused GPU
apply is ops/csrc/rope.hip; CPU fallback is the rotate-half formula.
"""

from __future__ import annotations

from typing import Optional

import torch
from torch import nn

from ..... import ops as _ops


class RotaryEmbedding(nn.Module):
    def __init__(self, kv_channels: int, rotary_percent: float = 1.0,
                 rotary_interleaved: bool = False, seq_len_interpolation_factor=None,
                 rotary_base: int = 10000, use_cpu_initialization: bool = False):
        super().__init__()
        dim = kv_channels
        if rotary_percent < 1.0:
            dim = int(dim * rotary_percent)
        self.dim = dim
        self.rotary_interleaved = rotary_interleaved
        self.seq_len_interpolation_factor = seq_len_interpolation_factor
        inv_freq = 1.0 / (rotary_base ** (
            torch.arange(0, dim, 2, dtype=torch.float32) / dim))
        self.register_buffer("inv_freq", inv_freq, persistent=False)
        self._cache = {}

    def forward(self, max_seq_len: int, offset: int = 0) -> torch.Tensor:
        """Returns freqs [s, 1, 1, dim] (angle per position/channel)."""
        key = (max_seq_len, offset, str(self.inv_freq.device))
        if key in self._cache:
            return self._cache[key]
        seq = torch.arange(max_seq_len, device=self.inv_freq.device,
                           dtype=torch.float32) + offset
        if self.seq_len_interpolation_factor is not None:
            seq = seq / self.seq_len_interpolation_factor
        freqs = torch.outer(seq, self.inv_freq)
        if not self.rotary_interleaved:
            emb = torch.cat((freqs, freqs), dim=-1)
        else:
            emb = torch.stack((freqs.reshape(-1, 1), freqs.reshape(-1, 1)),
                              dim=-1).reshape(freqs.shape[0], -1)
        emb = emb[:, None, None, :]
        self._cache[key] = emb
        return emb

    def get_rotary_seq_len(self, inference_context, transformer, transformer_input,
                           transformer_config, packed_seq_params=None) -> int:
        if inference_context is not None:
            return inference_context.max_sequence_length
        from .... import parallel_state
        seq_len = transformer_input.size(0)
        if transformer_config.sequence_parallel:
            seq_len *= parallel_state.get_tensor_model_parallel_world_size()
        seq_len *= transformer_config.context_parallel_size
        return seq_len


def _rotate_half(x):
    x1, x2 = torch.chunk(x, 2, dim=-1)
    return torch.cat((-x2, x1), dim=-1)


class _FusedRoPEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, t, cos, sin):
        ctx.save_for_backward(cos, sin)
        return _ops.get_ops().rope_fwd(t, cos, sin)

    @staticmethod
    def backward(ctx, dy):
        cos, sin = ctx.saved_tensors
        return _ops.get_ops().rope_bwd(dy.contiguous(), cos, sin), None, None


def apply_rotary_pos_emb(t: torch.Tensor, freqs: torch.Tensor,
                         config=None, cu_seqlens=None, fused: Optional[bool] = None
                         ) -> torch.Tensor:
    """t: [s, b, nh, hd]; freqs: [s, 1, 1, rot_dim] angles."""
    rot_dim = freqs.shape[-1]
    t_rot, t_pass = t[..., :rot_dim], t[..., rot_dim:]
    cos = torch.cos(freqs)
    sin = torch.sin(freqs)
    if fused is None:
        fused = (_ops.fused_enabled(t, "rope")
                 and (config is None or config.apply_rope_fusion))
    if fused and t.is_cuda:
        out = _FusedRoPEFn.apply(t_rot.contiguous(), cos, sin)
    else:
        cos = cos.to(t.dtype) if not t.is_cuda else cos
        sin = sin.to(t.dtype) if not t.is_cuda else sin
        out = (t_rot.float() * cos + _rotate_half(t_rot.float()) * sin).to(t.dtype)
    if t_pass.shape[-1] == 0:
        return out
    return torch.cat((out, t_pass), dim=-1)
