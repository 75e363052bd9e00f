"""Packed-sequence metadata (reference core/packed_seq_params.py).

Multiple variable-length sequences packed into one token stream; the
cu_seqlens prefix arrays mark boundaries.  DotProductAttention turns them
into a block-diagonal causal mask so tokens never attend across sequence
boundaries.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class PackedSeqParams:
    qkv_format: str = "thd"
    cu_seqlens_q: Optional[torch.Tensor] = None
    cu_seqlens_kv: Optional[torch.Tensor] = None
    max_seqlen_q: Optional[int] = None
    max_seqlen_kv: Optional[int] = None


def packed_attention_mask(cu_seqlens: torch.Tensor, total: int,
                          causal: bool = True) -> torch.Tensor:
    """[1, 1, total, total] bool mask, True = masked: block-diagonal over
    the packed segments, causal within each."""
    seg = torch.zeros(total, dtype=torch.long, device=cu_seqlens.device)
    seg[cu_seqlens[1:-1].long()] = 1
    seg = seg.cumsum(0)
    same = seg.unsqueeze(0) == seg.unsqueeze(1)
    mask = ~same
    if causal:
        pos = torch.arange(total, device=cu_seqlens.device)
        mask |= pos.unsqueeze(0) > pos.unsqueeze(1)
    return mask.view(1, 1, total, total)
