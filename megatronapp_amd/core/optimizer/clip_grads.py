"""Global gradient-norm clipping across the model-parallel world.

Reference: optimizer/clip_grads.py:233 — l2 norm over non-duplicate grads,
all-reduced across the model-parallel group, then a single scale pass
(HIP multi-tensor scale on GPU via torch._foreach_mul_).
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


def get_grad_norm_fp32(grads_for_norm: List[torch.Tensor],
                       model_parallel_group=None,
                       extra_groups: Optional[list] = None) -> float:
    if not grads_for_norm:
        total = torch.zeros(1, dtype=torch.float32,
                            device="cuda" if torch.cuda.is_available() else "cpu")
    else:
        norms = torch._foreach_norm(grads_for_norm, 2.0)
        total = torch.norm(torch.stack(norms), 2.0) ** 2
        total = total.reshape(1)
    if model_parallel_group is not None and \
            dist.get_world_size(model_parallel_group) > 1:
        dist.all_reduce(total, op=dist.ReduceOp.SUM, group=model_parallel_group)
    for g in (extra_groups or []):
        if g is not None and dist.get_world_size(g) > 1:
            dist.all_reduce(total, op=dist.ReduceOp.SUM, group=g)
    return total.item() ** 0.5


def clip_grad_by_total_norm_fp32(grads: List[torch.Tensor], max_norm: float,
                                 total_norm: float):
    clip_coeff = max_norm / (total_norm + 1.0e-6)
    if clip_coeff < 1.0 and grads:
        torch._foreach_mul_(grads, clip_coeff)


def count_zeros_fp32(grads: List[torch.Tensor], model_parallel_group=None) -> float:
    total = torch.zeros(1, dtype=torch.float32,
                        device=grads[0].device if grads else "cpu")
    for g in grads:
        total += (g == 0).sum().float()
    if model_parallel_group is not None and \
            dist.get_world_size(model_parallel_group) > 1:
        dist.all_reduce(total, group=model_parallel_group)
    return total.item()
