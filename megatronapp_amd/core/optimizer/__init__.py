"""Optimizer entry points (reference optimizer/__init__.py:431)."""

from __future__ import annotations

from typing import List, Optional

from .clip_grads import clip_grad_by_total_norm_fp32, get_grad_norm_fp32
from .distrib_optimizer import DistributedOptimizer
from .optimizer_config import OptimizerConfig


def get_megatron_optimizer(config: OptimizerConfig, model_chunks: List,
                           no_weight_decay_cond=None, scale_lr_cond=None,
                           lr_mult: float = 1.0,
                           use_gloo_process_groups: bool = True):
    """Build the buffer-aligned mixed-precision optimizer over every model
    chunk's DDP buffers (one optimizer for all chunks)."""
    return DistributedOptimizer(config, model_chunks)


__all__ = [
    "DistributedOptimizer",
    "OptimizerConfig",
    "get_megatron_optimizer",
    "clip_grad_by_total_norm_fp32",
    "get_grad_norm_fp32",
]
