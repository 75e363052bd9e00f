"""Optimizer config (reference optimizer/optimizer_config.py)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional


@dataclass
class OptimizerConfig:
    optimizer: str = "adam"
    lr: Optional[float] = None
    min_lr: Optional[float] = None
    weight_decay: float = 0.01
    fp16: bool = False
    bf16: bool = False
    params_dtype: object = None
    loss_scale: Optional[float] = None
    initial_loss_scale: float = 2 ** 32
    min_loss_scale: float = 1.0
    loss_scale_window: int = 1000
    hysteresis: int = 2
    adam_beta1: float = 0.9
    adam_beta2: float = 0.999
    adam_eps: float = 1e-8
    sgd_momentum: float = 0.9
    use_distributed_optimizer: bool = False
    overlap_param_gather: bool = False
    clip_grad: float = 1.0
    log_num_zeros_in_grad: bool = False
    barrier_with_L1_time: bool = False
    # precision-aware optimizer (reference --use-precision-aware-optimizer):
    # store Adam exp_avg / exp_avg_sq in bf16 (fp32 math in-kernel) —
    # halves optimizer-state memory and the optimizer HBM stream
    use_precision_aware_optimizer: bool = False
    exp_avg_dtype: str = "fp32"
    exp_avg_sq_dtype: str = "fp32"
