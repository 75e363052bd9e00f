"""LR / weight-decay scheduler (reference optimizer_param_scheduler.py).

Supports linear warmup + {constant, linear, cosine, WSD} decay.
"""

from __future__ import annotations

import math


class OptimizerParamScheduler:
    def __init__(self, optimizer, *, init_lr: float, max_lr: float, min_lr: float,
                 lr_warmup_steps: int, lr_decay_steps: int,
                 lr_decay_style: str = "linear",
                 start_wd: float = 0.01, end_wd: float = 0.01,
                 wd_incr_steps: int = 0, wd_incr_style: str = "constant",
                 use_checkpoint_opt_param_scheduler: bool = False,
                 override_opt_param_scheduler: bool = False,
                 wsd_decay_steps: int = None, lr_wsd_decay_style: str = "exponential"):
        self.optimizer = optimizer
        self.init_lr = init_lr
        self.max_lr = max_lr
        self.min_lr = min_lr
        self.lr_warmup_steps = lr_warmup_steps
        self.lr_decay_steps = max(lr_decay_steps, 1)
        self.lr_decay_style = lr_decay_style
        self.start_wd = start_wd
        self.end_wd = end_wd
        self.wd_incr_steps = wd_incr_steps
        self.wd_incr_style = wd_incr_style
        self.wsd_decay_steps = wsd_decay_steps
        self.lr_wsd_decay_style = lr_wsd_decay_style
        self.num_steps = 0
        self.step(0)

    def get_wd(self) -> float:
        if self.wd_incr_steps <= 0 or self.wd_incr_style == "constant":
            return self.end_wd
        frac = min(self.num_steps / self.wd_incr_steps, 1.0)
        if self.wd_incr_style == "linear":
            return self.start_wd + (self.end_wd - self.start_wd) * frac
        if self.wd_incr_style == "cosine":
            return self.end_wd + (self.start_wd - self.end_wd) * 0.5 * (
                math.cos(math.pi * frac) + 1.0)
        return self.end_wd

    def get_lr(self, param_group=None) -> float:
        if self.lr_warmup_steps > 0 and self.num_steps <= self.lr_warmup_steps:
            return self.init_lr + (self.max_lr - self.init_lr) * (
                self.num_steps / self.lr_warmup_steps)
        if self.lr_decay_style == "constant":
            return self.max_lr
        steps = self.num_steps - self.lr_warmup_steps
        decay_steps = self.lr_decay_steps - self.lr_warmup_steps
        frac = min(max(steps / max(decay_steps, 1), 0.0), 1.0)
        delta = self.max_lr - self.min_lr
        if self.lr_decay_style == "linear":
            return self.max_lr - delta * frac
        if self.lr_decay_style == "cosine":
            return self.min_lr + delta * 0.5 * (math.cos(math.pi * frac) + 1.0)
        if self.lr_decay_style == "inverse-square-root":
            warm = max(self.lr_warmup_steps, 1)
            return max(self.min_lr,
                       self.max_lr * math.sqrt(warm) / math.sqrt(max(self.num_steps, warm)))
        if self.lr_decay_style == "WSD":
            wsd_start = self.lr_decay_steps - (self.wsd_decay_steps or 0)
            if self.num_steps < wsd_start:
                return self.max_lr
            wf = (self.num_steps - wsd_start) / max(self.wsd_decay_steps or 1, 1)
            if self.lr_wsd_decay_style == "linear":
                return self.max_lr - delta * wf
            if self.lr_wsd_decay_style == "cosine":
                return self.min_lr + delta * 0.5 * (math.cos(math.pi * wf) + 1.0)
            return self.max_lr * (self.min_lr / self.max_lr) ** wf
        raise ValueError(f"unknown lr decay style {self.lr_decay_style}")

    def step(self, increment: int = 1):
        self.num_steps += increment
        lr = self.get_lr()
        wd = self.get_wd()
        for group in self.optimizer.param_groups:
            group["lr"] = lr * group.get("lr_mult", 1.0)
            group["weight_decay"] = wd * group.get("wd_mult", 1.0)

    def state_dict(self):
        return {"num_steps": self.num_steps, "max_lr": self.max_lr,
                "min_lr": self.min_lr,
                "lr_warmup_steps": self.lr_warmup_steps,
                "lr_decay_steps": self.lr_decay_steps,
                "lr_decay_style": self.lr_decay_style}

    def load_state_dict(self, sd):
        self.num_steps = sd.get("num_steps", 0)
        self.step(0)
