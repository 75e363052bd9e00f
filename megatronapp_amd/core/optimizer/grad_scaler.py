"""Dynamic loss scaler for fp16 (reference core/optimizer/grad_scaler.py).

Multiplies the loss by `scale` before backward; the optimizer unscales
gradients and skips the step when an inf/nan is found, backing the scale
off; after `growth_interval` clean steps the scale doubles.  bf16 needs
none of this (its exponent range matches fp32) — fp16 does.
"""

from __future__ import annotations

import torch


class ConstantGradScaler:
    def __init__(self, scale: float):
        self._scale = torch.tensor([float(scale)])

    @property
    def scale(self):
        return self._scale

    @property
    def inv_scale(self):
        return 1.0 / float(self._scale)

    def update(self, found_inf: bool):
        pass


class DynamicGradScaler:
    def __init__(self, initial_scale: float = 2 ** 32,
                 min_scale: float = 1.0, growth_factor: float = 2.0,
                 backoff_factor: float = 0.5, growth_interval: int = 1000,
                 hysteresis: int = 2):
        self._scale = torch.tensor([float(initial_scale)])
        self.min_scale = min_scale
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self.hysteresis = hysteresis
        self._growth_tracker = 0
        self._hysteresis_tracker = hysteresis

    @property
    def scale(self):
        return self._scale

    @property
    def inv_scale(self):
        return 1.0 / float(self._scale)

    def update(self, found_inf: bool):
        if found_inf:
            self._growth_tracker = 0
            self._hysteresis_tracker -= 1
            if self._hysteresis_tracker <= 0:
                self._scale = torch.clamp(
                    self._scale * self.backoff_factor, min=self.min_scale)
        else:
            self._growth_tracker += 1
            self._hysteresis_tracker = self.hysteresis
            if self._growth_tracker >= self.growth_interval:
                self._growth_tracker = 0
                self._scale = self._scale * self.growth_factor
