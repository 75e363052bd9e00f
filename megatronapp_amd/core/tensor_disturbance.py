"""MegaScope perturbation module (reference tensor_disturbance.py:1-75).

Three independent switches — weight / calculation / system perturbation —
with two noise factories: ``noise1`` additive Gaussian x + N(0, coef²) and
``noise2`` multiplicative uniform x · U(1−v, 1+v).  Injection points:
MLP output (transformer_layer), between layers (transformer_block), and
wherever parameters are read (weight perturbation).
"""

from __future__ import annotations

from typing import Callable, Dict, Optional

import torch


def noise1(x: torch.Tensor, coef: float) -> torch.Tensor:
    return x + torch.randn_like(x) * coef


def noise2(x: torch.Tensor, coef: float) -> torch.Tensor:
    return x * (1.0 + (torch.rand_like(x) * 2.0 - 1.0) * coef)


NOISE_REGISTRY: Dict[str, Callable] = {"noise1": noise1, "noise2": noise2}


class Disturbance:
    def __init__(self):
        self.weight_perturbation = False
        self.weight_fn = noise1
        self.weight_coef = 0.0
        self.calculation_perturbation = False
        self.calculation_fn = noise1
        self.calculation_coef = 0.0
        self.system_perturbation = False
        self.system_fn = noise2
        self.system_coef = 0.0

    def set_by_configs(self, configs: dict):
        configs = configs or {}

        def _b(v):
            return v == "True" if isinstance(v, str) else bool(v)

        self.weight_perturbation = _b(configs.get("weight_perturbation", False))
        self.weight_fn = NOISE_REGISTRY.get(
            configs.get("weight_perturbation_fn", "noise1"), noise1)
        self.weight_coef = float(configs.get("weight_perturbation_coef", 0.0))
        self.calculation_perturbation = _b(
            configs.get("calculation_perturbation", False))
        self.calculation_fn = NOISE_REGISTRY.get(
            configs.get("calculation_perturbation_fn", "noise1"), noise1)
        self.calculation_coef = float(
            configs.get("calculation_perturbation_coef", 0.0))
        self.system_perturbation = _b(configs.get("system_perturbation", False))
        self.system_fn = NOISE_REGISTRY.get(
            configs.get("system_perturbation_fn", "noise2"), noise2)
        self.system_coef = float(configs.get("system_perturbation_coef", 0.0))

    # injection helpers -----------------------------------------------
    def perturb_weight(self, w: torch.Tensor) -> torch.Tensor:
        if not self.weight_perturbation:
            return w
        with torch.no_grad():
            return self.weight_fn(w, self.weight_coef)

    def perturb_calculation(self, x: torch.Tensor) -> torch.Tensor:
        if not self.calculation_perturbation:
            return x
        return self.calculation_fn(x, self.calculation_coef)

    def perturb_system(self, x: torch.Tensor) -> torch.Tensor:
        if not self.system_perturbation:
            return x
        return self.system_fn(x, self.system_coef)

    @property
    def any_enabled(self) -> bool:
        return (self.weight_perturbation or self.calculation_perturbation
                or self.system_perturbation)


_DISTURBANCE: Optional[Disturbance] = None


def get_disturbance() -> Disturbance:
    global _DISTURBANCE
    if _DISTURBANCE is None:
        _DISTURBANCE = Disturbance()
    return _DISTURBANCE
