"""MoE top-k router (reference transformer/moe/router.py TopKRouter:102).

Score functions: softmax / sigmoid; load balancing: switch-style aux loss
(seq-level), sinkhorn, or none; z-loss; MoEAuxLossAutoScaler attaches the
aux loss to the main autograd graph without touching activations.
"""

from __future__ import annotations

import torch

from ...transformer_config import TransformerConfig


class MoEAuxLossAutoScaler(torch.autograd.Function):
    """Pass-through that injects d(aux_loss)/d(activation)=0 but scales
    the aux loss into the backward graph (reference moe_utils)."""

    main_loss_backward_scale = 1.0

    @staticmethod
    def forward(ctx, output, aux_loss):
        ctx.save_for_backward(aux_loss)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        (aux_loss,) = ctx.saved_tensors
        scale = MoEAuxLossAutoScaler.main_loss_backward_scale
        return grad_output, torch.ones_like(aux_loss) * scale


def switch_load_balancing_loss(probs: torch.Tensor, tokens_per_expert,
                               topk: int, coeff: float) -> torch.Tensor:
    """aux = E * sum_e f_e * P_e (Switch Transformer eq. 4)."""
    num_experts = probs.shape[-1]
    total = probs.shape[0] * topk
    frac = tokens_per_expert.float() / max(total, 1)
    mean_prob = probs.float().mean(dim=0)
    return coeff * num_experts * torch.sum(frac * mean_prob)


def z_loss_func(logits: torch.Tensor, coeff: float) -> torch.Tensor:
    return coeff * torch.logsumexp(logits.float(), dim=-1).square().mean()


def sinkhorn(cost: torch.Tensor, tol: float = 1e-4, iters: int = 8):
    """Sinkhorn normalization over (tokens, experts) (reference router)."""
    cost = torch.exp(cost.float())
    d0 = torch.ones(cost.size(0), device=cost.device)
    d1 = torch.ones(cost.size(1), device=cost.device)
    eps = 1e-8
    for _ in range(iters):
        d0 = (1.0 / cost.size(0)) / (torch.sum(d1 * cost, dim=1) + eps)
        d1 = (1.0 / cost.size(1)) / (torch.sum(d0.unsqueeze(1) * cost, dim=0) + eps)
    return d1 * cost * d0.unsqueeze(1)


class TopKRouter(torch.nn.Module):
    def __init__(self, config: TransformerConfig):
        super().__init__()
        self.config = config
        self.num_experts = config.num_moe_experts
        self.topk = config.moe_router_topk
        # stored in params_dtype so it lives in the DDP flat param buffer
        # like every other weight; gating MATH stays fp32 below
        self.weight = torch.nn.Parameter(torch.empty(
            self.num_experts, config.hidden_size,
            dtype=config.params_dtype))
        config.init_method(self.weight)
        setattr(self.weight, "sequence_parallel", config.sequence_parallel)

    def gating(self, x: torch.Tensor) -> torch.Tensor:
        return torch.nn.functional.linear(x.float(), self.weight.float())

    def forward(self, hidden: torch.Tensor):
        """hidden [n_tokens, h] -> (probs [n, topk], indices [n, topk],
        aux_loss or None)."""
        logits = self.gating(hidden)
        if self.config.moe_router_load_balancing_type == "sinkhorn":
            with torch.no_grad():
                norm = sinkhorn(logits.detach())
                _, indices = torch.topk(norm, self.topk, dim=1)
            scores = torch.softmax(logits, dim=-1)
            probs = scores.gather(1, indices)
            aux = None
        else:
            scores = torch.softmax(logits, dim=-1)
            probs, indices = torch.topk(scores, self.topk, dim=1)
            aux = None
            if (self.config.moe_router_load_balancing_type == "aux_loss" and
                    self.config.moe_aux_loss_coeff > 0):
                with torch.no_grad():
                    tokens_per_expert = torch.bincount(
                        indices.flatten(), minlength=self.num_experts)
                aux = switch_load_balancing_loss(
                    scores, tokens_per_expert, self.topk,
                    self.config.moe_aux_loss_coeff)
        if getattr(self.config, "moe_router_renormalize", False):
            probs = probs / probs.sum(dim=-1, keepdim=True)
        if self.config.moe_z_loss_coeff:
            z = z_loss_func(logits, self.config.moe_z_loss_coeff)
            aux = z if aux is None else aux + z
        return probs, indices, aux
