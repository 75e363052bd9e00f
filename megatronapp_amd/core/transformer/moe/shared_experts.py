"""Shared expert MLP run on every token alongside routed experts
(reference moe/shared_experts.py:247)."""

from __future__ import annotations

import torch

from ...transformer_config import TransformerConfig
from ..mlp import MLP, MLPSubmodules
from ..module import MegatronModule


class SharedExpertMLP(MegatronModule):
    def __init__(self, config: TransformerConfig, submodules: MLPSubmodules,
                 gate: bool = False):
        super().__init__(config)
        self.mlp = MLP(config, submodules,
                       ffn_hidden_size=config.moe_shared_expert_intermediate_size)
        self.gate = None
        if gate:
            self.gate = torch.nn.Parameter(
                torch.zeros(config.hidden_size, dtype=config.params_dtype))

    def forward(self, hidden_states):
        out, bias = self.mlp(hidden_states)
        if bias is not None:
            out = out + bias
        if self.gate is not None:
            out = out * torch.sigmoid(hidden_states @ self.gate).unsqueeze(-1)
        return out, None
