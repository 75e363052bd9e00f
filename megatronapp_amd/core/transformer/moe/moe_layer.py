"""MoE layer: router -> dispatcher -> experts -> combine (+ shared expert).

Reference: moe/moe_layer.py:76.  Drop-in replacement for MLP in the
transformer layer spec (same (output, bias) return contract).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional, Union

import torch

from ... import parallel_state
from ...transformer_config import TransformerConfig
from ..module import MegatronModule
from ..spec_utils import ModuleSpec, build_module
from .router import MoEAuxLossAutoScaler, TopKRouter
from .token_dispatcher import (
    MoEAllGatherTokenDispatcher,
    MoEAlltoAllTokenDispatcher,
)


@dataclass
class MoESubmodules:
    experts: Union[ModuleSpec, type] = None
    shared_experts: Union[ModuleSpec, type] = None


class MoELayer(MegatronModule):
    def __init__(self, config: TransformerConfig,
                 submodules: MoESubmodules = None, layer_number: int = 1):
        super().__init__(config)
        self.config = config
        ep = parallel_state.get_expert_model_parallel_world_size()
        assert config.num_moe_experts % ep == 0, (
            f"num_experts {config.num_moe_experts} not divisible by ep {ep}")
        self.num_local_experts = config.num_moe_experts // ep
        ep_rank = parallel_state.get_expert_model_parallel_rank()
        self.local_expert_indices = list(range(
            ep_rank * self.num_local_experts,
            (ep_rank + 1) * self.num_local_experts))

        self.router = TopKRouter(config)
        if config.moe_token_dispatcher_type == "allgather":
            self.token_dispatcher = MoEAllGatherTokenDispatcher(
                self.num_local_experts, self.local_expert_indices, config)
        else:
            self.token_dispatcher = MoEAlltoAllTokenDispatcher(
                self.num_local_experts, self.local_expert_indices, config)

        experts_spec = submodules.experts if submodules else None
        self.experts = build_module(experts_spec,
                                    self.num_local_experts, config)
        self.shared_experts = None
        if (submodules is not None and submodules.shared_experts is not None
                and config.moe_shared_expert_intermediate_size):
            self.shared_experts = build_module(
                submodules.shared_experts, config=config)

    def forward(self, hidden_states: torch.Tensor):
        # hidden_states: [s, b, h]
        s, b, h = hidden_states.shape
        tokens = hidden_states.reshape(-1, h)
        probs, indices, aux_loss = self.router(tokens)

        expert_in, tokens_per_expert, state = self.token_dispatcher.dispatch(
            tokens, probs, indices)
        expert_out = self.experts(expert_in, tokens_per_expert)
        combined = self.token_dispatcher.combine(expert_out, state)

        if self.shared_experts is not None:
            shared, _ = self.shared_experts(tokens)
            combined = combined + shared

        output = combined.reshape(s, b, h).to(hidden_states.dtype)
        if aux_loss is not None:
            output = MoEAuxLossAutoScaler.apply(output, aux_loss)
        return output, None
