"""Vision→language projector (reference:
core/models/vision/multimodal_projector.py:9-74).

'mlp' = the standard two-layer MLP with an ``input_size`` equal to the
vision hidden size; 'affine' = a single column-parallel linear.  Output
is always bias-added (the reference folds skip_bias_add back in).
"""

from __future__ import annotations

from ...tensor_parallel.layers import ColumnParallelLinear
from ...transformer.mlp import MLP, MLPSubmodules
from ...transformer.module import MegatronModule
from ...transformer_config import TransformerConfig


class MultimodalProjector(MegatronModule):
    def __init__(self, config: TransformerConfig, submodules: MLPSubmodules,
                 projector_type: str, input_size: int):
        super().__init__(config=config)
        self.projector_type = projector_type
        if projector_type == "mlp":
            self.encoder = MLP(config=config, submodules=submodules,
                               input_size=input_size)
        elif projector_type == "affine":
            self.encoder = ColumnParallelLinear(
                input_size, config.hidden_size, config=config,
                init_method=config.init_method, gather_output=True,
                bias=config.add_bias_linear, skip_bias_add=True)
        else:
            raise ValueError(
                f"unsupported multimodal projector type {projector_type}")

    def forward(self, hidden_states):
        out, bias = self.encoder(hidden_states)
        if bias is not None:
            out = out + bias
        return out
