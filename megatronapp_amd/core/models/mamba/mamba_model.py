"""Mamba language model (reference core/models/mamba/mamba_model.py).

A stack of pre-norm Mamba mixer blocks (optionally interleaved with
standard attention layers via ``hybrid_attention_ratio``-style patterns in
the reference; here a pattern string of 'M' (mamba) and '*' (attention)
selects per-layer) over the shared embedding/output-head machinery.
"""

from __future__ import annotations

import torch

from ...transformer_config import TransformerConfig
from ...transformer.module import MegatronModule
from ...transformer.transformer_layer import TransformerLayer
from ...fusions.fused_layer_norm import FusedRMSNorm
from ...ssm.mamba_mixer import MambaMixer
from ..common.language_module import LanguageModule
from ..common.embeddings.language_model_embedding import (
    LanguageModelEmbedding)


class MambaBlock(MegatronModule):
    """norm -> mixer -> residual."""

    def __init__(self, config, layer_number, attention_spec=None):
        super().__init__(config)
        self.norm = FusedRMSNorm(config=config,
                                 hidden_size=config.hidden_size,
                                 eps=config.layernorm_epsilon)
        if attention_spec is not None:
            from ...transformer.spec_utils import build_module
            self.mixer = None
            self.attn_layer = build_module(attention_spec, config=config,
                                           layer_number=layer_number)
        else:
            self.mixer = MambaMixer(config, layer_number=layer_number)
            self.attn_layer = None

    def forward(self, hidden_states, attention_mask=None,
                inference_context=None, **kwargs):
        if self.attn_layer is not None:
            out, _ = self.attn_layer(hidden_states,
                                     attention_mask=attention_mask,
                                     inference_context=inference_context)
            return out
        residual = hidden_states
        out, bias = self.mixer(self.norm(hidden_states),
                               inference_context=inference_context)
        if bias is not None:
            out = out + bias
        return residual + out


class MambaModel(LanguageModule):
    def __init__(self, config: TransformerConfig, vocab_size: int,
                 max_sequence_length: int, pre_process: bool = True,
                 post_process: bool = True, hybrid_pattern: str = None,
                 attention_spec=None, fp16_lm_cross_entropy: bool = False,
                 parallel_output: bool = True,
                 share_embeddings_and_output_weights: bool = True):
        super().__init__(config)
        self.pre_process = pre_process
        self.post_process = post_process
        self.vocab_size = vocab_size
        self.parallel_output = parallel_output
        self.share_embeddings_and_output_weights = \
            share_embeddings_and_output_weights

        if pre_process:
            self.embedding = LanguageModelEmbedding(
                config=config, vocab_size=vocab_size,
                max_sequence_length=max_sequence_length,
                position_embedding_type="none")

        pattern = hybrid_pattern or "M" * config.num_layers
        assert len(pattern) == config.num_layers
        self.layers = torch.nn.ModuleList([
            MambaBlock(config, i + 1,
                       attention_spec if c == "*" else None)
            for i, c in enumerate(pattern)])
        self.final_norm = FusedRMSNorm(config=config,
                                       hidden_size=config.hidden_size,
                                       eps=config.layernorm_epsilon)
        if post_process:
            from ...tensor_parallel.layers import ColumnParallelLinear
            self.output_layer = ColumnParallelLinear(
                config.hidden_size, vocab_size, config=config,
                init_method=config.init_method, bias=False,
                skip_bias_add=False, gather_output=not parallel_output,
                skip_weight_param_allocation=pre_process and
                share_embeddings_and_output_weights)
        if pre_process or post_process:
            self.setup_embeddings_and_output_layer()

    def set_input_tensor(self, input_tensor):
        # PP is not sliced through Mamba stacks yet (reference also runs
        # them PP=1 unless hybrid); accept the API for schedule parity
        if isinstance(input_tensor, list):
            input_tensor = input_tensor[0]
        self._input_tensor = input_tensor

    def forward(self, input_ids, position_ids=None, attention_mask=None,
                labels=None, inference_context=None, **kwargs):
        hidden = self.embedding(input_ids, position_ids)
        for layer in self.layers:
            hidden = layer(hidden, attention_mask=attention_mask,
                           inference_context=inference_context)
        hidden = self.final_norm(hidden)
        output_weight = None
        if self.share_embeddings_and_output_weights:
            output_weight = self.shared_embedding_or_output_weight()
        logits, _ = self.output_layer(hidden, weight=output_weight)
        if labels is None:
            return logits.transpose(0, 1).contiguous()
        return self.compute_language_model_loss(labels, logits)
