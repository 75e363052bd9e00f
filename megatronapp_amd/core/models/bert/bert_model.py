"""BERT model (reference core/models/bert/bert_model.py, 382 LoC).

Bidirectional encoder over the same TransformerBlock, with padding-mask
attention, token-type embeddings, an MLM language-model head and the
binary (NSP) head.
"""

from __future__ import annotations

from typing import Literal, Optional

import torch
from torch import nn

from ... import parallel_state
from ...enums import AttnMaskType
from ...tensor_parallel.layers import ColumnParallelLinear
from ...transformer.spec_utils import ModuleSpec
from ...transformer.transformer_block import TransformerBlock
from ...transformer_config import TransformerConfig
from ..common.embeddings.language_model_embedding import LanguageModelEmbedding
from ..common.language_module import LanguageModule
from ...fusions.fused_layer_norm import FusedLayerNorm


class BertLMHead(nn.Module):
    """dense -> gelu -> layernorm before the (tied) output projection."""

    def __init__(self, hidden_size: int, config: TransformerConfig):
        super().__init__()
        self.dense = nn.Linear(hidden_size, hidden_size,
                               dtype=config.params_dtype)
        config.init_method(self.dense.weight)
        self.layer_norm = FusedLayerNorm(config, hidden_size,
                                         eps=config.layernorm_epsilon)

    def forward(self, hidden_states):
        h = self.dense(hidden_states)
        h = torch.nn.functional.gelu(h.float()).to(hidden_states.dtype)
        return self.layer_norm(h)


class Pooler(nn.Module):
    def __init__(self, hidden_size: int, config: TransformerConfig):
        super().__init__()
        self.dense = nn.Linear(hidden_size, hidden_size,
                               dtype=config.params_dtype)
        config.init_method(self.dense.weight)

    def forward(self, hidden_states):
        # hidden: [s, b, h] -> pooled [b, h] from the first token
        return torch.tanh(self.dense(hidden_states[0]))


class BertModel(LanguageModule):
    def __init__(self, config: TransformerConfig, transformer_layer_spec,
                 vocab_size: int, max_sequence_length: int,
                 pre_process: bool = True, post_process: bool = True,
                 num_tokentypes: int = 2, add_binary_head: bool = True,
                 share_embeddings_and_output_weights: bool = True,
                 parallel_output: bool = True,
                 position_embedding_type: str = "learned_absolute",
                 vp_stage=None):
        super().__init__(config)
        self.vocab_size = vocab_size
        self.pre_process = pre_process
        self.post_process = post_process
        self.parallel_output = parallel_output
        self.share_embeddings_and_output_weights = share_embeddings_and_output_weights
        self.add_binary_head = add_binary_head
        self.position_embedding_type = position_embedding_type

        if pre_process:
            self.embedding = LanguageModelEmbedding(
                config=config, vocab_size=vocab_size,
                max_sequence_length=max_sequence_length,
                position_embedding_type=position_embedding_type,
                num_tokentypes=num_tokentypes)

        self.encoder = TransformerBlock(
            config=config, spec=transformer_layer_spec,
            pre_process=pre_process, post_process=post_process)

        if post_process:
            self.lm_head = BertLMHead(config.hidden_size, config)
            self.output_layer = ColumnParallelLinear(
                config.hidden_size, vocab_size, config=config,
                init_method=config.init_method, bias=True,
                skip_bias_add=False, gather_output=not parallel_output,
                skip_weight_param_allocation=pre_process and
                share_embeddings_and_output_weights)
            if add_binary_head:
                self.pooler = Pooler(config.hidden_size, config)
                self.binary_head = nn.Linear(config.hidden_size, 2,
                                             dtype=config.params_dtype)
        if pre_process or post_process:
            self.setup_embeddings_and_output_layer()

    def set_input_tensor(self, input_tensor):
        if isinstance(input_tensor, list):
            input_tensor = input_tensor[0]
        self.encoder.set_input_tensor(input_tensor)

    @staticmethod
    def _build_padding_mask(attention_mask_1d: torch.Tensor):
        """[b, s] 1=keep -> bool mask [b, 1, s, s] True=masked."""
        b, s = attention_mask_1d.shape
        keep = attention_mask_1d.bool()
        mask2d = keep.unsqueeze(1) & keep.unsqueeze(2)     # [b, s, s]
        return (~mask2d).unsqueeze(1)

    def forward(self, input_ids, attention_mask, tokentype_ids=None,
                lm_labels=None, inference_context=None):
        ext_mask = self._build_padding_mask(attention_mask)
        position_ids = torch.arange(
            input_ids.size(1), device=input_ids.device).unsqueeze(0).expand_as(
                input_ids)
        if self.pre_process:
            hidden = self.embedding(input_ids, position_ids,
                                    tokentype_ids=tokentype_ids)
        else:
            hidden = None
        hidden = self.encoder(hidden, attention_mask=ext_mask)
        if not self.post_process:
            return hidden

        binary_logits = None
        if self.add_binary_head:
            binary_logits = self.binary_head(self.pooler(hidden))

        hidden = self.lm_head(hidden)
        output_weight = None
        if self.share_embeddings_and_output_weights:
            output_weight = self.shared_embedding_or_output_weight()
        logits, _ = self.output_layer(hidden, weight=output_weight)

        if lm_labels is None:
            return logits.transpose(0, 1).contiguous(), binary_logits
        loss = self.compute_language_model_loss(lm_labels, logits)
        return loss, binary_logits
