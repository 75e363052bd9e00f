"""GPTModel: embedding -> rotary -> decoder block -> output layer -> loss.

Reference: models/gpt/gpt_model.py:32 (forward :235).  MegaScan wraps the
decoder in a "decoder" scope (reference gpt_model.py:333-340).
"""

from __future__ import annotations

from typing import Literal, Optional

import torch

from ... import parallel_state
from ...tensor_parallel.layers import ColumnParallelLinear
from ...transformer.spec_utils import ModuleSpec
from ...transformer.transformer_block import TransformerBlock
from ...transformer_config import TransformerConfig
from ...trace_hooks import trace_scope
from ..common.embeddings.language_model_embedding import LanguageModelEmbedding
from ..common.embeddings.rotary_pos_embedding import RotaryEmbedding
from ..common.language_module import LanguageModule


class GPTModel(LanguageModule):
    def __init__(self, config: TransformerConfig, transformer_layer_spec,
                 vocab_size: int, max_sequence_length: int,
                 pre_process: bool = True, post_process: bool = True,
                 fp16_lm_cross_entropy: bool = False, parallel_output: bool = True,
                 share_embeddings_and_output_weights: bool = False,
                 position_embedding_type: Literal["learned_absolute", "rope", "none"] = "learned_absolute",
                 rotary_percent: float = 1.0, rotary_base: int = 10000,
                 seq_len_interpolation_factor: Optional[float] = None,
                 vp_stage: Optional[int] = None):
        super().__init__(config=config)
        self.transformer_layer_spec = transformer_layer_spec
        self.vocab_size = vocab_size
        self.max_sequence_length = max_sequence_length
        self.pre_process = pre_process
        self.post_process = post_process
        self.fp16_lm_cross_entropy = fp16_lm_cross_entropy
        self.parallel_output = parallel_output
        self.share_embeddings_and_output_weights = share_embeddings_and_output_weights
        self.position_embedding_type = position_embedding_type
        self.vp_stage = vp_stage
        self.model_type = "encoder_or_decoder"

        if self.pre_process:
            self.embedding = LanguageModelEmbedding(
                config=config, vocab_size=vocab_size,
                max_sequence_length=max_sequence_length,
                position_embedding_type=position_embedding_type)

        if position_embedding_type == "rope":
            self.rotary_pos_emb = RotaryEmbedding(
                kv_channels=config.kv_channels, rotary_percent=rotary_percent,
                rotary_base=rotary_base,
                seq_len_interpolation_factor=seq_len_interpolation_factor)

        self.decoder = TransformerBlock(
            config=config, spec=transformer_layer_spec,
            pre_process=self.pre_process, post_process=self.post_process,
            vp_stage=vp_stage)

        self.mtp = None
        if (self.post_process and self.pre_process
                and getattr(config, "mtp_num_layers", None)):
            from ...transformer.multi_token_prediction import (
                MultiTokenPredictionBlock, get_gpt_mtp_block_spec)
            self.mtp = MultiTokenPredictionBlock(
                config, get_gpt_mtp_block_spec(config, transformer_layer_spec))

        if self.post_process:
            self.output_layer = ColumnParallelLinear(
                config.hidden_size, vocab_size, config=config,
                init_method=config.init_method, bias=False,
                skip_bias_add=False, gather_output=not parallel_output,
                skip_weight_param_allocation=self.pre_process and
                share_embeddings_and_output_weights)
            if share_embeddings_and_output_weights and not self.pre_process:
                pass  # weight allocated above; zeroed+synced in setup below

        if self.pre_process or self.post_process:
            self.setup_embeddings_and_output_layer()

    def set_input_tensor(self, input_tensor) -> None:
        if isinstance(input_tensor, list):
            input_tensor = input_tensor[0]
        self.decoder.set_input_tensor(input_tensor)

    def forward(self, input_ids, position_ids, attention_mask=None,
                decoder_input=None, labels=None, inference_context=None,
                packed_seq_params=None, extra_block_kwargs=None,
                runtime_gather_output=None, loss_mask=None):
        # hidden layout: [s, b, h]
        if decoder_input is not None:
            pass
        elif self.pre_process:
            decoder_input = self.embedding(input_ids, position_ids)
        else:
            decoder_input = None  # comes from set_input_tensor

        rotary_pos_emb = None
        if self.position_embedding_type == "rope":
            rope_input = decoder_input
            if rope_input is None:
                rope_input = self.decoder.input_tensor  # mid-pipeline stage
            rotary_seq_len = self.rotary_pos_emb.get_rotary_seq_len(
                inference_context, self.decoder, rope_input, self.config,
                packed_seq_params)
            rotary_pos_emb = self.rotary_pos_emb(rotary_seq_len)

        with trace_scope("decoder"):
            hidden_states = self.decoder(
                decoder_input, attention_mask=attention_mask,
                rotary_pos_emb=rotary_pos_emb,
                inference_context=inference_context,
                packed_seq_params=packed_seq_params,
                **(extra_block_kwargs or {}))

        if not self.post_process:
            return hidden_states

        output_weight = None
        if self.share_embeddings_and_output_weights:
            output_weight = self.shared_embedding_or_output_weight()

        if (self.mtp is not None and labels is not None
                and inference_context is None):
            hidden_states = self.mtp(
                input_ids, position_ids, hidden_states,
                attention_mask=attention_mask, rotary_pos_emb=rotary_pos_emb,
                embedding=self.embedding, output_layer=self.output_layer,
                output_weight=output_weight,
                compute_loss=self.compute_language_model_loss, labels=labels,
                loss_mask=loss_mask)

        logits, _ = self.output_layer(hidden_states, weight=output_weight)

        if labels is None:
            return logits.transpose(0, 1).contiguous()  # [b, s, v/tp]
        loss = self.compute_language_model_loss(labels, logits)
        return loss

    def sharded_state_dict(self, prefix: str = "", sharded_offsets=(), metadata=None):
        from ...dist_checkpointing.mapping import module_sharded_state_dict
        return module_sharded_state_dict(self, prefix)
