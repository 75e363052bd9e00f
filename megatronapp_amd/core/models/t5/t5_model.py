"""T5 encoder-decoder model (reference core/models/T5/t5_model.py, 523
LoC): bidirectional encoder, causal decoder with cross-attention to the
encoder output, shared embeddings, LM head over the decoder."""

from __future__ import annotations

import torch

from ...enums import AttnMaskType
from ...tensor_parallel.layers import ColumnParallelLinear
from ...transformer.transformer_block import TransformerBlock
from ...transformer_config import TransformerConfig
from ..common.embeddings.language_model_embedding import LanguageModelEmbedding
from ..common.language_module import LanguageModule


class T5Model(LanguageModule):
    def __init__(self, config: TransformerConfig, encoder_layer_spec,
                 decoder_layer_spec, vocab_size: int,
                 max_sequence_length: int, pre_process: bool = True,
                 post_process: bool = True,
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
        self.position_embedding_type = position_embedding_type

        self.embedding = LanguageModelEmbedding(
            config=config, vocab_size=vocab_size,
            max_sequence_length=max_sequence_length,
            position_embedding_type=position_embedding_type)
        self.encoder = TransformerBlock(config=config, spec=encoder_layer_spec)
        self.decoder = TransformerBlock(config=config, spec=decoder_layer_spec)
        self.output_layer = ColumnParallelLinear(
            config.hidden_size, vocab_size, config=config,
            init_method=config.init_method, bias=False, skip_bias_add=False,
            gather_output=not parallel_output,
            skip_weight_param_allocation=share_embeddings_and_output_weights)
        self.setup_embeddings_and_output_layer()

    def set_input_tensor(self, input_tensor):
        if isinstance(input_tensor, list):
            input_tensor = input_tensor[0]
        self.encoder.set_input_tensor(input_tensor)

    @staticmethod
    def _padding_mask(mask_1d):
        keep = mask_1d.bool()
        return (~(keep.unsqueeze(1) & keep.unsqueeze(2))).unsqueeze(1)

    @staticmethod
    def _cross_mask(dec_len, enc_mask_1d):
        keep = enc_mask_1d.bool().unsqueeze(1).expand(-1, dec_len, -1)
        return (~keep).unsqueeze(1)

    def forward(self, encoder_input_ids, decoder_input_ids, encoder_attn_mask,
                decoder_attn_mask=None, lm_labels=None, inference_context=None):
        b, s_enc = encoder_input_ids.shape
        s_dec = decoder_input_ids.shape[1]
        device = encoder_input_ids.device
        enc_pos = torch.arange(s_enc, device=device).unsqueeze(0).expand(b, -1)
        dec_pos = torch.arange(s_dec, device=device).unsqueeze(0).expand(b, -1)

        enc_hidden = self.embedding(encoder_input_ids, enc_pos)
        enc_out = self.encoder(
            enc_hidden, attention_mask=self._padding_mask(encoder_attn_mask))

        dec_hidden = self.embedding(decoder_input_ids, dec_pos)
        dec_out = self.decoder(
            dec_hidden, attention_mask=None,   # causal (spec default)
            context=enc_out,
            context_mask=self._cross_mask(s_dec, encoder_attn_mask))

        output_weight = None
        if self.share_embeddings_and_output_weights:
            output_weight = self.shared_embedding_or_output_weight()
        logits, _ = self.output_layer(dec_out, weight=output_weight)
        if lm_labels is None:
            return logits.transpose(0, 1).contiguous()
        return self.compute_language_model_loss(lm_labels, logits)
