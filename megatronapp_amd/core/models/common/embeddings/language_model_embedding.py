"""Word + positional embedding stage (reference language_model_embedding)."""

from __future__ import annotations

import torch
from torch import nn

from .... import parallel_state
from ....tensor_parallel.layers import VocabParallelEmbedding
from ....tensor_parallel.mappings import scatter_to_sequence_parallel_region
from ....transformer_config import TransformerConfig
from ....transformer.module import MegatronModule


class LanguageModelEmbedding(MegatronModule):
    def __init__(self, config: TransformerConfig, vocab_size: int,
                 max_sequence_length: int,
                 position_embedding_type: str = "learned_absolute",
                 num_tokentypes: int = 0,
                 scatter_to_sequence_parallel: bool = True):
        super().__init__(config)
        self.vocab_size = vocab_size
        self.max_sequence_length = max_sequence_length
        self.add_position_embedding = position_embedding_type == "learned_absolute"
        self.num_tokentypes = num_tokentypes
        self.reduce_scatter_embeddings = (
            scatter_to_sequence_parallel and config.sequence_parallel
            and not self.add_position_embedding)

        self.word_embeddings = VocabParallelEmbedding(
            vocab_size, config.hidden_size, init_method=config.init_method,
            config=config,
            reduce_scatter_embeddings=self.reduce_scatter_embeddings)

        if self.add_position_embedding:
            self.position_embeddings = nn.Embedding(
                max_sequence_length, config.hidden_size,
                dtype=config.params_dtype)
            config.init_method(self.position_embeddings.weight)
        if num_tokentypes > 0:
            self.tokentype_embeddings = nn.Embedding(
                num_tokentypes, config.hidden_size, dtype=config.params_dtype)
            config.init_method(self.tokentype_embeddings.weight)
        else:
            self.tokentype_embeddings = None
        self.embedding_dropout = nn.Dropout(config.hidden_dropout)

    def forward(self, input_ids, position_ids, tokentype_ids=None):
        word_embeddings = self.word_embeddings(input_ids)  # [b, s, h] or [s/tp, b, h]
        if self.reduce_scatter_embeddings:
            embeddings = word_embeddings
        else:
            embeddings = word_embeddings.transpose(0, 1).contiguous()  # [s, b, h]
        if self.add_position_embedding:
            # out-of-range lookups are an async DEVICE FAULT (core dump
            # with no traceback) — fail with a real error instead.  The
            # check is a device->host sync, which is forbidden while a
            # hipGraph is being captured (decode capture passes fixed
            # in-range position buffers, so skipping it there is safe).
            capturing = (position_ids.is_cuda
                         and torch.cuda.is_current_stream_capturing())
            if not capturing and position_ids.numel() and \
                    int(position_ids.max()) >= \
                    self.position_embeddings.num_embeddings:
                raise ValueError(
                    f"position id {int(position_ids.max())} exceeds "
                    f"max_sequence_length "
                    f"{self.position_embeddings.num_embeddings}")
            embeddings = embeddings + self.position_embeddings(
                position_ids).transpose(0, 1)
        if tokentype_ids is not None and self.tokentype_embeddings is not None:
            embeddings = embeddings + self.tokentype_embeddings(
                tokentype_ids).permute(1, 0, 2)
        if self.config.fp32_residual_connection:
            embeddings = embeddings.float()
        if self.config.sequence_parallel and not self.reduce_scatter_embeddings:
            embeddings = scatter_to_sequence_parallel_region(embeddings)
        embeddings = self.embedding_dropout(embeddings)
        return embeddings
