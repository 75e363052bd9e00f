"""BiEncoder retrieval model for ICT / DPR-style pretraining (reference
megatron/legacy/model/biencoder_model.py:65-310, pretrain_ict.py).

Two BERT towers (optionally shared) embed queries and context blocks;
the embedding is the [CLS] position of the final hidden state with an
optional projection.  Retrieval trains with in-batch softmax over
query·context scores, all-gathered across the data-parallel group so
every rank scores against the global batch
(``AllgatherFromDataParallelRegion``, reference pretrain_ict.py:45-70).
"""

from __future__ import annotations

import torch

from ... import parallel_state
from ...transformer.module import MegatronModule
from ..bert import BertModel


class AllgatherFromDataParallelRegion(torch.autograd.Function):
    """All-gather rows across DP; backward returns this rank's slice."""

    @staticmethod
    def forward(ctx, input_):
        assert input_.dim() == 2
        group = parallel_state.get_data_parallel_group()
        world = torch.distributed.get_world_size(group=group)
        if world == 1:
            return input_
        tensors = [torch.empty_like(input_) for _ in range(world)]
        tensors[torch.distributed.get_rank(group=group)] = input_
        torch.distributed.all_gather(tensors, input_, group=group)
        return torch.cat(tensors, dim=0).contiguous()

    @staticmethod
    def backward(ctx, grad_output):
        group = parallel_state.get_data_parallel_group()
        world = torch.distributed.get_world_size(group=group)
        if world == 1:
            return grad_output
        rank = torch.distributed.get_rank(group=group)
        rows = grad_output.shape[0] // world
        return grad_output[rank * rows:(rank + 1) * rows].contiguous()


class PretrainedBertEncoder(MegatronModule):
    """BERT tower pooled at [CLS], with optional projection
    (reference biencoder_model.py:246-301)."""

    def __init__(self, config, transformer_layer_spec, vocab_size,
                 max_sequence_length, projection_dim: int = 0,
                 num_tokentypes: int = 2):
        super().__init__(config=config)
        self.projection_dim = projection_dim
        self.language_model = BertModel(
            config=config, transformer_layer_spec=transformer_layer_spec,
            vocab_size=vocab_size,
            max_sequence_length=max_sequence_length,
            num_tokentypes=num_tokentypes,
            add_binary_head=False, post_process=False)
        if projection_dim > 0:
            self.projection_enc = torch.nn.Linear(
                config.hidden_size, projection_dim,
                dtype=config.params_dtype)

    def forward(self, input_ids, attention_mask, tokentype_ids=None):
        hidden = self.language_model(
            input_ids, attention_mask, tokentype_ids=tokentype_ids)
        pooled = hidden[0, :, :]            # [CLS] position, [b, h]
        if self.projection_dim > 0:
            pooled = self.projection_enc(pooled)
        return pooled


class BiEncoderModel(MegatronModule):
    """Query + context towers (reference biencoder_model.py:65-135)."""

    def __init__(self, config, transformer_layer_spec, vocab_size,
                 max_sequence_length, projection_dim: int = 0,
                 num_tokentypes: int = 2,
                 only_query_model: bool = False,
                 only_context_model: bool = False,
                 shared_query_context_model: bool = False):
        super().__init__(config=config)
        assert not (only_query_model and only_context_model)
        self.shared_query_context_model = shared_query_context_model

        def mk():
            return PretrainedBertEncoder(
                config, transformer_layer_spec, vocab_size,
                max_sequence_length, projection_dim, num_tokentypes)

        if shared_query_context_model:
            self.shared_model = mk()
            self.query_model = self.context_model = self.shared_model
        else:
            self.query_model = mk() if not only_context_model else None
            self.context_model = mk() if not only_query_model else None

    def set_input_tensor(self, input_tensor):
        """Biencoder runs without pipeline parallelism; the schedule
        still calls this with None."""
        assert input_tensor is None or (
            isinstance(input_tensor, list) and input_tensor[0] is None), \
            "BiEncoderModel does not support pipeline parallelism"

    def embed_query(self, query_tokens, query_attention_mask,
                    query_types=None):
        assert self.query_model is not None
        return self.query_model(query_tokens, query_attention_mask,
                                query_types)

    def embed_context(self, context_tokens, context_attention_mask,
                      context_types=None):
        assert self.context_model is not None
        return self.context_model(context_tokens, context_attention_mask,
                                  context_types)

    def forward(self, query_tokens, query_attention_mask, query_types,
                context_tokens, context_attention_mask, context_types):
        return (self.embed_query(query_tokens, query_attention_mask,
                                 query_types),
                self.embed_context(context_tokens, context_attention_mask,
                                   context_types))


def biencoder_model_provider(config, transformer_layer_spec, vocab_size,
                             max_sequence_length, projection_dim=0,
                             only_query_model=False,
                             only_context_model=False,
                             shared_query_context_model=False):
    """reference biencoder_model.py biencoder_model_provider."""
    return BiEncoderModel(
        config=config, transformer_layer_spec=transformer_layer_spec,
        vocab_size=vocab_size, max_sequence_length=max_sequence_length,
        projection_dim=projection_dim,
        only_query_model=only_query_model,
        only_context_model=only_context_model,
        shared_query_context_model=shared_query_context_model)
