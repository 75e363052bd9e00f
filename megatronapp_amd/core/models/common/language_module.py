"""LanguageModule: shared embedding/output-weight handling + LM loss.

Reference: models/common/language_module/language_module.py (tied
embeddings, grad-sync setup across the embedding group).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ... import parallel_state
from ...tensor_parallel.cross_entropy import vocab_parallel_cross_entropy
from ...transformer.module import MegatronModule
from ...transformer_config import TransformerConfig


class LanguageModule(MegatronModule):
    def __init__(self, config: TransformerConfig):
        super().__init__(config)

    def compute_language_model_loss(self, labels, logits) -> torch.Tensor:
        """labels [b, s]; logits [s, b, v/tp] -> loss [b, s]."""
        labels = labels.transpose(0, 1).contiguous()  # [s, b]
        loss = vocab_parallel_cross_entropy(logits.contiguous(), labels)
        return loss.transpose(0, 1).contiguous()

    def setup_embeddings_and_output_layer(self) -> None:
        """Mark shared embedding params; zero-init the duplicate so the
        first embedding-group all-reduce equalizes them (reference
        language_module.py)."""
        if self.pre_process:
            self.embedding.word_embeddings.weight.is_embedding_or_output_parameter = True
        if self.post_process and getattr(self, "output_layer", None) is not None \
                and self.output_layer.weight is not None:
            self.output_layer.weight.is_embedding_or_output_parameter = True

        if not self.share_embeddings_and_output_weights:
            return
        if parallel_state.get_pipeline_model_parallel_world_size() == 1:
            if self.pre_process and self.post_process:
                self.shared_embedding_or_output_weight().shared_embedding = True
            return

        if self.pre_process:
            self.shared_embedding_or_output_weight().shared_embedding = True
        if self.post_process and not self.pre_process:
            # last stage owns a zero-initialised copy; synced below
            self.output_layer.weight.data.fill_(0)
            self.output_layer.weight.shared = True
            self.output_layer.weight.shared_embedding = True

        # initial sync so both copies start identical
        if dist.is_initialized() and parallel_state.is_rank_in_embedding_group():
            weight = self.shared_embedding_or_output_weight()
            if weight is not None:
                dist.all_reduce(weight.data,
                                group=parallel_state.get_embedding_group())

    def shared_embedding_or_output_weight(self):
        if self.pre_process:
            return self.embedding.word_embeddings.weight
        if self.post_process and getattr(self, "output_layer", None) is not None:
            return self.output_layer.weight
        return None
