"""RetroModel (reference core/models/retro/model.py:20-107): GPT whose
forward additionally embeds the retrieved-neighbor token ids and threads
them to the chunked cross-attention layers as the block ``context``."""

from __future__ import annotations

from typing import Optional

from torch import Tensor

from ..gpt import GPTModel


class RetroModel(GPTModel):
    def forward(self, input_ids: Tensor, position_ids: Tensor,
                attention_mask: Optional[Tensor] = None,
                context_input_ids: Optional[Tensor] = None,
                context_position_ids: Optional[Tensor] = None,
                context_mask: Optional[Tensor] = None,
                decoder_input: Optional[Tensor] = None,
                labels: Optional[Tensor] = None,
                inference_context=None):
        # context_input_ids: [k*bs*l, r] -> context [r, k*bs*l, d]
        if context_input_ids is not None:
            context = self.embedding(context_input_ids,
                                     context_position_ids)
        else:
            context = None
        return super().forward(
            input_ids=input_ids, position_ids=position_ids,
            attention_mask=attention_mask, decoder_input=decoder_input,
            labels=labels, inference_context=inference_context,
            extra_block_kwargs={"context": context,
                                "context_mask": context_mask})
