"""Inference KV-cache context (reference core/inference_params.py +
inference context of model_inference_wrappers)."""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch


class InferenceParams:
    """Per-request decode state: preallocated KV cache + sequence offset."""

    def __init__(self, max_batch_size: int, max_sequence_length: int):
        self.max_batch_size = max_batch_size
        self.max_sequence_length = max_sequence_length
        self.sequence_len_offset = 0
        self.batch_size_offset = 0
        self.key_value_memory_dict: Dict[int, Tuple[torch.Tensor, torch.Tensor]] = {}

    def reset(self):
        self.sequence_len_offset = 0
        self.key_value_memory_dict.clear()

    def update_kv_cache(self, layer_number: int, key: torch.Tensor,
                        value: torch.Tensor):
        """Append this step's K/V ([s_new, b, ng, hd]) and return the full
        prefix ([s_total, b, ng, hd]) for attention."""
        s_new, b, ng, hd = key.shape
        if layer_number not in self.key_value_memory_dict:
            k_cache = torch.empty(self.max_sequence_length, b, ng, hd,
                                  dtype=key.dtype, device=key.device)
            # v head dim may differ from k (MLA: v_head_dim != qk dims)
            v_cache = torch.empty(self.max_sequence_length, b, ng,
                                  value.shape[-1], dtype=value.dtype,
                                  device=value.device)
            self.key_value_memory_dict[layer_number] = (k_cache, v_cache)
        k_cache, v_cache = self.key_value_memory_dict[layer_number]
        start = self.sequence_len_offset
        end = start + s_new
        assert end <= self.max_sequence_length, "KV cache overflow"
        k_cache[start:end, :b].copy_(key)
        v_cache[start:end, :b].copy_(value)
        return k_cache[:end, :b], v_cache[:end, :b]

    def increment_sequence_len_offset(self, n: int):
        self.sequence_len_offset += n

    def reorder_batch(self, indices: torch.Tensor):
        """Permute the cache's batch dim (beam-search hypothesis
        reordering): cache[:, i] <- cache[:, indices[i]]."""
        end = self.sequence_len_offset
        for layer, (k, v) in self.key_value_memory_dict.items():
            k[:end] = k[:end].index_select(1, indices)
            v[:end] = v[:end].index_select(1, indices)


# alias matching newer reference naming
class StaticInferenceContext(InferenceParams):
    @classmethod
    def from_config(cls, config, max_batch_size, max_sequence_length):
        return cls(max_batch_size, max_sequence_length)
