"""Per-slot KV-cache context for continuous batching (reference
core/inference/engines/dynamic_engine.py's context).

Unlike the static :class:`InferenceParams` (one shared sequence offset),
every slot has its own length: decode steps scatter each row's new K/V
at that row's position, attention reads the union window with a per-row
padding mask, and RoPE positions are per-row.  Slots join (prefill) and
retire independently while the decode batch keeps running.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import torch


class DynamicInferenceContext:
    is_dynamic = True
    is_graph_context = False

    def __init__(self, max_slots: int, max_sequence_length: int):
        self.max_slots = max_slots
        self.max_sequence_length = max_sequence_length
        self.key_value_memory_dict: Dict[int, Tuple[torch.Tensor,
                                                    torch.Tensor]] = {}
        self.host_lengths = [0] * max_slots     # tokens already cached
        self.active: List[int] = []             # slot ids in batch order
        self.prefill = False                    # current call is a prefill
        self._device = None
        self.sequence_len_offset = 0            # 0 for fresh-slot prefill

    # ------------------------------------------------------------- slots
    def free_slot(self):
        for s in range(self.max_slots):
            if self.host_lengths[s] == 0:
                return s
        return None

    def release(self, slot: int):
        self.host_lengths[slot] = 0

    def set_active(self, slots: List[int], prefill: bool):
        self.active = list(slots)
        self.prefill = prefill

    def advance(self, n: int):
        for s in self.active:
            self.host_lengths[s] += n

    # ------------------------------------------------------------ lengths
    def row_positions(self) -> torch.Tensor:
        return torch.tensor([self.host_lengths[s] for s in self.active],
                            dtype=torch.long, device=self._device)

    def window(self) -> int:
        return max(self.host_lengths[s] for s in self.active) + 1

    def decode_padding_mask(self) -> torch.Tensor:
        """[b, 1, 1, window] True = masked (beyond this row's prefix+new
        token)."""
        w = self.window()
        lens = self.row_positions() + 1    # new token included
        ar = torch.arange(w, device=self._device)
        return (ar.unsqueeze(0) >= lens.unsqueeze(1)).view(
            len(self.active), 1, 1, w)

    # ----------------------------------------------------------- kv cache
    def _cache(self, layer: int, key: torch.Tensor, value: torch.Tensor):
        if layer not in self.key_value_memory_dict:
            k = torch.empty(self.max_sequence_length, self.max_slots,
                            key.shape[2], key.shape[3], dtype=key.dtype,
                            device=key.device)
            v = torch.empty(self.max_sequence_length, self.max_slots,
                            value.shape[2], value.shape[3],
                            dtype=value.dtype, device=value.device)
            self.key_value_memory_dict[layer] = (k, v)
        return self.key_value_memory_dict[layer]

    def update_kv_cache(self, layer: int, key: torch.Tensor,
                        value: torch.Tensor):
        """key/value [s_new, b_active, ng, hd] -> window views."""
        self._device = key.device
        kc, vc = self._cache(layer, key, value)
        s_new, b = key.shape[0], key.shape[1]
        if self.prefill:
            assert b == 1
            sid = self.active[0]
            start = self.host_lengths[sid]
            kc[start:start + s_new, sid] = key[:, 0]
            vc[start:start + s_new, sid] = value[:, 0]
            end = start + s_new
            return kc[:end, sid:sid + 1], vc[:end, sid:sid + 1]
        act = torch.tensor(self.active, dtype=torch.long, device=key.device)
        pos = self.row_positions()
        kc[pos, act] = key[0]
        vc[pos, act] = value[0]
        w = self.window()
        return (kc[:w].index_select(1, act), vc[:w].index_select(1, act))
