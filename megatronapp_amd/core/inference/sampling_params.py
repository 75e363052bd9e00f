"""Sampling parameters (reference core/inference/sampling_params.py)."""

from __future__ import annotations

from dataclasses import dataclass


@dataclass
class SamplingParams:
    num_tokens_to_generate: int = 64
    temperature: float = 1.0
    top_k: int = 0
    top_p: float = 0.0
    return_log_probs: bool = False
    add_BOS: bool = False
    termination_id: int = -1
    prevent_newline_after_colon: bool = False
    # beam search (reference text_generation_server.py:243-267): > 0
    # switches generation to beam decoding with this many hypotheses
    beam_width: int = 0
    length_penalty: float = 1.0
