"""Retro config (reference core/models/retro/config.py:14-88).

Retrieval-augmented GPT: ``retro_retrieved_length`` is derived as
``retro_num_retrieved_chunks * retro_chunk_length`` (neighbor +
continuation tokens).  The reference's TE-env and preprocessing-split
validations do not apply here.
"""

from __future__ import annotations

from dataclasses import dataclass

from ...transformer_config import TransformerConfig


@dataclass
class RetroConfig(TransformerConfig):
    retro_project_dir: str = None
    retro_block_size: int = None
    retro_chunk_length: int = 64
    retro_encoder_num_layers: int = 2
    retro_encoder_hidden_dropout: float = 0.1
    retro_encoder_attention_dropout: float = 0.1
    retro_neighbor_dirs: dict = None
    retro_num_neighbors: int = 2
    retro_num_retrieved_chunks: int = 2
    retro_retrieved_length: int = None
    retro_split_preprocessing: str = None
    retro_verify_neighbor_count: bool = True

    def __post_init__(self):
        super().__post_init__()
        self.retro_retrieved_length = (
            self.retro_num_retrieved_chunks * self.retro_chunk_length)
