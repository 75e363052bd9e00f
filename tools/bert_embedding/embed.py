#!/usr/bin/env python3
"""BERT embedding extraction (reference tools/bert_embedding/embed.py).

Embeds token chunks (or byte-tokenized text) with a BERT tower,
mean-pooling the final hidden states over non-pad positions, and writes
``.npy`` shards (the reference writes HDF5; h5py is not in this image).
Used standalone and by the retro preprocessing pipeline
(tools/retro/preprocess.py).
"""

from __future__ import annotations

import numpy as np
import torch


class BertEmbedder:
    """Mean-pooled BERT chunk embedder."""

    def __init__(self, config, vocab_size: int, max_sequence_length: int,
                 load_path: str = None, device: str = None):
        from megatronapp_amd.core.models.bert import BertModel
        from megatronapp_amd.core.models.bert.bert_layer_specs import (
            get_bert_layer_local_spec)
        self.device = device or (
            "cuda" if torch.cuda.is_available() else "cpu")
        self.model = BertModel(
            config=config,
            transformer_layer_spec=get_bert_layer_local_spec(),
            vocab_size=vocab_size,
            max_sequence_length=max_sequence_length,
            add_binary_head=False, post_process=False,
        ).to(self.device).eval()
        if load_path:
            sd = torch.load(load_path, map_location="cpu",
                            weights_only=False)
            self.model.load_state_dict(sd.get("model", sd), strict=False)

    @torch.no_grad()
    def embed_tokens(self, token_chunks: np.ndarray, pad_id: int,
                     batch_size: int = 64) -> np.ndarray:
        """[n, chunk_len] int tokens -> [n, hidden] fp32 embeddings."""
        outs = []
        for i in range(0, len(token_chunks), batch_size):
            ids = torch.as_tensor(
                np.ascontiguousarray(token_chunks[i:i + batch_size]),
                dtype=torch.long, device=self.device)
            mask = (ids != pad_id).float()
            hidden = self.model(ids, mask)          # [s, b, h]
            hidden = hidden.transpose(0, 1).float()  # [b, s, h]
            denom = mask.sum(1, keepdim=True).clamp(min=1)
            pooled = (hidden * mask.unsqueeze(-1)).sum(1) / denom
            outs.append(pooled.cpu().numpy())
        return np.concatenate(outs, axis=0)

    def embed_texts(self, texts, tokenizer, seq_length: int,
                    batch_size: int = 64) -> np.ndarray:
        pad = tokenizer.pad
        chunks = np.full((len(texts), seq_length), pad, dtype=np.int64)
        for i, t in enumerate(texts):
            ids = tokenizer.tokenize(t)[:seq_length]
            chunks[i, :len(ids)] = ids
        return self.embed_tokens(chunks, pad, batch_size)
