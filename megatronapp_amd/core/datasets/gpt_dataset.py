"""GPT datasets: MockGPTDataset (synthetic, the bring-up/bench path) and
a token-stream GPTDataset over a memory-mapped numpy token file.

Reference: core/datasets/gpt_dataset.py (GPTDataset:811, MockGPTDataset).
The reference's .bin/.idx IndexedDataset + C++ sample-index builder is
covered by indexed_dataset.py / helpers in this package; the mock dataset
reproduces the reference's role exactly: deterministic per-sample tokens,
causal labels, loss mask, position ids, attention mask.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import numpy as np
import torch


@dataclass
class GPTDatasetConfig:
    random_seed: int = 1234
    sequence_length: int = 1024
    reset_position_ids: bool = False
    reset_attention_mask: bool = False
    eod_mask_loss: bool = False
    create_attention_mask: bool = False
    vocab_size: int = 51200
    blend: Optional[list] = None
    split: Optional[str] = None
    path_to_cache: Optional[str] = None
    tokenizer: Optional[object] = None
    mock: bool = True


def _build_sample(tokens: np.ndarray, config: GPTDatasetConfig):
    seq = config.sequence_length
    tokens_t = torch.from_numpy(tokens.astype(np.int64))
    labels = tokens_t[1:].contiguous()
    tokens_t = tokens_t[:-1].contiguous()
    loss_mask = torch.ones(seq, dtype=torch.float)
    position_ids = torch.arange(seq, dtype=torch.int64)
    sample = {
        "tokens": tokens_t,
        "labels": labels,
        "loss_mask": loss_mask,
        "position_ids": position_ids,
    }
    if config.create_attention_mask:
        att = torch.tril(torch.ones((seq, seq), dtype=torch.bool)).unsqueeze(0)
        sample["attention_mask"] = ~att
    return sample


class MockGPTDataset(torch.utils.data.Dataset):
    """Deterministic synthetic tokens — the data-free bring-up dataset."""

    def __init__(self, config: GPTDatasetConfig, num_samples: int = 1 << 20,
                 name: str = "mock"):
        self.config = config
        self.num_samples = num_samples
        self.name = name

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        rng = np.random.default_rng(self.config.random_seed + int(idx))
        tokens = rng.integers(0, self.config.vocab_size,
                              self.config.sequence_length + 1, dtype=np.int64)
        return _build_sample(tokens, self.config)


class GPTDataset(torch.utils.data.Dataset):
    """Causal-LM windows over a flat token stream (np.memmap .npy/.bin)."""

    def __init__(self, config: GPTDatasetConfig, token_file: str,
                 num_samples: Optional[int] = None, name: str = "gpt"):
        self.config = config
        self.name = name
        if token_file.endswith(".npy"):
            self.tokens = np.load(token_file, mmap_mode="r")
        elif os.path.exists(token_file + ".idx"):
            # reference-format indexed dataset: the .bin is a flat token
            # stream; document boundaries live in the .idx
            from .indexed_dataset import IndexedDataset
            self.tokens = IndexedDataset(token_file).bin
        else:
            self.tokens = np.memmap(token_file, dtype=np.int32, mode="r")
        max_samples = (len(self.tokens) - 1) // config.sequence_length
        self.num_samples = min(num_samples or max_samples, max_samples)

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        s = self.config.sequence_length
        start = idx * s
        window = np.asarray(self.tokens[start:start + s + 1])
        return _build_sample(window, self.config)


class BlendedMegatronDatasetBuilder:
    """Builds train/valid/test datasets (reference
    blended_megatron_dataset_builder.py).  Mock or single-source token
    streams; multi-source blending by sample-proportional round-robin."""

    def __init__(self, cls, sizes, is_built_on_rank, config: GPTDatasetConfig):
        self.cls = cls
        self.sizes = sizes
        self.is_built_on_rank = is_built_on_rank
        self.config = config

    def build(self):
        out = []
        for split_idx, size in enumerate(self.sizes):
            if size is None or size == 0:
                out.append(None)
                continue
            if self.config.mock or self.config.blend is None:
                cfg = GPTDatasetConfig(**{**self.config.__dict__,
                                          "random_seed": self.config.random_seed + split_idx})
                out.append(MockGPTDataset(cfg, num_samples=size,
                                          name=f"split{split_idx}"))
            else:
                paths = self.config.blend
                out.append(self.cls(self.config, paths[0], num_samples=size))
        return out
