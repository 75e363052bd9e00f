"""Masked-LM dataset (reference core/datasets/masked_dataset.py): BERT
span masking over mock or token-stream data."""

from __future__ import annotations

import numpy as np
import torch

from .gpt_dataset import GPTDatasetConfig


class MockBertDataset(torch.utils.data.Dataset):
    """Synthetic masked-LM samples: 15% of tokens masked (80% [MASK],
    10% random, 10% kept), labels = original at masked positions."""

    def __init__(self, config: GPTDatasetConfig, num_samples: int = 1 << 20,
                 mask_prob: float = 0.15, name: str = "mock_bert"):
        self.config = config
        self.num_samples = num_samples
        self.mask_prob = mask_prob
        # reserve the last 3 ids: [CLS], [SEP], [MASK]
        self.cls_id = config.vocab_size - 3
        self.sep_id = config.vocab_size - 2
        self.mask_id = config.vocab_size - 1

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        rng = np.random.default_rng(self.config.random_seed + int(idx))
        seq = self.config.sequence_length
        tokens = rng.integers(0, self.cls_id, seq, dtype=np.int64)
        tokens[0] = self.cls_id
        tokens[seq // 2] = self.sep_id
        tokens[-1] = self.sep_id

        labels = np.full(seq, -1, dtype=np.int64)
        loss_mask = np.zeros(seq, dtype=np.float32)
        candidates = [i for i in range(seq)
                      if tokens[i] < self.cls_id]
        n_mask = max(1, int(len(candidates) * self.mask_prob))
        picked = rng.choice(candidates, size=n_mask, replace=False)
        masked = tokens.copy()
        for i in picked:
            labels[i] = tokens[i]
            loss_mask[i] = 1.0
            r = rng.random()
            if r < 0.8:
                masked[i] = self.mask_id
            elif r < 0.9:
                masked[i] = rng.integers(0, self.cls_id)
        tokentypes = np.zeros(seq, dtype=np.int64)
        tokentypes[seq // 2 + 1:] = 1
        is_next = int(rng.random() < 0.5)
        return {
            "text": torch.from_numpy(masked),
            "labels": torch.from_numpy(labels),
            "loss_mask": torch.from_numpy(loss_mask),
            "padding_mask": torch.ones(seq, dtype=torch.int64),
            "types": torch.from_numpy(tokentypes),
            "is_random": torch.tensor(is_next, dtype=torch.int64),
        }
