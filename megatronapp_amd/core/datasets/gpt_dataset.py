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


def _build_sample_idx_py(sizes, doc_idx, seq_length, num_epochs,
                         tokens_per_epoch):
    """Pure-python reference for the native build_sample_idx (tests)."""
    num_samples = (num_epochs * tokens_per_epoch - 1) // seq_length
    out = np.zeros((num_samples + 1, 2), dtype=np.int64)
    doc_pos, doc_off = 0, 0
    for s in range(1, num_samples + 1):
        remaining = seq_length
        while remaining > 0:
            doc_len = sizes[doc_idx[doc_pos]] - doc_off
            if doc_len > remaining:
                doc_off += remaining
                remaining = 0
            else:
                remaining -= doc_len
                doc_pos += 1
                doc_off = 0
        out[s] = (doc_pos, doc_off)
    return out


class GPTDataset(torch.utils.data.Dataset):
    """Causal-LM samples (reference gpt_dataset.py three-index design).

    With a reference-format .idx/.bin pair: documents are shuffled per
    epoch (doc_idx), samples map to (document position, offset) honoring
    document boundaries via the native ``build_sample_idx`` helper, and a
    shuffle_idx permutes sample order — so a sample can span documents and
    epochs differ in order, like the reference.  Flat .npy/.bin streams
    fall back to fixed windows."""

    def __init__(self, config: GPTDatasetConfig, token_file: str,
                 num_samples: Optional[int] = None, name: str = "gpt",
                 sample_offset: int = 0):
        self.config = config
        self.name = name
        self.sample_offset = sample_offset
        self.indexed = None
        if token_file.endswith(".npy"):
            self.tokens = np.load(token_file, mmap_mode="r")
        elif os.path.exists(token_file + ".idx"):
            from .indexed_dataset import IndexedDataset
            self.indexed = IndexedDataset(token_file)
            self.tokens = self.indexed.bin
        else:
            self.tokens = np.memmap(token_file, dtype=np.int32, mode="r")

        seq = config.sequence_length
        if self.indexed is not None and self.indexed.document_count > 1:
            self._build_doc_aware_indices(num_samples)
        else:
            max_samples = (len(self.tokens) - 1) // seq
            avail = max(0, max_samples - sample_offset)
            self.num_samples = min(num_samples or avail, avail)
            self.sample_idx = None

    def _build_doc_aware_indices(self, num_samples):
        cfg = self.config
        seq = cfg.sequence_length
        sizes = self.indexed.sequence_lengths
        docs = np.arange(len(sizes), dtype=np.int32)
        tokens_per_epoch = int(sizes.sum())
        want = num_samples or max(1, (tokens_per_epoch - 1) // seq)
        num_epochs = max(1, -(-(want * seq + 1) // tokens_per_epoch))
        rng = np.random.RandomState(cfg.random_seed)
        doc_idx = np.concatenate(
            [rng.permutation(docs) for _ in range(num_epochs)])
        try:
            from .build_helpers import load_helpers
            self.sample_idx = load_helpers().build_sample_idx(
                sizes.astype(np.int32), doc_idx, seq, num_epochs,
                tokens_per_epoch)
        except Exception:
            self.sample_idx = _build_sample_idx_py(
                sizes, doc_idx, seq, num_epochs, tokens_per_epoch)
        self.doc_idx = doc_idx
        avail = self.sample_idx.shape[0] - 1
        perm = rng.permutation(avail)
        off = min(self.sample_offset, avail)
        self.num_samples = min(want, avail - off)
        self.shuffle_idx = perm[off:off + self.num_samples]

    def __len__(self):
        return self.num_samples

    def _doc_aware_window(self, idx):
        seq = self.config.sequence_length
        i = int(self.shuffle_idx[idx])
        pos0, off0 = self.sample_idx[i]
        pos1, off1 = self.sample_idx[i + 1]
        parts = []
        need = seq + 1
        pos, off = int(pos0), int(off0)
        while need > 0:
            doc = int(self.doc_idx[pos])
            chunk = self.indexed.get(doc, offset=off, length=need)
            parts.append(chunk)
            need -= len(chunk)
            pos += 1
            off = 0
        return np.concatenate(parts)

    def __getitem__(self, idx):
        s = self.config.sequence_length
        if self.sample_idx is not None:
            return _build_sample(self._doc_aware_window(idx), self.config)
        start = (idx + self.sample_offset) * s
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
                out.append(self._build_real(split_idx, size))
        return out

    def _split_offsets(self, path):
        """Partition a source's sample space by config.split ratios."""
        probe = self.cls(self.config, path)
        total = len(probe)
        if not self.config.split:
            return {0: (0, total), 1: (0, 0), 2: (0, 0)}
        parts = [float(x) for x in str(self.config.split).split(",")]
        parts += [0.0] * (3 - len(parts))
        norm = sum(parts) or 1.0
        counts = [int(total * p / norm) for p in parts]
        offs, off = {}, 0
        for i, c in enumerate(counts):
            offs[i] = (off, c)
            off += c
        return offs

    def _build_real(self, split_idx, size):
        """Real token files: split-ratio windows per source; multiple
        sources interleave by the native blending indices."""
        blend = self.config.blend
        # reference blend format may interleave weights and paths
        if all(isinstance(b, str) for b in blend):
            paths = list(blend)
            weights = [1.0] * len(paths)
        else:
            weights = [float(b) for b in blend[0::2]]
            paths = list(blend[1::2])
        sources = []
        for path in paths:
            off, avail = self._split_offsets(path)[split_idx]
            n = min(size, avail) if avail else 0
            sources.append(self.cls(self.config, path, num_samples=n,
                                    sample_offset=off,
                                    name=f"split{split_idx}"))
        if len(sources) == 1:
            return sources[0]
        return BlendedDataset(sources, weights, size)


class BlendedDataset(torch.utils.data.Dataset):
    """Proportional interleave of sources via the native
    build_blending_indices helper (reference blended_dataset.py)."""

    def __init__(self, datasets, weights, size):
        import numpy as _np
        self.datasets = datasets
        cap = sum(len(d) for d in datasets)
        self.size = min(size, cap) if cap else 0
        w = _np.array(weights, dtype=_np.float64)
        w = w / w.sum()
        try:
            from .build_helpers import load_helpers
            di, dsi = load_helpers().build_blending_indices(w, self.size)
            self.dataset_index = _np.asarray(di)
            self.dataset_sample_index = _np.asarray(dsi)
        except Exception:
            counts = _np.zeros(len(datasets), dtype=_np.int64)
            self.dataset_index = _np.zeros(self.size, dtype=_np.int16)
            self.dataset_sample_index = _np.zeros(self.size, dtype=_np.int64)
            for i in range(self.size):
                err = w * (i + 1) - counts
                d = int(err.argmax())
                self.dataset_index[i] = d
                self.dataset_sample_index[i] = counts[d]
                counts[d] += 1

    def __len__(self):
        return self.size

    def __getitem__(self, idx):
        d = int(self.dataset_index[idx])
        s = int(self.dataset_sample_index[idx]) % max(len(self.datasets[d]), 1)
        return self.datasets[d][s]
