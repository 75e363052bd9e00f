"""Shared data utilities for the downstream-task harness (reference
tasks/data_utils.py).

``ByteTokenizer`` makes the harness runnable without any vocab file:
256 byte ids + [CLS]/[SEP]/[PAD] specials.  ``build_sample`` packs a
(text_a, text_b) pair into BERT-style ids/token-types/padding-mask,
truncating the longer text first like the reference's
``build_tokens_types_paddings_from_text``.
"""

from __future__ import annotations

import csv
from typing import List, Optional

import torch


class ByteTokenizer:
    CLS = 256
    SEP = 257
    PAD = 258

    @property
    def vocab_size(self):
        return 259

    @property
    def cls(self):
        return self.CLS

    @property
    def sep(self):
        return self.SEP

    @property
    def pad(self):
        return self.PAD

    def tokenize(self, text: str) -> List[int]:
        return list(text.encode("utf-8", errors="replace"))

    def detokenize(self, ids: List[int]) -> str:
        return bytes(i for i in ids if i < 256).decode(
            "utf-8", errors="replace")

    @property
    def eod(self):
        return self.PAD


def truncate_pair(a: List[int], b: Optional[List[int]], max_tokens: int):
    """Drop tokens from the longer sequence until the pair fits."""
    if b is None:
        return a[:max_tokens], None
    while len(a) + len(b) > max_tokens:
        if len(a) >= len(b):
            a = a[:-1]
        else:
            b = b[:-1]
    return a, b


def build_sample(tokenizer, text_a: str, text_b: Optional[str],
                 seq_length: int):
    """[CLS] a [SEP] (b [SEP]) with token types and 1/0 padding mask."""
    ids_a = tokenizer.tokenize(text_a)
    ids_b = tokenizer.tokenize(text_b) if text_b is not None else None
    specials = 2 if ids_b is None else 3
    ids_a, ids_b = truncate_pair(ids_a, ids_b, seq_length - specials)

    ids = [tokenizer.cls] + ids_a + [tokenizer.sep]
    types = [0] * len(ids)
    if ids_b is not None:
        ids += ids_b + [tokenizer.sep]
        types += [1] * (len(ids_b) + 1)
    mask = [1] * len(ids)
    pad = seq_length - len(ids)
    ids += [tokenizer.pad] * pad
    types += [0] * pad
    mask += [0] * pad
    return (torch.tensor(ids), torch.tensor(types),
            torch.tensor(mask, dtype=torch.float))


def read_tsv(path: str, skip_header: bool = True):
    with open(path, newline="", encoding="utf-8") as f:
        rows = list(csv.reader(f, delimiter="\t", quotechar=None))
    return rows[1:] if skip_header and rows else rows
