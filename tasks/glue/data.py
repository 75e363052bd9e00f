"""GLUE dataset base (reference tasks/glue/data.py:1-56,
tasks/glue/{mnli,qqp}.py).

Each task subclass parses its TSV column layout into
{text_a, text_b, label} samples; tokenization to BERT-style
ids/types/mask happens lazily in ``__getitem__``.
"""

from __future__ import annotations

from abc import ABC, abstractmethod

import torch

from ..data_utils import build_sample

MNLI_LABELS = {"contradiction": 0, "entailment": 1, "neutral": 2}


class GLUEAbstractDataset(ABC, torch.utils.data.Dataset):
    def __init__(self, task_name, dataset_name, datapaths, tokenizer,
                 max_seq_length):
        self.task_name = task_name
        self.dataset_name = dataset_name
        self.tokenizer = tokenizer
        self.max_seq_length = max_seq_length
        self.samples = []
        for path in datapaths:
            self.samples.extend(self.process_samples_from_single_path(path))

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        s = self.samples[idx]
        ids, types, mask = build_sample(
            self.tokenizer, s["text_a"], s["text_b"], self.max_seq_length)
        return {"ids": ids, "types": types, "mask": mask,
                "label": torch.tensor(s["label"]),
                "uid": torch.tensor(s["uid"])}

    @abstractmethod
    def process_samples_from_single_path(self, path):
        ...


class MNLIDataset(GLUEAbstractDataset):
    """MNLI TSV: col 0 = uid, 8 = premise, 9 = hypothesis, last = label
    (reference mnli.py)."""

    num_classes = 3

    def __init__(self, name, datapaths, tokenizer, max_seq_length,
                 test_label="contradiction"):
        self.test_label = test_label
        super().__init__("MNLI", name, datapaths, tokenizer, max_seq_length)

    def process_samples_from_single_path(self, path):
        samples = []
        with open(path, encoding="utf-8") as f:
            for i, line in enumerate(f):
                row = line.rstrip("\n").split("\t")
                if i == 0:
                    self.is_test = len(row) == 10
                    continue
                label = self.test_label if self.is_test \
                    else row[-1].strip()
                samples.append({"uid": int(row[0]),
                                "text_a": row[8].strip(),
                                "text_b": row[9].strip(),
                                "label": MNLI_LABELS[label]})
        return samples


class QQPDataset(GLUEAbstractDataset):
    """QQP TSV: train cols (uid, _, _, q1, q2, is_duplicate); test cols
    (uid, q1, q2) (reference qqp.py)."""

    num_classes = 2

    def __init__(self, name, datapaths, tokenizer, max_seq_length,
                 test_label=0):
        self.test_label = test_label
        super().__init__("QQP", name, datapaths, tokenizer, max_seq_length)

    def process_samples_from_single_path(self, path):
        samples = []
        with open(path, encoding="utf-8") as f:
            for i, line in enumerate(f):
                row = line.rstrip("\n").split("\t")
                if i == 0:
                    self.is_test = len(row) == 3
                    continue
                if self.is_test:
                    uid, a, b = int(row[0]), row[1].strip(), row[2].strip()
                    label = self.test_label
                else:
                    if len(row) != 6 or not row[3] or not row[4]:
                        continue
                    uid, a, b = int(row[0]), row[3].strip(), row[4].strip()
                    label = int(row[5])
                samples.append({"uid": uid, "text_a": a, "text_b": b,
                                "label": label})
        return samples
