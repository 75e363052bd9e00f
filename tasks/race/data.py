"""RACE multiple-choice dataset (reference tasks/race/data.py).

Each sample packs NUM_CHOICES (question+choice, article) pairs; the
classification model scores each pair with one logit and the loss is a
softmax over choices.  Files are JSON-lines with
{article, questions, options, answers} as in the RACE release.
"""

from __future__ import annotations

import glob
import json
import os

import torch

from ..data_utils import build_sample

NUM_CHOICES = 4
MAX_QA_LENGTH = 128


class RaceDataset(torch.utils.data.Dataset):
    sample_multiplier = NUM_CHOICES
    num_classes = 1          # one score per (qa, article) pair

    def __init__(self, dataset_name, datapaths, tokenizer, max_seq_length,
                 max_qa_length=MAX_QA_LENGTH):
        self.dataset_name = dataset_name
        self.tokenizer = tokenizer
        self.max_seq_length = max_seq_length
        self.max_qa_length = max_qa_length
        self.samples = []
        for path in datapaths:
            self.samples.extend(self._read(path))

    def _read(self, datapath):
        if os.path.isdir(datapath):
            filenames = sorted(glob.glob(os.path.join(datapath, "*.txt")))
        else:
            filenames = [datapath]
        samples = []
        for filename in filenames:
            with open(filename, encoding="utf-8") as f:
                for line in f:
                    data = json.loads(line)
                    article = data["article"]
                    for qi, question in enumerate(data["questions"]):
                        label = ord(data["answers"][qi]) - ord("A")
                        choices = data["options"][qi]
                        assert len(choices) == NUM_CHOICES
                        qas = []
                        for choice in choices:
                            qa = question.replace("_", choice) \
                                if "_" in question else \
                                " ".join([question, choice])
                            qas.append(qa[:self.max_qa_length])
                        samples.append(
                            {"article": article, "qas": qas,
                             "label": label})
        return samples

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        s = self.samples[idx]
        ids, types, masks = [], [], []
        for qa in s["qas"]:
            i, t, m = build_sample(self.tokenizer, qa, s["article"],
                                   self.max_seq_length)
            ids.append(i)
            types.append(t)
            masks.append(m)
        return {"ids": torch.stack(ids),        # [choices, s]
                "types": torch.stack(types),
                "mask": torch.stack(masks),
                "label": torch.tensor(s["label"]),
                "uid": torch.tensor(idx)}
