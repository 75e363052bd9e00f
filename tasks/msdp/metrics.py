"""Token-level F1 for generated text (reference tasks/msdp/metrics.py).

The normalization matches the reference (SQuAD-style: lowercase, strip
punctuation and articles, collapse whitespace)."""

from __future__ import annotations

import re
import string
from collections import Counter


def normalize_answer(s: str) -> str:
    s = s.lower()
    s = "".join(ch for ch in s if ch not in string.punctuation)
    s = re.sub(r"\b(a|an|the)\b", " ", s)
    return " ".join(s.split())


class F1Metric:
    @staticmethod
    def _prec_recall_f1_score(pred_items, gold_items):
        common = Counter(gold_items) & Counter(pred_items)
        num_same = sum(common.values())
        if num_same == 0:
            return 0.0, 0.0, 0.0
        precision = num_same / len(pred_items)
        recall = num_same / len(gold_items)
        f1 = (2 * precision * recall) / (precision + recall)
        return precision, recall, f1

    @staticmethod
    def compute_each_pair(guess: str, answer: str):
        g = normalize_answer(guess).split()
        a = normalize_answer(answer).split()
        if not a:
            return None, None, None
        if not g:
            return 0.0, 0.0, 0.0
        return F1Metric._prec_recall_f1_score(g, a)

    @staticmethod
    def compute_all_pairs(guesses, answers):
        assert len(guesses) == len(answers)
        ps, rs, fs = [], [], []
        for g, a in zip(guesses, answers):
            p, r, f = F1Metric.compute_each_pair(g, a)
            if p is not None:
                ps.append(p)
                rs.append(r)
                fs.append(f)
        n = max(len(fs), 1)
        return sum(ps) / n, sum(rs) / n, sum(fs) / n
