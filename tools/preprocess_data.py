#!/usr/bin/env python3
"""Tokenize a JSONL corpus into the .bin/.idx indexed-dataset format
(reference tools/preprocess_data.py).

  python tools/preprocess_data.py --input corpus.jsonl \
      --json-keys text --output-prefix my_corpus \
      --tokenizer-type HuggingFaceTokenizer --tokenizer-model gpt2 \
      [--append-eod] [--workers 4]
"""

import argparse
import json
import multiprocessing
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np


def get_args():
    p = argparse.ArgumentParser()
    p.add_argument("--input", required=True)
    p.add_argument("--json-keys", nargs="+", default=["text"])
    p.add_argument("--output-prefix", required=True)
    p.add_argument("--tokenizer-type", default="NullTokenizer")
    p.add_argument("--tokenizer-model", default=None)
    p.add_argument("--vocab-file", default=None)
    p.add_argument("--merge-file", default=None)
    p.add_argument("--vocab-size", type=int, default=131072)
    p.add_argument("--append-eod", action="store_true")
    p.add_argument("--workers", type=int, default=1)
    p.add_argument("--log-interval", type=int, default=10000)
    return p.parse_args()


class Encoder:
    tokenizer = None

    def __init__(self, args):
        self.args = args

    def _build(self):
        from megatronapp_amd.training.tokenizer import build_tokenizer

        class _A:
            pass
        a = _A()
        a.tokenizer_type = self.args.tokenizer_type
        a.tokenizer_model = self.args.tokenizer_model
        a.vocab_file = self.args.vocab_file
        a.merge_file = self.args.merge_file
        a.padded_vocab_size = self.args.vocab_size
        a.vocab_size = self.args.vocab_size
        Encoder.tokenizer = build_tokenizer(a)

    def encode(self, line):
        if Encoder.tokenizer is None:
            self._build()
        try:
            data = json.loads(line)
        except json.JSONDecodeError:
            return None
        out = {}
        for key in self.args.json_keys:
            text = data.get(key, "")
            ids = Encoder.tokenizer.tokenize(text)
            if self.args.append_eod:
                ids = list(ids) + [Encoder.tokenizer.eod]
            out[key] = ids
        return out


def main():
    args = get_args()
    from megatronapp_amd.core.datasets.indexed_dataset import (
        IndexedDatasetBuilder)

    encoder = Encoder(args)
    builders = {key: IndexedDatasetBuilder(
        f"{args.output_prefix}_{key}_document", dtype=np.int32)
        for key in args.json_keys}

    t0 = time.time()
    n = 0
    with open(args.input) as f:
        if args.workers > 1:
            pool = multiprocessing.Pool(args.workers)
            docs = pool.imap(encoder.encode, f, chunksize=32)
        else:
            docs = map(encoder.encode, f)
        for doc in docs:
            if doc is None:
                continue
            for key, ids in doc.items():
                if ids:
                    builders[key].add_item(ids)
                    builders[key].end_document()
            n += 1
            if n % args.log_interval == 0:
                rate = n / (time.time() - t0)
                print(f"processed {n} documents ({rate:.0f} docs/s)",
                      flush=True)
    for key, b in builders.items():
        b.finalize()
        print(f"wrote {args.output_prefix}_{key}_document.bin/.idx")


if __name__ == "__main__":
    main()
