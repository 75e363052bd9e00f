#!/usr/bin/env python3
"""Merge several .bin/.idx indexed datasets into one (reference
tools/merge_datasets.py)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from megatronapp_amd.core.datasets.indexed_dataset import (
    IndexedDataset, IndexedDatasetBuilder)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--input", nargs="+", required=True,
                   help="dataset path prefixes")
    p.add_argument("--output-prefix", required=True)
    args = p.parse_args()
    first = IndexedDataset(args.input[0])
    builder = IndexedDatasetBuilder(args.output_prefix, dtype=first.dtype)
    for prefix in args.input:
        ds = IndexedDataset(prefix)
        for i in range(len(ds)):
            builder.add_item(ds.get(i))
            builder.end_document()
        print(f"merged {prefix}: {len(ds)} sequences")
    builder.finalize()
    print(f"wrote {args.output_prefix}.bin/.idx")


if __name__ == "__main__":
    main()
