"""End-to-end learning sanity: build a structured token corpus (arithmetic
ramps — predictable continuations), train a small GPT on it through the
full stack (indexed dataset -> doc-aware sampling -> fused kernels ->
dist optimizer), and require a large loss drop.

python scripts/train_sanity.py [--iters 300]
"""
import argparse
import os
import subprocess
import sys
import tempfile

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def build_corpus(prefix, vocab=512, docs=2000):
    from megatronapp_amd.core.datasets.indexed_dataset import (
        IndexedDatasetBuilder)
    rng = np.random.RandomState(0)
    b = IndexedDatasetBuilder(prefix)
    for _ in range(docs):
        start = rng.randint(0, vocab)
        step = rng.randint(1, 4)
        n = rng.randint(64, 256)
        toks = (start + step * np.arange(n)) % vocab
        b.add_item(toks.astype(np.int32))
        b.end_document()
    b.finalize()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=300)
    args = ap.parse_args()
    d = tempfile.mkdtemp()
    prefix = os.path.join(d, "ramps")
    build_corpus(prefix)
    cmd = [sys.executable, os.path.join(REPO, "pretrain_gpt.py"),
           "--num-layers", "4", "--hidden-size", "256",
           "--num-attention-heads", "8", "--seq-length", "256",
           "--max-position-embeddings", "256", "--micro-batch-size", "8",
           "--global-batch-size", "32", "--vocab-size", "512",
           "--data-path", prefix, "--train-iters", str(args.iters),
           "--lr", os.environ.get("SANITY_LR", "1e-3"), "--lr-decay-style", "cosine",
           "--lr-warmup-iters", "20", "--log-interval", "20",
           "--eval-iters", "0", "--hidden-dropout", "0",
           "--attention-dropout", "0"]
    import torch
    if os.environ.get("SANITY_FORCE_BF16") == "1" or (
            torch.cuda.is_available()
            and os.environ.get("SANITY_FP32") != "1"):
        cmd.append("--bf16")
    if os.environ.get("SANITY_FP8") == "1":
        cmd.extend(["--fp8", "hybrid"])
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29673",
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    out = subprocess.run(cmd, capture_output=True, text=True, env=env)
    losses = []
    for line in out.stdout.splitlines():
        if "lm loss:" in line:
            losses.append(float(line.split("lm loss:")[1].split("|")[0]))
            print(line.strip())
    assert losses, out.stdout[-2000:] + out.stderr[-2000:]
    first, last = losses[0], min(losses[-3:])
    print(f"first={first:.3f} best_final={last:.3f}")
    assert last < first * 0.35, "loss did not drop enough"
    print("LEARNING SANITY OK")


if __name__ == "__main__":
    main()
