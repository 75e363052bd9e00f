"""Checkpoint save -> resume round trip (CPU, single rank)."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

ARGS = [
    "--num-layers", "2", "--hidden-size", "64", "--num-attention-heads", "4",
    "--seq-length", "32", "--micro-batch-size", "2", "--global-batch-size", "4",
    "--mock-data", "--lr", "1e-3", "--log-interval", "1", "--vocab-size", "128",
    "--eval-iters", "0", "--hidden-dropout", "0", "--attention-dropout", "0",
]


def _run(extra, port):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "pretrain_gpt.py")] + ARGS + extra,
        capture_output=True, text=True, cwd=REPO, env=env, timeout=300)
    assert out.returncode == 0, out.stderr[-3000:]
    return out.stdout


def test_save_and_resume(tmp_path):
    save = str(tmp_path / "ckpt")
    out1 = _run(["--train-iters", "4", "--save", save, "--save-interval", "2",
                 "--ckpt-format", "torch"], 29631)
    assert os.path.exists(os.path.join(save, "latest_checkpointed_iteration.txt"))
    with open(os.path.join(save, "latest_checkpointed_iteration.txt")) as f:
        assert f.read().strip() == "4"
    assert os.path.exists(os.path.join(save, "iter_0000004", "mp_rank_00",
                                       "model_optim_rng.pt"))
    # resume: continues from iteration 4 to 6
    out2 = _run(["--train-iters", "6", "--save", save, "--load", save,
                 "--ckpt-format", "torch", "--save-interval", "100"], 29632)
    assert "loaded checkpoint" in out2
    assert "iteration        5/6" in out2 or "iteration        6/6" in out2
    # iterations 1-4 must NOT rerun
    assert "iteration        1/6" not in out2


def test_indexed_dataset_roundtrip(tmp_path):
    import numpy as np
    from megatronapp_amd.core.datasets.indexed_dataset import (
        IndexedDataset, IndexedDatasetBuilder)
    prefix = str(tmp_path / "corpus")
    b = IndexedDatasetBuilder(prefix, dtype=np.int32)
    docs = [[1, 2, 3], [7, 8, 9, 10, 11], [42]]
    for d in docs:
        b.add_item(d)
        b.end_document()
    b.finalize()
    ds = IndexedDataset(prefix)
    assert len(ds) == 3
    for i, d in enumerate(docs):
        assert list(ds.get(i)) == d
    assert list(ds.get(1, offset=1, length=2)) == [8, 9]
    # flat .bin stream view used by GPTDataset
    assert list(ds.bin[:3]) == [1, 2, 3]


def test_save_and_resume_torch_dist(tmp_path):
    save = str(tmp_path / "ckpt_dist")
    out1 = _run(["--train-iters", "3", "--save", save, "--save-interval", "3",
                 "--ckpt-format", "torch_dist"], 29633)
    assert os.path.exists(os.path.join(save, "iter_0000003", "index.json"))
    out2 = _run(["--train-iters", "5", "--save", save, "--load", save,
                 "--ckpt-format", "torch_dist", "--save-interval", "100"],
                29634)
    assert "loaded checkpoint (torch_dist)" in out2
    assert "iteration        1/5" not in out2
