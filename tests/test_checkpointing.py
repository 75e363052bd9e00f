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
    assert os.path.exists(os.path.join(save, "iter_0000003", ".metadata"))
    out2 = _run(["--train-iters", "5", "--save", save, "--load", save,
                 "--ckpt-format", "torch_dist", "--save-interval", "100"],
                29634)
    assert "loaded checkpoint (torch_dist)" in out2
    assert "iteration        1/5" not in out2


def test_doc_aware_sample_idx(tmp_path):
    """Native build_sample_idx == python reference; doc-aware GPTDataset
    yields seq+1 windows that respect the shuffled document stream."""
    import numpy as np
    from megatronapp_amd.core.datasets.build_helpers import load_helpers
    from megatronapp_amd.core.datasets.gpt_dataset import (
        GPTDataset, GPTDatasetConfig, _build_sample_idx_py)
    from megatronapp_amd.core.datasets.indexed_dataset import (
        IndexedDatasetBuilder)

    rng = np.random.RandomState(0)
    sizes = rng.randint(3, 40, size=57).astype(np.int32)
    doc_idx = np.concatenate([rng.permutation(57).astype(np.int32)
                              for _ in range(3)])
    tpe = int(sizes.sum())
    h = load_helpers()
    a = h.build_sample_idx(sizes, doc_idx, 16, 3, tpe)
    b = _build_sample_idx_py(sizes, doc_idx, 16, 3, tpe)
    assert np.array_equal(a, b)

    # real dataset round trip
    prefix = str(tmp_path / "docs")
    builder = IndexedDatasetBuilder(prefix)
    stream = {}
    for d in range(10):
        toks = rng.randint(0, 1000, size=rng.randint(20, 60)).astype(np.int32)
        builder.add_item(toks)
        builder.end_document()
        stream[d] = toks
    builder.finalize()
    cfg = GPTDatasetConfig(sequence_length=16, random_seed=7, mock=False)
    ds = GPTDataset(cfg, prefix, num_samples=12)
    assert len(ds) == 12
    for i in range(12):
        s = ds[i]
        assert s["tokens"].shape == (16,)
        # labels are the stream shifted by one
        full = ds._doc_aware_window(i)
        assert np.array_equal(s["tokens"].numpy(), full[:-1])
        assert np.array_equal(s["labels"].numpy(), full[1:])


def test_torch_dist_save_resume_pp2(tmp_path):
    """Sharded save + resume across a 2-rank pipeline: each stage writes
    its shards, the gathered index covers both, and resume continues."""
    import subprocess
    import sys
    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    save = str(tmp_path / "ck_pp2")
    base_args = [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
        "--master-port", "29667", os.path.join(REPO, "pretrain_gpt.py"),
        "--num-layers", "4", "--hidden-size", "64",
        "--num-attention-heads", "4", "--seq-length", "32",
        "--max-position-embeddings", "32", "--micro-batch-size", "2",
        "--global-batch-size", "8", "--pipeline-model-parallel-size", "2",
        "--mock-data", "--lr", "1e-3", "--log-interval", "1",
        "--vocab-size", "128", "--eval-iters", "0", "--hidden-dropout", "0",
        "--attention-dropout", "0", "--ckpt-format", "torch_dist",
        "--save", save]
    out = subprocess.run(base_args + ["--train-iters", "3",
                                      "--save-interval", "3"],
                         capture_output=True, text=True, cwd=REPO,
                         timeout=420)
    assert out.returncode == 0, out.stderr[-3000:]
    base_dir = os.path.join(save, "iter_0000003")
    assert os.path.exists(os.path.join(base_dir, ".metadata"))
    # both PP stages wrote DCP shard files
    distcp = [f for f in os.listdir(base_dir) if f.endswith(".distcp")]
    assert len(distcp) == 2, distcp
    out2 = subprocess.run(base_args + ["--train-iters", "5", "--load", save,
                                       "--save-interval", "100"],
                          capture_output=True, text=True, cwd=REPO,
                          timeout=420)
    assert out2.returncode == 0, out2.stderr[-3000:]
    assert "loaded checkpoint (torch_dist)" in out2.stdout

    # VPP save -> PP=1 resume (unified namespace + globalized layer keys)
    save_v = str(tmp_path / "ck_vpp")
    outv = subprocess.run(
        [a if a != save else save_v for a in base_args]
        + ["--train-iters", "2", "--save-interval", "2",
           "--num-layers-per-virtual-pipeline-stage", "1"],
        capture_output=True, text=True, cwd=REPO, timeout=420)
    assert outv.returncode == 0, outv.stderr[-3000:]
    outv2 = _run(["--num-layers", "4", "--hidden-size", "64",
                  "--num-attention-heads", "4", "--seq-length", "32",
                  "--max-position-embeddings", "32", "--micro-batch-size",
                  "2", "--global-batch-size", "8", "--vocab-size", "128",
                  "--hidden-dropout", "0", "--attention-dropout", "0",
                  "--ckpt-format", "torch_dist", "--train-iters", "4",
                  "--load", save_v, "--save-interval", "100",
                  "--eval-iters", "0"], 29670)
    assert "loaded checkpoint (torch_dist)" in outv2

    # cross-topology resume: the PP=2 sharded checkpoint loads at PP=1
    out3 = _run(["--num-layers", "4", "--hidden-size", "64",
                 "--num-attention-heads", "4", "--seq-length", "32",
                 "--max-position-embeddings", "32", "--micro-batch-size",
                 "2", "--global-batch-size", "8", "--vocab-size", "128",
                 "--hidden-dropout", "0", "--attention-dropout", "0",
                 "--ckpt-format", "torch_dist", "--train-iters", "5",
                 "--load", save, "--save-interval", "100",
                 "--eval-iters", "0"], 29668)
    assert "loaded checkpoint (torch_dist)" in out3


def test_non_persistent_local_checkpoint_resume(tmp_path):
    """Local (node-scratch) checkpoints save on their own interval and a
    NEWER local checkpoint wins over the persistent one at resume."""
    save = str(tmp_path / "persist")
    local = str(tmp_path / "scratch")
    args = ["--num-layers", "2", "--hidden-size", "64",
            "--num-attention-heads", "4", "--seq-length", "32",
            "--micro-batch-size", "2", "--global-batch-size", "4",
            "--mock-data", "--lr", "1e-3", "--log-interval", "1",
            "--vocab-size", "128", "--eval-iters", "0",
            "--hidden-dropout", "0", "--attention-dropout", "0",
            "--save", save, "--save-interval", "4",
            "--non-persistent-save-interval", "2",
            "--non-persistent-ckpt-dir", local]
    out = _run(["--train-iters", "6"] + args[:0] + args, 29761) \
        if False else None
    import subprocess, sys
    env = {**os.environ, "MASTER_ADDR": "127.0.0.1",
           "MASTER_PORT": "29761", "RANK": "0", "WORLD_SIZE": "1"}
    r = subprocess.run([sys.executable, "pretrain_gpt.py",
                        "--train-iters", "6"] + args,
                       capture_output=True, text=True, cwd=REPO,
                       timeout=420, env=env)
    assert r.returncode == 0, r.stderr[-2500:]
    assert open(os.path.join(local, "latest")).read().strip() == "6"
    # simulate a crash after the local save but before the final
    # persistent save: roll the persistent tracker back to 4
    with open(os.path.join(save,
              "latest_checkpointed_iteration.txt"), "w") as f:
        f.write("4")
    r2 = subprocess.run([sys.executable, "pretrain_gpt.py",
                         "--train-iters", "8", "--load", save] + args,
                        capture_output=True, text=True, cwd=REPO,
                        timeout=420, env=env)
    assert r2.returncode == 0, r2.stderr[-2500:]
    assert "loaded LOCAL (non-persistent) checkpoint at iteration 6" \
        in r2.stdout
