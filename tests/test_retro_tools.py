"""Retro preprocessing pipeline + BERT embedder (reference tools/retro,
tools/bert_embedding)."""
import numpy as np
import torch

from tests.utils import initialize_model_parallel, destroy


def test_chunk_db_and_neighbor_exclusion():
    from tools.retro.preprocess import (
        build_chunk_db, BruteForceMIPSIndex)
    docs = [np.arange(20), np.arange(100, 113), np.arange(200, 216)]
    chunks, doc_ids = build_chunk_db(docs, chunk_length=8, pad_id=0)
    # 20 -> 3 chunks (last padded), 13 -> 2, 16 -> 2
    assert chunks.shape == (7, 8)
    assert doc_ids.tolist() == [0, 0, 0, 1, 1, 2, 2]
    assert chunks[2, 4] == 0          # tail padding
    # identical embeddings within a doc; search must exclude same-doc
    emb = np.stack([np.eye(4)[d] for d in doc_ids]).astype(np.float32)
    idx = BruteForceMIPSIndex(emb, device="cpu")
    nb = idx.search(emb, 2, query_docs=doc_ids, base_docs=doc_ids)
    for q in range(7):
        assert all(doc_ids[n] != doc_ids[q] for n in nb[q])


def test_neighbor_continuation_windows():
    from tools.retro.preprocess import load_neighbor_tokens
    chunks = np.arange(40).reshape(5, 8)
    doc_ids = np.array([0, 0, 0, 1, 1])
    out = load_neighbor_tokens(chunks, doc_ids, np.array([1, 2, 4]),
                               pad_id=-1, num_retrieved_chunks=2)
    assert out.shape == (3, 16)
    # chunk 1 continues into chunk 2 (same doc)
    assert out[0].tolist() == list(range(8, 24))
    # chunk 2 has no same-doc continuation -> padded
    assert out[1, :8].tolist() == list(range(16, 24))
    assert all(v == -1 for v in out[1, 8:])
    # chunk 4 is the last chunk -> padded continuation
    assert out[2, :8].tolist() == list(range(32, 40))


def test_end_to_end_project_build(tmp_path):
    initialize_model_parallel()
    try:
        from megatronapp_amd.core.transformer_config import (
            TransformerConfig)
        from tools.bert_embedding.embed import BertEmbedder
        from tools.retro.preprocess import (
            build_retro_project, load_retro_project, load_neighbor_tokens)
        torch.manual_seed(0)
        cfg = TransformerConfig(
            num_layers=2, hidden_size=32, num_attention_heads=4,
            ffn_hidden_size=64, hidden_dropout=0.0, attention_dropout=0.0,
            masked_softmax_fusion=False)
        embedder = BertEmbedder(cfg, vocab_size=64, max_sequence_length=32,
                                device="cpu")
        rng = np.random.RandomState(0)
        docs = [rng.randint(1, 64, size=rng.randint(20, 50))
                for _ in range(5)]
        chunks, doc_ids, neighbors = build_retro_project(
            str(tmp_path / "proj"), docs, embedder, pad_id=0,
            chunk_length=8, num_neighbors=2)
        assert neighbors.shape[1] == 2
        c2, d2, n2, meta = load_retro_project(str(tmp_path / "proj"))
        assert np.array_equal(c2, chunks) and meta["chunk_length"] == 8
        nt = load_neighbor_tokens(c2, d2, n2[0], meta["pad_id"],
                                  meta["num_retrieved_chunks"])
        assert nt.shape == (2, 16)
        # neighbors of chunk 0 are never from document 0
        assert all(d2[n] != d2[0] for n in n2[0])
    finally:
        destroy()


def test_pretrain_retro_with_project_dir(tmp_path):
    """Full pipeline: corpus -> chunk db -> embed -> neighbors ->
    pretrain_retro trains from the project directory."""
    import os
    import subprocess
    import sys
    initialize_model_parallel()
    try:
        from megatronapp_amd.core.transformer_config import (
            TransformerConfig)
        from tools.bert_embedding.embed import BertEmbedder
        from tools.retro.preprocess import build_retro_project
        cfg = TransformerConfig(
            num_layers=2, hidden_size=32, num_attention_heads=4,
            ffn_hidden_size=64, hidden_dropout=0.0, attention_dropout=0.0,
            masked_softmax_fusion=False)
        emb = BertEmbedder(cfg, vocab_size=128, max_sequence_length=32,
                           device="cpu")
        rng = np.random.RandomState(0)
        docs = [rng.randint(1, 128, size=rng.randint(30, 60))
                for _ in range(6)]
        proj = str(tmp_path / "proj")
        build_retro_project(proj, docs, emb, pad_id=0, chunk_length=8,
                            num_neighbors=2)
    finally:
        destroy()
    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29712",
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "pretrain_retro.py"),
         "--num-layers", "4", "--hidden-size", "64",
         "--num-attention-heads", "4", "--seq-length", "32",
         "--max-position-embeddings", "64", "--micro-batch-size", "2",
         "--global-batch-size", "2", "--vocab-size", "128",
         "--retro-chunk-length", "8", "--retro-num-neighbors", "2",
         "--retro-project-dir", proj,
         "--train-iters", "2", "--lr", "1e-4", "--eval-iters", "1",
         "--hidden-dropout", "0", "--attention-dropout", "0"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "lm loss" in out.stdout
