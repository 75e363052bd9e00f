"""ICT biencoder retrieval model (reference
legacy/model/biencoder_model.py + pretrain_ict.py)."""
import os
import subprocess
import sys

import torch

from tests.utils import initialize_model_parallel, destroy

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _build(shared=False, proj=0):
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.models.bert.bert_layer_specs import (
        get_bert_layer_local_spec)
    from megatronapp_amd.core.models.biencoder import (
        biencoder_model_provider)
    cfg = TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        ffn_hidden_size=128, hidden_dropout=0.0, attention_dropout=0.0,
        masked_softmax_fusion=False)
    return biencoder_model_provider(
        config=cfg, transformer_layer_spec=get_bert_layer_local_spec(),
        vocab_size=128, max_sequence_length=64, projection_dim=proj,
        shared_query_context_model=shared)


def test_biencoder_forward_and_grad():
    initialize_model_parallel()
    try:
        torch.manual_seed(0)
        m = _build(proj=32)
        b, s = 4, 16
        q = torch.randint(0, 128, (b, s))
        c = torch.randint(0, 128, (b, s))
        mask = torch.ones(b, s)
        types = torch.zeros(b, s, dtype=torch.long)
        qe, ce = m(q, mask, types, c, mask, types)
        assert qe.shape == (b, 32) and ce.shape == (b, 32)
        # towers are independent
        assert m.query_model is not m.context_model
        scores = qe @ ce.t()
        loss = torch.nn.functional.cross_entropy(
            scores, torch.arange(b))
        loss.backward()
        assert any(p.grad is not None
                   for p in m.query_model.parameters())
    finally:
        destroy()


def test_biencoder_shared_towers():
    initialize_model_parallel()
    try:
        m = _build(shared=True)
        assert m.query_model is m.context_model
        b, s = 2, 8
        tok = torch.randint(0, 128, (b, s))
        mask = torch.ones(b, s)
        qe = m.embed_query(tok, mask)
        ce = m.embed_context(tok, mask)
        assert torch.allclose(qe, ce)      # same tower, same input
    finally:
        destroy()


def _allgather_worker(rank, world, port):
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from megatronapp_amd.core import parallel_state
    parallel_state.initialize_model_parallel()
    from megatronapp_amd.core.models.biencoder import (
        AllgatherFromDataParallelRegion)
    x = torch.full((2, 3), float(rank + 1), requires_grad=True)
    y = AllgatherFromDataParallelRegion.apply(x)
    assert y.shape == (2 * world, 3)
    assert torch.all(y[:2] == 1.0) and torch.all(y[2:] == 2.0)
    # backward returns only this rank's slice
    grad = torch.arange(float(2 * world * 3)).reshape(2 * world, 3)
    y.backward(grad)
    expect = grad[rank * 2:(rank + 1) * 2]
    assert torch.allclose(x.grad, expect)
    parallel_state.destroy_model_parallel()
    dist.destroy_process_group()


def test_allgather_dp_region_two_ranks():
    torch.multiprocessing.spawn(
        _allgather_worker, args=(2, 29679), nprocs=2, join=True)


def test_pretrain_ict_entry_runs(tmp_path):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29681",
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "pretrain_ict.py"),
         "--num-layers", "2", "--hidden-size", "64",
         "--num-attention-heads", "4", "--seq-length", "16",
         "--max-position-embeddings", "64", "--micro-batch-size", "4",
         "--global-batch-size", "4", "--vocab-size", "128",
         "--train-iters", "2", "--lr", "1e-4", "--eval-iters", "1",
         "--hidden-dropout", "0", "--attention-dropout", "0",
         "--retriever-score-scaling"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "lm loss" in out.stdout
