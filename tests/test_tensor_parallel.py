"""TP correctness: TP=2 matches TP=1 numerics (gloo, 2 CPU processes)."""

import os

import pytest
import torch

from .utils import spawn_ranks


def _tp2_linear_matches_single(rank, world_size):
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.tensor_parallel.layers import (
        ColumnParallelLinear, RowParallelLinear)
    from megatronapp_amd.core.transformer_config import TransformerConfig

    parallel_state.initialize_model_parallel(tensor_model_parallel_size=2)
    config = TransformerConfig(num_layers=1, hidden_size=8,
                               num_attention_heads=2,
                               gradient_accumulation_fusion=False)

    torch.manual_seed(99)
    w_full = torch.randn(16, 8)
    b_full = torch.randn(16)
    x = torch.randn(4, 3, 8, requires_grad=True)

    col = ColumnParallelLinear(8, 16, config=config,
                               init_method=lambda t: t,
                               bias=True, gather_output=True)
    shard = w_full.chunk(2, dim=0)[rank]
    with torch.no_grad():
        col.weight.copy_(shard)
        col.bias.copy_(b_full.chunk(2)[rank])
    out, _ = col(x)
    expected = x @ w_full.t() + b_full
    assert torch.allclose(out, expected, atol=1e-5), (out - expected).abs().max()

    # backward: grad wrt input must match the single-rank result
    out.sum().backward()
    x2 = x.detach().clone().requires_grad_(True)
    (x2 @ w_full.t() + b_full).sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)

    # RowParallelLinear: input sharded along last dim
    w2_full = torch.randn(8, 16)
    row = RowParallelLinear(16, 8, config=config, init_method=lambda t: t,
                            bias=True, input_is_parallel=True)
    with torch.no_grad():
        row.weight.copy_(w2_full.chunk(2, dim=1)[rank])
        row.bias.copy_(torch.zeros(8))
    y = torch.randn(4, 3, 16)
    y_shard = y.chunk(2, dim=-1)[rank]
    out2, _ = row(y_shard)
    expected2 = y @ w2_full.t()
    assert torch.allclose(out2, expected2, atol=1e-5)

    parallel_state.destroy_model_parallel()


def _vocab_parallel_ce(rank, world_size):
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.tensor_parallel.cross_entropy import (
        vocab_parallel_cross_entropy)

    parallel_state.initialize_model_parallel(tensor_model_parallel_size=2)
    torch.manual_seed(3)
    V, S, B = 32, 5, 2
    logits_full = torch.randn(S, B, V)
    target = torch.randint(0, V, (S, B))
    shard = logits_full.chunk(2, dim=-1)[rank].clone().requires_grad_(True)
    loss = vocab_parallel_cross_entropy(shard, target)
    ref = torch.nn.functional.cross_entropy(
        logits_full.reshape(-1, V), target.reshape(-1), reduction="none"
    ).reshape(S, B)
    assert torch.allclose(loss, ref, atol=1e-5), (loss - ref).abs().max()

    # gradient check against autograd on the full logits
    loss.sum().backward()
    full = logits_full.clone().requires_grad_(True)
    torch.nn.functional.cross_entropy(full.reshape(-1, V), target.reshape(-1),
                                      reduction="sum").backward()
    expected_grad = full.grad.chunk(2, dim=-1)[rank]
    assert torch.allclose(shard.grad, expected_grad, atol=1e-5)
    parallel_state.destroy_model_parallel()


def _embedding_tp2(rank, world_size):
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.tensor_parallel.layers import VocabParallelEmbedding
    from megatronapp_amd.core.transformer_config import TransformerConfig

    parallel_state.initialize_model_parallel(tensor_model_parallel_size=2)
    config = TransformerConfig(num_layers=1, hidden_size=8, num_attention_heads=2)
    torch.manual_seed(11)
    w_full = torch.randn(64, 8)
    emb = VocabParallelEmbedding(64, 8, init_method=lambda t: t, config=config)
    with torch.no_grad():
        emb.weight.copy_(w_full.chunk(2, dim=0)[rank])
    ids = torch.randint(0, 64, (2, 10))
    out = emb(ids)
    expected = torch.nn.functional.embedding(ids, w_full)
    assert torch.allclose(out, expected, atol=1e-5)
    parallel_state.destroy_model_parallel()


def test_tp2_linears():
    spawn_ranks(_tp2_linear_matches_single, world_size=2)


def test_tp2_vocab_cross_entropy():
    spawn_ranks(_vocab_parallel_ce, world_size=2)


def test_tp2_embedding():
    spawn_ranks(_embedding_tp2, world_size=2)
