"""Sharded checkpointing: save/load round trip + cross-topology reshard
(save at TP=2, load at TP=1 and vice versa)."""

import os

import pytest
import torch

from .utils import destroy, initialize_model_parallel, spawn_ranks

H = 32


def _build_linear_state(tp_rank, tp_world, seed=3):
    """A fake column-parallel weight [out=64/tp, in=32] + a replicated
    bias-like tensor, with deterministic global contents."""
    from megatronapp_amd.core.dist_checkpointing import ShardedTensor
    g = torch.Generator().manual_seed(seed)
    full = torch.randn(64, H, generator=g)
    shard = full.chunk(tp_world, dim=0)[tp_rank].clone()
    st = ShardedTensor.from_rank_offsets("w", shard, (0, tp_rank, tp_world))
    norm = torch.randn(H, generator=g)
    from megatronapp_amd.core import parallel_state
    rep = ShardedTensor("norm", norm.clone(), tuple(norm.shape), (0,),
                        replica_id=tp_rank)
    return {"w": st, "norm": rep}, full, norm


def test_single_rank_roundtrip(tmp_path):
    initialize_model_parallel()
    from megatronapp_amd.core.dist_checkpointing import load, save, load_common
    sd, full, norm = _build_linear_state(0, 1)
    save(sd, str(tmp_path), common_state={"iteration": 7})
    sd2, _, _ = _build_linear_state(0, 1, seed=99)  # different contents
    load(sd2, str(tmp_path))
    assert torch.equal(sd2["w"].data, full)
    assert torch.equal(sd2["norm"].data, norm)
    assert load_common(str(tmp_path))["iteration"] == 7
    destroy()


def _save_tp2(rank, world, ckpt_dir):
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.dist_checkpointing import save
    parallel_state.initialize_model_parallel(tensor_model_parallel_size=2)
    sd, full, norm = _build_linear_state(rank, 2)
    save(sd, ckpt_dir, common_state={"iteration": 3})
    parallel_state.destroy_model_parallel()


def _load_tp2(rank, world, ckpt_dir):
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.dist_checkpointing import load
    parallel_state.initialize_model_parallel(tensor_model_parallel_size=2)
    sd, _, _ = _build_linear_state(rank, 2, seed=123)
    load(sd, ckpt_dir)
    # expected contents come from the SAVED checkpoint (seed 3)
    g = torch.Generator().manual_seed(3)
    full = torch.randn(64, 32, generator=g)
    norm = torch.randn(32, generator=g)
    expected = full.chunk(2, dim=0)[rank]
    assert torch.equal(sd["w"].data, expected), rank
    assert torch.equal(sd["norm"].data, norm)
    parallel_state.destroy_model_parallel()


def test_reshard_tp2_to_tp1(tmp_path):
    ckpt = str(tmp_path / "tp2")
    spawn_ranks(_save_tp2, world_size=2, args=(ckpt,))
    # load the TP2-saved checkpoint at TP1
    initialize_model_parallel()
    from megatronapp_amd.core.dist_checkpointing import load
    g = torch.Generator().manual_seed(3)
    full = torch.randn(64, H, generator=g)
    norm = torch.randn(H, generator=g)
    sd, _, _ = _build_linear_state(0, 1, seed=55)
    load(sd, ckpt)
    assert torch.equal(sd["w"].data, full)
    assert torch.equal(sd["norm"].data, norm)
    destroy()


def test_reshard_tp1_to_tp2(tmp_path):
    ckpt = str(tmp_path / "tp1")
    initialize_model_parallel()
    from megatronapp_amd.core.dist_checkpointing import save
    sd, full, norm = _build_linear_state(0, 1)
    save(sd, ckpt, common_state={})
    destroy()
    spawn_ranks(_load_tp2, world_size=2, args=(ckpt,))


def test_model_sharded_state_dict(tmp_path):
    initialize_model_parallel()
    from megatronapp_amd.core.dist_checkpointing import (
        load, module_sharded_state_dict, save)
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    torch.manual_seed(4)
    config = TransformerConfig(num_layers=2, hidden_size=32,
                               num_attention_heads=4, hidden_dropout=0.0,
                               attention_dropout=0.0)
    m = GPTModel(config=config,
                 transformer_layer_spec=get_gpt_layer_local_spec(use_flash=False),
                 vocab_size=64, max_sequence_length=32)
    sd = m.sharded_state_dict()
    assert "embedding.word_embeddings.weight" in sd
    save(sd, str(tmp_path))
    # perturb then restore
    with torch.no_grad():
        before = {k: p.clone() for k, p in m.named_parameters()}
        for p in m.parameters():
            p.add_(1.0)
    load(m.sharded_state_dict(), str(tmp_path))
    for k, p in m.named_parameters():
        assert torch.allclose(p, before[k]), k
    destroy()


def test_reshard_property_random_grids(tmp_path):
    """Property test: save sharded on grid A, load on grid B — exact for
    arbitrary shard factorizations of 2-D tensors."""
    from hypothesis import given, settings, strategies as st
    import torch
    from megatronapp_amd.core.dist_checkpointing import (
        ShardedTensor, save as dist_save, load as dist_load)

    case_idx = [0]

    @settings(max_examples=25, deadline=None)
    @given(
        rows=st.sampled_from([8, 12, 16, 24]),
        cols=st.sampled_from([4, 8, 16]),
        grid_a=st.sampled_from([(1, 1), (2, 1), (1, 2), (2, 2), (4, 1)]),
        grid_b=st.sampled_from([(1, 1), (2, 1), (1, 2), (2, 2), (1, 4)]),
        seed=st.integers(0, 1000))
    def check(rows, cols, grid_a, grid_b, seed):
        ar, ac = grid_a
        br, bc = grid_b
        if rows % (ar * br) or cols % (ac * bc):
            return
        case_idx[0] += 1
        d = tmp_path / f"case{case_idx[0]}"
        g = torch.Generator().manual_seed(seed)
        full = torch.randn(rows, cols, generator=g)
        # simulate grid-A ranks saving in one process: each writes its
        # shard file; rank 0 of the simulation writes the index
        import json, os
        os.makedirs(d, exist_ok=True)
        index = {}
        for i in range(ar):
            for j in range(ac):
                r0, c0 = i * rows // ar, j * cols // ac
                shard = full[r0:r0 + rows // ar, c0:c0 + cols // ac]
                fname = f"shards_rank{(i * ac + j):05d}.pt"
                torch.save({"t": {"offset": (r0, c0),
                                  "global_shape": (rows, cols),
                                  "tensor": shard.clone()}}, d / fname)
                index.setdefault("t", []).append(
                    {"file": fname, "offset": [r0, c0],
                     "shape": [rows // ar, cols // ac],
                     "global_shape": [rows, cols]})
        json.dump(index, open(d / "index.json", "w"))
        # load every grid-B shard and verify
        for i in range(br):
            for j in range(bc):
                r0, c0 = i * rows // br, j * cols // bc
                out = torch.zeros(rows // br, cols // bc)
                dist_load({"t": ShardedTensor("t", out, (rows, cols),
                                              (r0, c0))}, str(d))
                assert torch.equal(
                    out, full[r0:r0 + rows // br, c0:c0 + cols // bc])

    check()


def test_torch_dcp_interchange(tmp_path):
    """The sharded format IS torch-DCP: plain torch.distributed.checkpoint
    loads our checkpoints, and we load checkpoints produced by plain DCP
    (what reference/upstream tooling writes)."""
    import torch.distributed.checkpoint as dcp
    from megatronapp_amd.core.dist_checkpointing.mapping import ShardedTensor
    from megatronapp_amd.core.dist_checkpointing.torch_dcp import (
        is_dcp_checkpoint, load_dcp, load_dcp_consolidated, save_dcp)
    initialize_model_parallel()
    d = str(tmp_path / "ours")
    t1 = torch.randn(8, 16)
    t2 = torch.arange(24.0).view(4, 6)
    sd = {"model.a.weight": ShardedTensor("model.a.weight", t1, (8, 16),
                                          (0, 0), 0),
          "model.b.bias": ShardedTensor("model.b.bias", t2, (4, 6),
                                        (0, 0), 0)}
    save_dcp(sd, d, common_state={"iteration": 7})
    assert is_dcp_checkpoint(d)
    # upstream direction: vanilla DCP reads our checkpoint
    target = {"model.a.weight": torch.empty(8, 16),
              "model.b.bias": torch.empty(4, 6)}
    dcp.load(target, storage_reader=dcp.FileSystemReader(d))
    assert torch.equal(target["model.a.weight"], t1)
    assert torch.equal(target["model.b.bias"], t2)
    # import direction: we read a vanilla-DCP (reference-style) checkpoint
    d2 = str(tmp_path / "theirs")
    dcp.save({"x": t1}, storage_writer=dcp.FileSystemWriter(d2))
    full = load_dcp_consolidated(d2)
    assert torch.equal(full["x"], t1)
    # resharded load: two half-windows of our own checkpoint
    h1, h2 = torch.empty(4, 16), torch.empty(4, 16)
    load_dcp({"model.a.weight": ShardedTensor("model.a.weight", h1,
                                              (8, 16), (0, 0), 0)}, d)
    load_dcp({"model.a.weight": ShardedTensor("model.a.weight", h2,
                                              (8, 16), (4, 0), 0)}, d)
    assert torch.equal(torch.cat([h1, h2]), t1)
