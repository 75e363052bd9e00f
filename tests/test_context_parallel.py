"""Context parallelism: CP=2 must reproduce the single-rank forward/loss."""

import hashlib

import pytest
import torch

from .utils import spawn_ranks

SEQ = 32
VOCAB = 64


def _fill(model):
    for name, p in model.named_parameters():
        g = torch.Generator()
        g.manual_seed(int(hashlib.md5(name.encode()).hexdigest()[:8], 16))
        with torch.no_grad():
            p.copy_(torch.randn(p.shape, generator=g) * 0.05)


def _build(cp_comm_type):
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core import parallel_state
    cp = parallel_state.get_context_parallel_world_size()
    config = TransformerConfig(
        num_layers=2, hidden_size=32, num_attention_heads=4,
        hidden_dropout=0.0, attention_dropout=0.0,
        position_embedding_type="rope", normalization="RMSNorm",
        activation_func="swiglu", add_bias_linear=False,
        context_parallel_size=cp, cp_comm_type=cp_comm_type,
        masked_softmax_fusion=True)
    m = GPTModel(config=config,
                 transformer_layer_spec=get_gpt_layer_local_spec(
                     normalization="RMSNorm", use_flash=False),
                 vocab_size=VOCAB, max_sequence_length=SEQ,
                 position_embedding_type="rope")
    _fill(m)
    return m


def _batch():
    g = torch.Generator().manual_seed(5)
    tokens = torch.randint(0, VOCAB, (2, SEQ), generator=g)
    pos = torch.arange(SEQ).unsqueeze(0).expand(2, -1)
    return tokens, pos


def _reference_logits(cp_comm_type="p2p"):
    from megatronapp_amd.core import parallel_state
    from .utils import init_distributed
    init_distributed()
    if parallel_state.model_parallel_is_initialized():
        parallel_state.destroy_model_parallel()
    parallel_state.initialize_model_parallel()
    m = _build("p2p")
    tokens, pos = _batch()
    with torch.no_grad():
        logits = m(tokens, pos)
    parallel_state.destroy_model_parallel()
    return logits


def _cp_run(rank, world, cp_comm_type, ref):
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.utils import get_batch_on_this_cp_rank
    parallel_state.initialize_model_parallel(context_parallel_size=world)
    m = _build(cp_comm_type)
    tokens, pos = _batch()
    batch = get_batch_on_this_cp_rank(
        {"tokens": tokens, "position_ids": pos})
    with torch.no_grad():
        logits_local = m(batch["tokens"], batch["position_ids"])
    # local rows are global chunks (rank, 2cp-1-rank)
    half = SEQ // (2 * world)
    c0, c1 = rank, 2 * world - 1 - rank
    expected = torch.cat([ref[:, c0 * half:(c0 + 1) * half],
                          ref[:, c1 * half:(c1 + 1) * half]], dim=1)
    err = (logits_local - expected).abs().max().item()
    assert err < 1e-4, f"cp rank {rank} {cp_comm_type} err {err}"
    parallel_state.destroy_model_parallel()


@pytest.mark.parametrize("mode", ["allgather", "a2a", "ring"])
def test_cp2_matches_single_rank(mode):
    ref = _reference_logits()
    spawn_ranks(_cp_run, world_size=2, args=(mode, ref))


def _ring_grad_run(rank, world, tmpdir):
    """Ring CP backward: dQ/dK/dV equal the full-attention grads."""
    import os
    import math
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.transformer.cp_attention import (
        _RingAttention, _local_global_positions)
    parallel_state.initialize_model_parallel(context_parallel_size=world)
    group = parallel_state.get_context_parallel_group()
    torch.manual_seed(11)
    sl_full, b, nh, hd = 32, 2, 2, 16
    qg = torch.randn(sl_full, b, nh, hd)
    kg = torch.randn(sl_full, b, nh, hd)
    vg = torch.randn(sl_full, b, nh, hd)
    dog = torch.randn(sl_full, b, nh, hd)

    # full reference
    qr = qg.clone().requires_grad_(True)
    kr = kg.clone().requires_grad_(True)
    vr = vg.clone().requires_grad_(True)
    scale = 1.0 / math.sqrt(hd)
    s = torch.einsum("qbnd,kbnd->bnqk", qr, kr) * scale
    mask = torch.triu(torch.ones(sl_full, sl_full, dtype=torch.bool), 1)
    s = s.masked_fill(mask, float("-inf"))
    o = torch.einsum("bnqk,kbnd->qbnd", torch.softmax(s, -1), vr)
    o.backward(dog)

    # local chunks for this rank
    half = sl_full // (2 * world)
    c0, c1 = rank, 2 * world - 1 - rank
    def take(t):
        return torch.cat([t[c0 * half:(c0 + 1) * half],
                          t[c1 * half:(c1 + 1) * half]], dim=0)
    ql = take(qg).clone().requires_grad_(True)
    kl = take(kg).clone().requires_grad_(True)
    vl = take(vg).clone().requires_grad_(True)
    out = _RingAttention.apply(ql, kl, vl, group, world, rank, scale)
    out.backward(take(dog))
    assert (out - take(o.detach())).abs().max() < 1e-4
    assert (ql.grad - take(qr.grad)).abs().max() < 1e-4, "dq mismatch"
    assert (kl.grad - take(kr.grad)).abs().max() < 1e-4, "dk mismatch"
    assert (vl.grad - take(vr.grad)).abs().max() < 1e-4, "dv mismatch"
    parallel_state.destroy_model_parallel()


def test_ring_attention_grads_match_full(tmp_path):
    spawn_ranks(_ring_grad_run, world_size=2, args=(str(tmp_path),))


def test_packed_sequences_block_diagonal():
    """Packed THD attention == running the segments separately."""
    from tests.utils import initialize_model_parallel, destroy
    from megatronapp_amd.core.packed_seq_params import PackedSeqParams
    from megatronapp_amd.core.transformer.dot_product_attention import (
        DotProductAttention)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.enums import AttnMaskType
    initialize_model_parallel()
    torch.manual_seed(2)
    cfg = TransformerConfig(num_layers=1, hidden_size=32,
                            num_attention_heads=2, hidden_dropout=0.0,
                            attention_dropout=0.0,
                            masked_softmax_fusion=False)
    attn = DotProductAttention(cfg, 1, AttnMaskType.causal)
    lens = [12, 8, 4]
    total = sum(lens)
    q = torch.randn(total, 1, 2, 16)
    k = torch.randn(total, 1, 2, 16)
    v = torch.randn(total, 1, 2, 16)
    cu = torch.tensor([0, 12, 20, 24], dtype=torch.int32)
    packed = attn(q, k, v,
                  packed_seq_params=PackedSeqParams(cu_seqlens_q=cu,
                                                    cu_seqlens_kv=cu))
    off = 0
    for ln in lens:
        sep = attn(q[off:off + ln], k[off:off + ln], v[off:off + ln],
                   attn_mask_type=AttnMaskType.causal)
        assert torch.allclose(packed[off:off + ln], sep, atol=1e-5), off
        off += ln
    destroy()
