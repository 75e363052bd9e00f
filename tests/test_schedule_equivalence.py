"""Gradient equivalence: PP schedules vs single-process reference.

Every parameter is filled deterministically from a hash of its global
layer/name so the sharded and unsharded builds have identical weights;
after one fixed step the gradients must match to fp32 tolerance.
"""

import hashlib

import pytest
import torch

VOCAB = 64
SEQ = 16
MBS = 2
NUM_MICRO = 4
LAYERS = 4


def _fill_params_deterministic(model, layer_offset_map=None):
    """Fill each param from a generator seeded by its canonical name."""
    for name, p in model.named_parameters():
        canon = name.replace("module.", "")
        g = torch.Generator()
        seed = int(hashlib.md5(canon.encode()).hexdigest()[:8], 16)
        g.manual_seed(seed)
        with torch.no_grad():
            p.copy_(torch.randn(p.shape, generator=g) * 0.02)


def _canon_layer_name(name, offset):
    """Rewrite local layer indices to global: layers.<i> -> layers.<i+off>."""
    import re

    def repl(m):
        return f"layers.{int(m.group(1)) + offset}."
    return re.sub(r"layers\.(\d+)\.", repl, name)


def _make_config(pp, vpp, **kw):
    from megatronapp_amd.core.distributed.finalize_model_grads import (
        finalize_model_grads)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    defaults = dict(
        num_layers=LAYERS, hidden_size=32, num_attention_heads=4,
        ffn_hidden_size=64, pipeline_dtype=torch.float32,
        pipeline_model_parallel_size=pp,
        virtual_pipeline_model_parallel_size=vpp,
        hidden_dropout=0.0, attention_dropout=0.0,
        position_embedding_type="rope", normalization="RMSNorm",
        activation_func="swiglu", add_bias_linear=False,
        finalize_model_grads_func=finalize_model_grads)
    defaults.update(kw)
    return TransformerConfig(**defaults)


def _fixed_batches():
    """NUM_MICRO fixed batches, deterministic."""
    g = torch.Generator().manual_seed(777)
    batches = []
    for i in range(NUM_MICRO):
        tokens = torch.randint(0, VOCAB, (MBS, SEQ + 1), generator=g)
        batches.append({
            "tokens": tokens[:, :-1].contiguous(),
            "labels": tokens[:, 1:].contiguous(),
            "loss_mask": torch.ones(MBS, SEQ),
            "position_ids": torch.arange(SEQ).unsqueeze(0).expand(MBS, -1).contiguous(),
        })
    return batches


def _cyclic_batches():
    batches = _fixed_batches()
    i = 0
    while True:
        yield batches[i % NUM_MICRO]
        i += 1


def _forward_step_maker():
    fallback = _cyclic_batches()

    def forward_step(data_iterator, model):
        batch = next(data_iterator if data_iterator is not None else fallback)

        def loss_func(output_tensor):
            loss_mask = batch["loss_mask"].view(-1)
            loss = torch.sum(output_tensor.float().view(-1) * loss_mask) / loss_mask.sum()
            return loss, {"lm loss": loss.detach()}

        out = model(batch["tokens"], batch["position_ids"], labels=batch["labels"])
        return out, loss_func
    return forward_step


def _reference_grads():
    """Single-process full model grads keyed by canonical param name."""
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.distributed import (
        DistributedDataParallel, DistributedDataParallelConfig)
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.pipeline_parallel import get_forward_backward_func
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    from .utils import init_distributed

    init_distributed()
    parallel_state.initialize_model_parallel()
    model_parallel_cuda_manual_seed(1)
    config = _make_config(1, None)
    m = GPTModel(config=config,
                 transformer_layer_spec=get_gpt_layer_local_spec(
                     normalization="RMSNorm", use_flash=False),
                 vocab_size=VOCAB, max_sequence_length=SEQ,
                 position_embedding_type="rope",
                 share_embeddings_and_output_weights=False)
    _fill_params_deterministic(m)
    ddp = DistributedDataParallel(config, DistributedDataParallelConfig(), m)
    fb = get_forward_backward_func()
    fb(forward_step_func=_forward_step_maker(), data_iterator=None,
       model=ddp, num_microbatches=NUM_MICRO, seq_length=SEQ,
       micro_batch_size=MBS, forward_only=False)
    grads = {}
    for name, p in ddp.named_parameters():
        canon = name.replace("module.", "")
        grads[canon] = p.main_grad.clone()
    parallel_state.destroy_model_parallel()
    return grads


def _pp_run(rank, world_size, vpp, result_q):
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.distributed import (
        DistributedDataParallel, DistributedDataParallelConfig)
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.pipeline_parallel import get_forward_backward_func
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)

    pp = world_size
    parallel_state.initialize_model_parallel(
        pipeline_model_parallel_size=pp,
        virtual_pipeline_model_parallel_size=vpp)
    model_parallel_cuda_manual_seed(1)
    config = _make_config(pp, vpp)

    chunks, offsets = [], []
    vpp_eff = vpp or 1
    per_chunk = LAYERS // pp // vpp_eff
    for v in range(vpp_eff):
        vp_stage = v if vpp else None
        if vpp:
            parallel_state.set_virtual_pipeline_model_parallel_rank(v)
        pre = parallel_state.is_pipeline_first_stage()
        post = parallel_state.is_pipeline_last_stage()
        m = GPTModel(config=config,
                     transformer_layer_spec=get_gpt_layer_local_spec(
                         normalization="RMSNorm", use_flash=False),
                     vocab_size=VOCAB, max_sequence_length=SEQ,
                     position_embedding_type="rope", pre_process=pre,
                     post_process=post, vp_stage=vp_stage,
                     share_embeddings_and_output_weights=False)
        offset = (v * pp + rank) * per_chunk
        # fill with canonical (global-layer-index) names
        for name, p in m.named_parameters():
            canon = _canon_layer_name(name, offset)
            import hashlib as _h
            g = torch.Generator()
            g.manual_seed(int(_h.md5(canon.encode()).hexdigest()[:8], 16))
            with torch.no_grad():
                p.copy_(torch.randn(p.shape, generator=g) * 0.02)
        chunks.append(DistributedDataParallel(
            config, DistributedDataParallelConfig(), m))
        offsets.append(offset)

    fb = get_forward_backward_func()
    fs = _forward_step_maker()
    if vpp:
        fb(forward_step_func=fs,
           data_iterator=[_cyclic_batches() for _ in range(vpp_eff)],
           model=chunks,
           num_microbatches=NUM_MICRO, seq_length=SEQ, micro_batch_size=MBS,
           forward_only=False)
    else:
        fb(forward_step_func=fs, data_iterator=None, model=chunks[0],
           num_microbatches=NUM_MICRO, seq_length=SEQ, micro_batch_size=MBS,
           forward_only=False)

    out = {}
    for chunk, offset in zip(chunks, offsets):
        for name, p in chunk.named_parameters():
            canon = _canon_layer_name(name.replace("module.", ""), offset)
            out[canon] = p.main_grad.clone()
    result_q.put((rank, out))
    parallel_state.destroy_model_parallel()


@pytest.mark.parametrize("vpp", [None, 2])
def test_pp2_grads_match_single(vpp):
    import torch.multiprocessing as mp
    from .utils import _free_port
    import os

    ref = _reference_grads()

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = []
    for r in range(2):
        env = dict(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                   RANK=str(r), WORLD_SIZE="2")
        p = ctx.Process(target=_spawn_pp, args=(r, 2, vpp, q, port))
        p.start()
        procs.append(p)
    results = {}
    for _ in range(2):
        r, grads = q.get(timeout=300)
        results[r] = grads
    for p in procs:
        p.join(timeout=60)

    merged = {}
    for grads in results.values():
        merged.update(grads)

    # forward_step divides by num_microbatches in both runs; grads comparable
    missing = set(ref) - set(merged)
    assert not missing, f"missing grads: {missing}"
    for name, g in ref.items():
        got = merged[name]
        assert torch.allclose(got, g, atol=2e-4, rtol=1e-3), (
            name, (got - g).abs().max().item())


def _spawn_pp(rank, world_size, vpp, q, port):
    import os
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    dist.init_process_group(backend="gloo", rank=rank, world_size=world_size)
    try:
        _pp_run(rank, world_size, vpp, q)
    finally:
        dist.destroy_process_group()


def _tp_pp_run(rank, world_size, result_q):
    """TP=2 x PP=2 grid: every shard grad must equal the corresponding
    slice of the single-process reference grads."""
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.distributed import (
        DistributedDataParallel, DistributedDataParallelConfig)
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.pipeline_parallel import get_forward_backward_func
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    import hashlib as _h

    parallel_state.initialize_model_parallel(
        tensor_model_parallel_size=2, pipeline_model_parallel_size=2)
    tp_rank = parallel_state.get_tensor_model_parallel_rank()
    pp_rank = parallel_state.get_pipeline_model_parallel_rank()
    model_parallel_cuda_manual_seed(1)
    # non-gated activation: TP sharding is then a contiguous row/col slice
    config = _make_config(2, None, tensor_model_parallel_size=2,
                          activation_func="gelu")

    pre = parallel_state.is_pipeline_first_stage()
    post = parallel_state.is_pipeline_last_stage()
    m = GPTModel(config=config,
                 transformer_layer_spec=get_gpt_layer_local_spec(
                     normalization="RMSNorm", use_flash=False),
                 vocab_size=VOCAB, max_sequence_length=SEQ,
                 position_embedding_type="rope", pre_process=pre,
                 post_process=post,
                 share_embeddings_and_output_weights=False)
    offset = pp_rank * (LAYERS // 2)

    def full_shape_and_slice(p):
        if getattr(p, "tensor_model_parallel", False):
            dim = getattr(p, "partition_dim", 0)
            shape = list(p.shape)
            shape[dim] *= 2
            sl = [slice(None)] * len(shape)
            sl[dim] = slice(tp_rank * p.shape[dim],
                            (tp_rank + 1) * p.shape[dim])
            return tuple(shape), tuple(sl)
        return tuple(p.shape), tuple([slice(None)] * max(p.dim(), 1))

    for name, p in m.named_parameters():
        canon = _canon_layer_name(name, offset)
        shape, sl = full_shape_and_slice(p)
        g = torch.Generator()
        g.manual_seed(int(_h.md5(canon.encode()).hexdigest()[:8], 16))
        full = torch.randn(shape, generator=g) * 0.02
        with torch.no_grad():
            p.copy_(full[sl])

    ddp = DistributedDataParallel(config, DistributedDataParallelConfig(), m)
    fb = get_forward_backward_func()
    fb(forward_step_func=_forward_step_maker(), data_iterator=None,
       model=ddp, num_microbatches=NUM_MICRO, seq_length=SEQ,
       micro_batch_size=MBS, forward_only=False)

    out = {}
    for name, p in ddp.named_parameters():
        canon = _canon_layer_name(name.replace("module.", ""), offset)
        _, sl = full_shape_and_slice(p)
        out[canon] = (p.main_grad.clone(), sl)
    result_q.put((rank, out))
    parallel_state.destroy_model_parallel()


def test_tp2_pp2_grads_match_single():
    import torch.multiprocessing as mp
    from .utils import _free_port
    import os

    ref = None
    # reference with the same (gelu) activation
    global _make_config_orig
    import functools
    ref_cfg_kw = dict(activation_func="gelu")
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.distributed import (
        DistributedDataParallel, DistributedDataParallelConfig)
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.pipeline_parallel import get_forward_backward_func
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    from .utils import init_distributed

    init_distributed()
    parallel_state.initialize_model_parallel()
    model_parallel_cuda_manual_seed(1)
    config = _make_config(1, None, **ref_cfg_kw)
    mref = GPTModel(config=config,
                    transformer_layer_spec=get_gpt_layer_local_spec(
                        normalization="RMSNorm", use_flash=False),
                    vocab_size=VOCAB, max_sequence_length=SEQ,
                    position_embedding_type="rope",
                    share_embeddings_and_output_weights=False)
    _fill_params_deterministic(mref)
    ddp = DistributedDataParallel(config, DistributedDataParallelConfig(),
                                  mref)
    fb = get_forward_backward_func()
    fb(forward_step_func=_forward_step_maker(), data_iterator=None,
       model=ddp, num_microbatches=NUM_MICRO, seq_length=SEQ,
       micro_batch_size=MBS, forward_only=False)
    ref = {n.replace("module.", ""): p.main_grad.clone()
           for n, p in ddp.named_parameters()}
    parallel_state.destroy_model_parallel()
    import torch.distributed as dist
    dist.destroy_process_group()

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = []
    for r in range(4):
        env = dict(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                   RANK=str(r), WORLD_SIZE="4")
        p = ctx.Process(target=_grid_entry, args=(r, 4, port, q))
        p.start()
        procs.append(p)
    results = [q.get(timeout=300) for _ in range(4)]
    for p in procs:
        p.join(timeout=120)
    for rank, grads in results:
        for canon, (g, sl) in grads.items():
            assert canon in ref, canon
            expected = ref[canon][sl]
            err = (g - expected).abs().max()
            assert err < 2e-4, (rank, canon, float(err))


def _grid_entry(rank, world, port, q):
    import os
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        _tp_pp_run(rank, world, q)
    finally:
        dist.destroy_process_group()


def _sp_run(rank, world_size, result_q):
    """TP=2 with sequence parallelism: shard grads == reference slices."""
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.distributed import (
        DistributedDataParallel, DistributedDataParallelConfig)
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.pipeline_parallel import get_forward_backward_func
    import hashlib as _h
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)

    parallel_state.initialize_model_parallel(tensor_model_parallel_size=2)
    tp_rank = parallel_state.get_tensor_model_parallel_rank()
    model_parallel_cuda_manual_seed(1)
    config = _make_config(1, None, tensor_model_parallel_size=2,
                          activation_func="gelu", sequence_parallel=True)
    m = GPTModel(config=config,
                 transformer_layer_spec=get_gpt_layer_local_spec(
                     normalization="RMSNorm", use_flash=False),
                 vocab_size=VOCAB, max_sequence_length=SEQ,
                 position_embedding_type="rope",
                 share_embeddings_and_output_weights=False)

    def full_shape_and_slice(p):
        if getattr(p, "tensor_model_parallel", False):
            dim = getattr(p, "partition_dim", 0)
            shape = list(p.shape)
            shape[dim] *= 2
            sl = [slice(None)] * len(shape)
            sl[dim] = slice(tp_rank * p.shape[dim],
                            (tp_rank + 1) * p.shape[dim])
            return tuple(shape), tuple(sl)
        return tuple(p.shape), tuple([slice(None)] * max(p.dim(), 1))

    for name, p in m.named_parameters():
        shape, sl = full_shape_and_slice(p)
        g = torch.Generator()
        g.manual_seed(int(_h.md5(name.encode()).hexdigest()[:8], 16))
        full = torch.randn(shape, generator=g) * 0.02
        with torch.no_grad():
            p.copy_(full[sl])

    ddp = DistributedDataParallel(config, DistributedDataParallelConfig(), m)
    fb = get_forward_backward_func()
    fb(forward_step_func=_forward_step_maker(), data_iterator=None,
       model=ddp, num_microbatches=NUM_MICRO, seq_length=SEQ,
       micro_batch_size=MBS, forward_only=False)
    out = {}
    for name, p in ddp.named_parameters():
        _, sl = full_shape_and_slice(p)
        out[name.replace("module.", "")] = (p.main_grad.clone(), sl)
    result_q.put((rank, out))
    parallel_state.destroy_model_parallel()


def test_tp2_sequence_parallel_grads_match_single():
    import torch.multiprocessing as mp
    from .utils import _free_port, init_distributed
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.distributed import (
        DistributedDataParallel, DistributedDataParallelConfig)
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.pipeline_parallel import get_forward_backward_func
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    import torch.distributed as dist

    init_distributed()
    parallel_state.initialize_model_parallel()
    model_parallel_cuda_manual_seed(1)
    config = _make_config(1, None, activation_func="gelu")
    mref = GPTModel(config=config,
                    transformer_layer_spec=get_gpt_layer_local_spec(
                        normalization="RMSNorm", use_flash=False),
                    vocab_size=VOCAB, max_sequence_length=SEQ,
                    position_embedding_type="rope",
                    share_embeddings_and_output_weights=False)
    _fill_params_deterministic(mref)
    ddp = DistributedDataParallel(config, DistributedDataParallelConfig(),
                                  mref)
    fb = get_forward_backward_func()
    fb(forward_step_func=_forward_step_maker(), data_iterator=None,
       model=ddp, num_microbatches=NUM_MICRO, seq_length=SEQ,
       micro_batch_size=MBS, forward_only=False)
    ref = {n.replace("module.", ""): p.main_grad.clone()
           for n, p in ddp.named_parameters()}
    parallel_state.destroy_model_parallel()
    dist.destroy_process_group()

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [ctx.Process(target=_sp_entry, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in procs:
        p.join(timeout=120)
    for rank, grads in results:
        for canon, (g, sl) in grads.items():
            err = (g - ref[canon][sl]).abs().max()
            assert err < 2e-4, (rank, canon, float(err))


def _sp_entry(rank, world, port, q):
    import os
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        _sp_run(rank, world, q)
    finally:
        dist.destroy_process_group()
