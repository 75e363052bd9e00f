"""Pipeline-parallel schedule tests (gloo, 2 CPU processes)."""

import pytest
import torch

from .utils import spawn_ranks

VOCAB = 64
SEQ = 16


def _build(rank, pp, vpp=None):
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.distributed import (
        DistributedDataParallel, DistributedDataParallelConfig)
    from megatronapp_amd.core.distributed.finalize_model_grads import (
        finalize_model_grads)
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.optimizer import (
        OptimizerConfig, get_megatron_optimizer)
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    from megatronapp_amd.core.transformer_config import TransformerConfig

    parallel_state.initialize_model_parallel(
        pipeline_model_parallel_size=pp,
        virtual_pipeline_model_parallel_size=vpp)
    torch.manual_seed(1234)
    model_parallel_cuda_manual_seed(1234)
    config = TransformerConfig(
        num_layers=4, hidden_size=32, num_attention_heads=4,
        ffn_hidden_size=64, pipeline_dtype=torch.float32,
        pipeline_model_parallel_size=pp,
        virtual_pipeline_model_parallel_size=vpp,
        hidden_dropout=0.0, attention_dropout=0.0,
        position_embedding_type="rope", normalization="RMSNorm",
        activation_func="swiglu", add_bias_linear=False,
        finalize_model_grads_func=finalize_model_grads)

    def make_chunk(vp_stage=None):
        if vpp is None:
            pre = parallel_state.is_pipeline_first_stage()
            post = parallel_state.is_pipeline_last_stage()
        else:
            parallel_state.set_virtual_pipeline_model_parallel_rank(vp_stage)
            pre = parallel_state.is_pipeline_first_stage()
            post = parallel_state.is_pipeline_last_stage()
        m = GPTModel(
            config=config,
            transformer_layer_spec=get_gpt_layer_local_spec(
                normalization="RMSNorm", use_flash=False),
            vocab_size=VOCAB, max_sequence_length=SEQ,
            position_embedding_type="rope", pre_process=pre, post_process=post,
            share_embeddings_and_output_weights=False, vp_stage=vp_stage)
        ddp = DistributedDataParallel(
            config, DistributedDataParallelConfig(), m)
        return ddp

    if vpp is None:
        model = [make_chunk()]
    else:
        model = [make_chunk(v) for v in range(vpp)]
    opt = get_megatron_optimizer(
        OptimizerConfig(lr=5e-3, weight_decay=0.0, clip_grad=1.0), model)
    return config, model, opt


def _data_iter(batch_size):
    from megatronapp_amd.core.datasets import GPTDatasetConfig, MockGPTDataset
    ds = MockGPTDataset(GPTDatasetConfig(sequence_length=SEQ, vocab_size=VOCAB,
                                         random_seed=21), num_samples=8)
    idx = 0
    while True:
        samples = [ds[(idx + i) % len(ds)] for i in range(batch_size)]
        idx += batch_size
        yield {k: torch.stack([s[k] for s in samples]) for k in samples[0]}


def _forward_step(data_iterator, model):
    batch = next(data_iterator)

    def loss_func(output_tensor):
        loss_mask = batch["loss_mask"].view(-1).float()
        loss = torch.sum(output_tensor.float().view(-1) * loss_mask) / loss_mask.sum()
        return loss, {"lm loss": loss.detach()}

    from megatronapp_amd.core import parallel_state
    output = model(batch["tokens"], batch["position_ids"],
                   labels=batch["labels"])
    return output, loss_func


def _pp2_trains(rank, world_size, vpp=None):
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.pipeline_parallel import get_forward_backward_func

    config, model, opt = _build(rank, pp=2, vpp=vpp)
    fb = get_forward_backward_func()
    mbs = 2
    num_micro = 4
    losses = []
    # one iterator per model chunk: chunk c's j-th pull must be batch j so
    # labels on the last chunk align with tokens fed to the first chunk
    data_iterator = ([_data_iter(mbs) for _ in model] if vpp
                     else _data_iter(mbs))
    for step in range(20):
        for chunk in model:
            chunk.zero_grad_buffer()
        opt.zero_grad()
        out = fb(forward_step_func=_forward_step, data_iterator=data_iterator,
                 model=model if vpp else model[0], num_microbatches=num_micro,
                 seq_length=SEQ, micro_batch_size=mbs, forward_only=False)
        ok, grad_norm, _ = opt.step()
        assert ok
        if parallel_state.is_pipeline_last_stage(ignore_virtual=True):
            losses.append(torch.stack(
                [d["lm loss"] for d in out]).mean().item())
    if parallel_state.is_pipeline_last_stage(ignore_virtual=True):
        assert losses[-1] < losses[0] - 0.3, losses
    parallel_state.destroy_model_parallel()


def test_pp2_1f1b_trains():
    spawn_ranks(_pp2_trains, world_size=2)


def test_pp2_interleaved_trains():
    spawn_ranks(_pp2_trains, world_size=2, args=(2,))
