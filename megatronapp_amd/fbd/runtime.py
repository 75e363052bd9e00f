"""MegaFBD training runtime: the disaggregated pretrain loop.

Reference wiring: training.py:991-1010 (thread spawn — here a clean
process split), checkpointing via the legacy path (reference
save_checkpoint_legacy, training.py:833-834).
"""

from __future__ import annotations

import time

import torch
import torch.distributed as dist

from ..core import parallel_state
from ..core.num_microbatches_calculator import get_num_microbatches
from ..training.arguments import core_transformer_config_from_args
from ..training.training import (
    _build_train_valid_test_data_iterators,
    get_model,
    get_optimizer_param_scheduler,
    print_rank_0,
)
from .schedule import forward_backward_disaggregated


def _sync_params_to_forward(model, dual_rank: int, is_forward: bool):
    """Broadcast updated params from the backward instance to its dual
    forward instance over the pair group."""
    group = parallel_state.get_forward_backward_parallel_group()
    src = dual_rank if is_forward else dist.get_rank()
    for param in model.module.parameters() if hasattr(model, "module") else \
            model.parameters():
        dist.broadcast(param.data, src=src, group=group)


def pretrain_fbd(args, train_valid_test_dataset_provider, model_provider,
                 forward_step_func):
    from ..core.distributed import (
        DistributedDataParallel, DistributedDataParallelConfig)
    from ..core.optimizer import OptimizerConfig, get_megatron_optimizer

    is_forward = parallel_state.is_forward_stage()
    dual = parallel_state.get_forward_backward_parallel_dual_rank()
    config = core_transformer_config_from_args(args)

    model = get_model(model_provider, wrap_with_ddp=not is_forward, args=args)
    chunk = model[0]

    optimizer = None
    opt_param_scheduler = None
    if not is_forward:
        opt_config = OptimizerConfig(
            lr=args.lr, weight_decay=args.weight_decay,
            adam_beta1=args.adam_beta1, adam_beta2=args.adam_beta2,
            adam_eps=args.adam_eps, clip_grad=args.clip_grad,
            fp16=args.fp16, bf16=args.bf16)
        optimizer = get_megatron_optimizer(opt_config, model)
        opt_param_scheduler = get_optimizer_param_scheduler(optimizer, args)

    train_it, valid_it, test_it = _build_train_valid_test_data_iterators(
        train_valid_test_dataset_provider, args)

    args.iteration = 0
    iteration = 0
    t0 = time.time()
    while iteration < args.train_iters:
        if not is_forward:
            chunk.zero_grad_buffer()
            optimizer.zero_grad()
        losses = forward_backward_disaggregated(
            forward_step_func=forward_step_func, data_iterator=train_it,
            model=model, num_microbatches=get_num_microbatches(),
            seq_length=args.seq_length,
            micro_batch_size=args.micro_batch_size, forward_only=False)
        if not is_forward:
            ok, grad_norm, _ = optimizer.step()
            opt_param_scheduler.step(increment=1)
        # keep the forward instance's weights in lockstep
        _sync_params_to_forward(chunk, dual, is_forward)
        iteration += 1
        if (not is_forward and losses and
                parallel_state.is_pipeline_last_stage() and
                iteration % args.log_interval == 0):
            loss = torch.stack([d["lm loss"] for d in losses]).mean().item()
            elapsed = (time.time() - t0) / args.log_interval
            t0 = time.time()
            print(f" [FBD] iteration {iteration:6d}/{args.train_iters} | "
                  f"lm loss: {loss:.6f} | elapsed/iter: {elapsed*1000:.1f} ms",
                  flush=True)
    print_rank_0("[FBD] training complete")
    return model
