"""Training runtime: pretrain() and the train loop.

Reference: training/training.py (pretrain :894, get_model :1043,
train :1967, train_step :1367, training_log :1488, evaluate :2371).
MegaScan iteration hooks wrap each step (reference :2167-2292); MegaScope's
training WS server gates steps when --enable-ws-server (reference
:1991-2023); MegaDPP/MegaFBD schedule selection happens in
core.pipeline_parallel / fbd.
"""

from __future__ import annotations

import gc
import sys
import time
from typing import Callable, List, Optional

import torch
import torch.distributed as dist

from ..core import parallel_state
from ..core.distributed import (
    DistributedDataParallel,
    DistributedDataParallelConfig,
)
from ..core.enums import ModelType
from ..core.num_microbatches_calculator import (
    get_current_global_batch_size,
    get_num_microbatches,
    update_num_microbatches,
)
from ..core.optimizer import OptimizerConfig, get_megatron_optimizer
from ..core.optimizer.optimizer_param_scheduler import OptimizerParamScheduler
from ..core.pipeline_parallel import get_forward_backward_func
from ..core.utils import check_param_hashes_across_dp_replicas, num_floating_point_operations
from .arguments import core_transformer_config_from_args
from .checkpointing import load_checkpoint, save_checkpoint
from .global_vars import get_args, get_timers, get_tokenizer, get_tracer
from .initialize import initialize_megatron


def print_rank_0(msg):
    if not dist.is_initialized() or dist.get_rank() == 0:
        print(msg, flush=True)


def print_rank_last(msg):
    if not dist.is_initialized() or dist.get_rank() == dist.get_world_size() - 1:
        print(msg, flush=True)


def get_model(model_provider_func, model_type=ModelType.encoder_or_decoder,
              wrap_with_ddp=True, args=None):
    """Build model chunk(s) for this rank (reference training.py:1043)."""
    args = args or get_args()
    vpp = args.virtual_pipeline_model_parallel_size
    model = []
    if vpp is not None and parallel_state.get_pipeline_model_parallel_world_size() > 1:
        for v in range(vpp):
            parallel_state.set_virtual_pipeline_model_parallel_rank(v)
            pre = parallel_state.is_pipeline_first_stage()
            post = parallel_state.is_pipeline_last_stage()
            chunk = model_provider_func(pre_process=pre, post_process=post,
                                        vp_stage=v)
            model.append(chunk)
        parallel_state.set_virtual_pipeline_model_parallel_rank(0)
    else:
        pre = parallel_state.is_pipeline_first_stage()
        post = parallel_state.is_pipeline_last_stage()
        model.append(model_provider_func(pre_process=pre, post_process=post))

    for chunk in model:
        for param in chunk.parameters():
            if torch.cuda.is_available():
                param.data = param.data.cuda(torch.cuda.current_device())

    num_params = sum(p.nelement() for c in model for p in c.parameters())
    print_rank_0(f" > number of parameters on (tp, pp) rank "
                 f"({parallel_state.get_tensor_model_parallel_rank()}, "
                 f"{parallel_state.get_pipeline_model_parallel_rank()}): "
                 f"{num_params}")

    if getattr(args, "fp8", None) and torch.cuda.is_available():
        from ..core.fp8 import enable_fp8_training
        n8 = sum(enable_fp8_training(chunk) for chunk in model)
        print_rank_0(f" > fp8 training: {n8} linears flagged ({args.fp8})")

    if wrap_with_ddp:
        config = core_transformer_config_from_args(args)
        ddp_config = DistributedDataParallelConfig(
            grad_reduce_in_fp32=args.accumulate_allreduce_grads_in_fp32,
            overlap_grad_reduce=args.overlap_grad_reduce,
            use_distributed_optimizer=args.use_distributed_optimizer,
            bucket_size=args.ddp_bucket_size)
        model = [DistributedDataParallel(config, ddp_config, chunk,
                                         disable_bucketing=(i > 0))
                 for i, chunk in enumerate(model)]
    return model


def get_optimizer_param_scheduler(optimizer, args=None):
    args = args or get_args()
    if args.train_iters is None and args.train_samples:
        args.train_iters = args.train_samples // args.global_batch_size
    return OptimizerParamScheduler(
        optimizer, init_lr=0.0, max_lr=args.lr or 0.0, min_lr=args.min_lr,
        lr_warmup_steps=args.lr_warmup_iters,
        lr_decay_steps=args.lr_decay_iters or (args.train_iters or 1),
        lr_decay_style=args.lr_decay_style,
        start_wd=args.start_weight_decay, end_wd=args.end_weight_decay,
        wd_incr_steps=args.train_iters or 0, wd_incr_style="constant",
        wsd_decay_steps=args.lr_wsd_decay_iters,
        lr_wsd_decay_style=args.lr_wsd_decay_style)


def setup_model_and_optimizer(model_provider_func, model_type,
                              no_wd_decay_cond=None, scale_lr_cond=None,
                              lr_mult=1.0, args=None):
    args = args or get_args()
    model = get_model(model_provider_func, model_type, args=args)
    opt_config = OptimizerConfig(
        optimizer=args.optimizer, lr=args.lr, min_lr=args.min_lr,
        weight_decay=args.weight_decay, fp16=args.fp16, bf16=args.bf16,
        adam_beta1=args.adam_beta1, adam_beta2=args.adam_beta2,
        adam_eps=args.adam_eps, clip_grad=args.clip_grad,
        loss_scale=args.loss_scale,
        initial_loss_scale=args.initial_loss_scale,
        min_loss_scale=args.min_loss_scale,
        loss_scale_window=args.loss_scale_window,
        hysteresis=args.hysteresis,
        use_distributed_optimizer=args.use_distributed_optimizer,
        overlap_param_gather=args.overlap_param_gather,
        log_num_zeros_in_grad=args.log_num_zeros_in_grad,
        use_precision_aware_optimizer=getattr(
            args, "use_precision_aware_optimizer", False),
        exp_avg_dtype=getattr(args, "exp_avg_dtype", "fp32"),
        exp_avg_sq_dtype=getattr(args, "exp_avg_sq_dtype", "fp32"))
    optimizer = get_megatron_optimizer(opt_config, model)
    if args.fp16 and hasattr(optimizer, "scale_loss"):
        # fp16: schedules scale the loss before backward; the optimizer
        # unscales gradients and skips overflowed steps
        config = core_transformer_config_from_args(args)
        for chunk in model:
            mod = chunk.module if hasattr(chunk, "module") else chunk
            mod.config.grad_scale_func = optimizer.scale_loss
    opt_param_scheduler = get_optimizer_param_scheduler(optimizer, args)

    if args.load is not None:
        args.iteration, args.num_floating_point_operations_so_far = \
            load_checkpoint(model, optimizer, opt_param_scheduler)
    else:
        args.iteration = 0
        args.num_floating_point_operations_so_far = 0
    # a NEWER node-local non-persistent checkpoint (crash restart) wins
    if getattr(args, "non_persistent_save_interval", None) or \
            getattr(args, "non_persistent_ckpt_dir", None):
        from .checkpointing import load_local_checkpoint_if_newer
        li = load_local_checkpoint_if_newer(model, optimizer,
                                            opt_param_scheduler,
                                            args.iteration)
        if li is not None:
            args.iteration = li
    return model, optimizer, opt_param_scheduler


def train_step(forward_step_func, data_iterator, model, optimizer,
               opt_param_scheduler, config, args=None):
    """One optimizer step (reference training.py:1367), wrapped by the
    rerun state machine (reference training.py:1387-1410)."""
    args = args or get_args()
    from ..core.rerun_state_machine import get_rerun_state_machine
    rerun = get_rerun_state_machine()

    if hasattr(optimizer, "finish_param_sync"):
        optimizer.finish_param_sync()   # overlap-param-gather from last step
    fb_func = get_forward_backward_func()
    losses_reduced = []
    while rerun.should_run_forward_backward(data_iterator):
        for chunk in model:
            chunk.zero_grad_buffer()
        optimizer.zero_grad()
        losses_reduced = fb_func(
            forward_step_func=forward_step_func, data_iterator=data_iterator,
            model=model if len(model) > 1 else model[0],
            num_microbatches=get_num_microbatches(),
            seq_length=args.seq_length,
            micro_batch_size=args.micro_batch_size,
            forward_only=False)
        if losses_reduced and "lm loss" in losses_reduced[0]:
            rerun.record_result(sum(
                float(d["lm loss"]) for d in losses_reduced))
        else:
            rerun.record_result(0.0)

    tracer = get_tracer()
    if tracer is not None and tracer.is_tracing_active():
        with tracer.scope("optimizer"):
            ok, grad_norm, num_zeros = optimizer.step()
    else:
        ok, grad_norm, num_zeros = optimizer.step()
    if ok:
        increment = (get_num_microbatches() * args.micro_batch_size *
                     args.data_parallel_size)
        opt_param_scheduler.step(increment=1)
        skipped_iter = 0
    else:
        skipped_iter = 1

    loss_reduced = {}
    if parallel_state.is_pipeline_last_stage(ignore_virtual=True) and losses_reduced:
        for key in losses_reduced[0]:
            vals = [d[key] for d in losses_reduced if key in d]
            loss_reduced[key] = torch.stack(
                [v if torch.is_tensor(v) else torch.tensor(v) for v in vals]
            ).mean().item()
    return loss_reduced, skipped_iter, grad_norm, num_zeros


def training_log(loss_dict, total_loss_dict, learning_rate, iteration,
                 loss_scale, report_memory_flag, skipped_iter, grad_norm,
                 params_norm, num_zeros_in_grad, elapsed_per_iter, args=None):
    args = args or get_args()
    if iteration % args.log_interval != 0:
        return report_memory_flag
    throughput_tps = (args.global_batch_size * args.seq_length /
                      max(elapsed_per_iter, 1e-9))
    flops = num_floating_point_operations(args, args.global_batch_size)
    tflops = flops / max(elapsed_per_iter, 1e-9) / 1e12 / args.world_size
    loss_str = " | ".join(f"{k}: {v:.6f}" for k, v in loss_dict.items())
    print_rank_last(
        f" iteration {iteration:8d}/{args.train_iters} | "
        f"elapsed (ms): {elapsed_per_iter*1000:.1f} | tokens/s: "
        f"{throughput_tps:.0f} | TFLOP/s/GPU: {tflops:.1f} | lr: "
        f"{learning_rate:.3E} | {loss_str} | grad norm: "
        f"{grad_norm if grad_norm is not None else 0:.3f}"
        + (f" | params norm: {params_norm:.3f}"
           if params_norm is not None else ""))
    from .global_vars import get_tensorboard_writer
    writer = get_tensorboard_writer()
    if writer is not None:
        for k, v in loss_dict.items():
            writer.add_scalar(k, v, iteration)
        writer.add_scalar("learning-rate", learning_rate, iteration)
        writer.add_scalar("throughput", throughput_tps, iteration)
        if getattr(args, "log_timers_to_tensorboard", False):
            from .global_vars import get_timers
            timers = get_timers()
            if timers is not None:
                timers.write(list(timers._timers), writer, iteration)
    if report_memory_flag:
        # once, after optimizer state exists (reference training.py:1745)
        if (not torch.distributed.is_initialized()
                or torch.distributed.get_rank() == 0):
            from .theoretical_memory_usage import report_theoretical_memory
            try:
                report_theoretical_memory(
                    args, num_microbatches=get_num_microbatches())
            except Exception as e:  # never let a report kill training
                print(f"  (theoretical memory report failed: {e})")
            if torch.cuda.is_available():
                mem = torch.cuda.memory_stats()
                print(f"Measured memory (rank 0): allocated "
                      f"{mem.get('allocated_bytes.all.current', 0) / 2**20:.0f}"
                      f" MiB | reserved "
                      f"{mem.get('reserved_bytes.all.current', 0) / 2**20:.0f}"
                      f" MiB | max allocated "
                      f"{mem.get('allocated_bytes.all.peak', 0) / 2**20:.0f} MiB")
        report_memory_flag = False
    return report_memory_flag


def evaluate(forward_step_func, data_iterator, model, config, args=None,
             verbose=False):
    args = args or get_args()
    for chunk in model:
        chunk.eval()
    total = {}
    with torch.no_grad():
        for _ in range(args.eval_iters):
            fb_func = get_forward_backward_func()
            losses = fb_func(
                forward_step_func=forward_step_func,
                data_iterator=data_iterator,
                model=model if len(model) > 1 else model[0],
                num_microbatches=get_num_microbatches(),
                seq_length=args.seq_length,
                micro_batch_size=args.micro_batch_size, forward_only=True)
            if parallel_state.is_pipeline_last_stage(ignore_virtual=True):
                for d in losses:
                    for k, v in d.items():
                        total.setdefault(k, []).append(
                            v if torch.is_tensor(v) else torch.tensor(v))
    for chunk in model:
        chunk.train()
    return {k: torch.stack(v).mean().item() for k, v in total.items()}


def evaluate_and_print_results(prefix, forward_step_func, data_iterator,
                               model, iteration, process_non_loss_data_func,
                               config, verbose=False):
    results = evaluate(forward_step_func, data_iterator, model, config)
    string = " | ".join(f"{k} value: {v:.6E}" for k, v in results.items())
    print_rank_last(f" validation loss at {prefix} | {string}")
    return results


def _build_train_valid_test_data_iterators(
        build_train_valid_test_datasets_provider, args):
    """Per-rank dataloaders; data-parallel sharding via DistributedSampler."""
    if args.train_iters:
        train_samples = args.train_iters * args.global_batch_size
    else:
        train_samples = args.train_samples or args.global_batch_size
    eval_iters = (args.train_iters // args.eval_interval + 1) * args.eval_iters \
        if args.train_iters else args.eval_iters
    sizes = [train_samples, eval_iters * args.global_batch_size,
             args.eval_iters * args.global_batch_size]
    train_ds, valid_ds, test_ds = build_train_valid_test_datasets_provider(sizes)

    def make_iter(ds):
        if ds is None:
            return None
        dp_rank = parallel_state.get_data_parallel_rank()
        dp_world = parallel_state.get_data_parallel_world_size()
        sampler = torch.utils.data.distributed.DistributedSampler(
            ds, num_replicas=dp_world, rank=dp_rank, shuffle=False,
            drop_last=True)
        loader = torch.utils.data.DataLoader(
            ds, batch_size=args.micro_batch_size, sampler=sampler,
            num_workers=args.num_workers, drop_last=True, pin_memory=True,
            persistent_workers=args.num_workers > 0)

        def cyclic():
            while True:
                for batch in loader:
                    yield batch
        return cyclic()

    return make_iter(train_ds), make_iter(valid_ds), make_iter(test_ds)


def train(forward_step_func, model, optimizer, opt_param_scheduler,
          train_data_iterator, valid_data_iterator,
          process_non_loss_data_func, config, args, checkpointing_context=None):
    """Main loop (reference training.py:1967)."""
    for chunk in model:
        chunk.train()
    iteration = args.iteration
    total_loss_dict = {}
    report_memory_flag = True
    tracer = get_tracer()

    from ..core.rerun_state_machine import (
        RerunDataIterator, initialize_rerun_state_machine)
    initialize_rerun_state_machine(
        args.rerun_mode,
        args.rerun_validate_interval if args.rerun_mode != "disabled" else 0)
    if args.rerun_mode == "validate_results" and train_data_iterator is not None:
        if isinstance(train_data_iterator, list):
            train_data_iterator = [RerunDataIterator(it)
                                   for it in train_data_iterator]
        else:
            train_data_iterator = RerunDataIterator(train_data_iterator)

    from ..core.straggler_detector import StragglerDetector
    from ..core.utils import num_floating_point_operations as _nfpo
    straggler = StragglerDetector(
        args.straggler_report_interval if args.log_straggler else 0,
        _nfpo(args, args.global_batch_size) / args.world_size)

    ws_server = None
    if args.enable_ws_server and args.training_ws_port and \
            parallel_state.get_tensor_model_parallel_rank() == 0 and \
            parallel_state.get_pipeline_model_parallel_rank() == 0:
        from .training_wsserver import TrainingWSServer
        ws_server = TrainingWSServer(args.training_ws_port, args)
        ws_server.start()

    t_step = time.time()
    while iteration < args.train_iters:
        if ws_server is not None:
            ws_server.wait_for_step_and_broadcast()
        elif args.enable_ws_server and args.training_ws_port:
            from .training_wsserver import follower_sync_configs
            follower_sync_configs()

        update_num_microbatches(args.consumed_train_samples,
                                consistency_check=True)
        if tracer is not None:
            if dist.is_initialized():
                dist.barrier()
            tracer.iteration_begin(iteration)

        from . import ft_integration
        ft_integration.on_training_step_start()
        with straggler:
            loss_dict, skipped_iter, grad_norm, num_zeros = train_step(
                forward_step_func, train_data_iterator, model, optimizer,
                opt_param_scheduler, config, args)
        ft_integration.on_training_step_end()
        params_norm = None
        if args.log_params_norm:
            sq = sum(float(p.data.float().norm() ** 2)
                     for chunk in model for p in chunk.parameters())
            t = torch.tensor([sq])
            if dist.is_initialized() and \
                    parallel_state.get_model_parallel_group() is not None:
                dist.all_reduce(t, group=parallel_state.get_model_parallel_group())
            params_norm = float(t.sqrt())
        from ..core.rerun_state_machine import get_rerun_state_machine
        if get_rerun_state_machine().should_checkpoint_and_exit():
            print_rank_0("rerun state machine requested checkpoint + exit "
                         "(irreproducible result detected)")
            if args.save:
                save_checkpoint(iteration + 1, model, optimizer,
                                opt_param_scheduler)
            break
        iteration += 1
        args.curr_iteration = iteration
        args.consumed_train_samples += get_current_global_batch_size()
        if args.empty_unused_memory_level >= 1 and torch.cuda.is_available():
            # reference training.py empty_unused_memory: release cached
            # blocks between iterations (level 2: every iteration)
            if args.empty_unused_memory_level >= 2 or iteration % 50 == 0:
                torch.cuda.empty_cache()

        if tracer is not None:
            tracer.iteration_end()

        elapsed = time.time() - t_step
        t_step = time.time()
        lr = optimizer.param_groups[0]["lr"]
        report_memory_flag = training_log(
            loss_dict, total_loss_dict, lr, iteration, 1.0,
            report_memory_flag, skipped_iter, grad_norm, params_norm,
            num_zeros,
            elapsed, args)
        if ws_server is not None:
            ws_server.step_finished(iteration, loss_dict)

        if args.check_weight_hash_across_dp_replicas_interval and \
                iteration % args.check_weight_hash_across_dp_replicas_interval == 0:
            assert check_param_hashes_across_dp_replicas(model), \
                "parameter hashes differ across DP replicas"

        if args.eval_interval and iteration % args.eval_interval == 0 and \
                args.eval_iters > 0 and valid_data_iterator is not None:
            evaluate_and_print_results(
                f"iteration {iteration}", forward_step_func,
                valid_data_iterator, model, iteration, None, config)

        if args.save and args.save_interval and \
                iteration % args.save_interval == 0:
            save_checkpoint(iteration, model, optimizer, opt_param_scheduler,
                            args.num_floating_point_operations_so_far)

        if getattr(args, "non_persistent_save_interval", None) and \
                iteration % args.non_persistent_save_interval == 0:
            from .checkpointing import save_local_checkpoint
            save_local_checkpoint(iteration, model, optimizer,
                                  opt_param_scheduler)

        if args.exit_interval and iteration % args.exit_interval == 0:
            break
        from .global_vars import get_signal_handler
        sh = get_signal_handler()
        if sh is not None and any(sh.signals_received()):
            if args.save:
                save_checkpoint(iteration, model, optimizer,
                                opt_param_scheduler)
            print_rank_0("exiting on signal")
            break
    return iteration


def pretrain(train_valid_test_dataset_provider, model_provider,
             model_type=ModelType.encoder_or_decoder,
             forward_step_func=None, process_non_loss_data_func=None,
             extra_args_provider=None, args_defaults={},
             get_embedding_ranks=None, get_position_embedding_ranks=None,
             non_loss_data_func=None):
    """Main entry (reference training.py:894)."""
    args = initialize_megatron(extra_args_provider=extra_args_provider,
                               args_defaults=args_defaults)
    if getattr(args, "forward_backward_disaggregating", False):
        from ..fbd.runtime import pretrain_fbd
        return pretrain_fbd(args, train_valid_test_dataset_provider,
                            model_provider, forward_step_func)

    config = core_transformer_config_from_args(args)
    if getattr(args, "use_dpp", False):
        from ..dpp.transport import initialize_dpp
        initialize_dpp(args, config)
    model, optimizer, opt_param_scheduler = setup_model_and_optimizer(
        model_provider, model_type, args=args)

    train_it, valid_it, test_it = _build_train_valid_test_data_iterators(
        train_valid_test_dataset_provider, args)
    if args.virtual_pipeline_model_parallel_size is not None:
        # per-chunk iterators (schedule pulls once per chunk per microbatch)
        train_its = [train_it] + [
            _clone_iter(train_valid_test_dataset_provider, args)
            for _ in range(args.virtual_pipeline_model_parallel_size - 1)]
    else:
        train_its = train_it

    iteration = train(forward_step_func, model, optimizer,
                      opt_param_scheduler, train_its, valid_it, None, config,
                      args)

    if args.save:
        save_checkpoint(iteration, model, optimizer, opt_param_scheduler,
                        args.num_floating_point_operations_so_far)
    from .checkpointing import finalize_async_save
    finalize_async_save(blocking=True)

    if test_it is not None and args.eval_iters > 0:
        evaluate_and_print_results("the end of training", forward_step_func,
                                   test_it, model, iteration, None, config)
    tracer = get_tracer()
    if tracer is not None:
        tracer.shutdown()
    if getattr(args, "use_dpp", False):
        from ..dpp.transport import shutdown_dpp
        shutdown_dpp()
    return model


def _clone_iter(provider, args):
    train_it, _, _ = _build_train_valid_test_data_iterators(provider, args)
    return train_it
