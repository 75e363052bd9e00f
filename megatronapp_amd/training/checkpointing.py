"""Checkpoint save/load (reference training/checkpointing.py:315).

Layout (reference-compatible, §2.6 SURVEY):
  <save>/iter_{it:07d}/mp_rank_{tp:02d}[_{pp:03d}]/model_optim_rng.pt
  <save>/latest_checkpointed_iteration.txt

"torch" format: one file per (tp, pp) rank written by dp rank 0 — the
reference's legacy layout, loadable by its conversion tooling.
"torch_dist" format: torch.distributed.checkpoint (DCP) sharded save with
per-rank key namespacing for TP/PP shards + async save option.
"""

from __future__ import annotations

import os
import random
import shutil
import sys
from typing import List, Optional

import numpy as np
import torch
import torch.distributed as dist

from ..core import parallel_state
from ..core.tensor_parallel.random import get_cuda_rng_tracker
from .global_vars import get_args


def get_checkpoint_name(checkpoints_path, iteration, release=False,
                        tp_rank=None, pp_rank=None, return_base_dir=False):
    if release:
        directory = "release"
    else:
        directory = f"iter_{iteration:07d}"
    base = os.path.join(checkpoints_path, directory)
    if return_base_dir:
        return base
    tp_rank = (parallel_state.get_tensor_model_parallel_rank()
               if tp_rank is None else tp_rank)
    pp_rank = (parallel_state.get_pipeline_model_parallel_rank()
               if pp_rank is None else pp_rank)
    if parallel_state.get_pipeline_model_parallel_world_size() == 1:
        common_path = os.path.join(base, f"mp_rank_{tp_rank:02d}")
    else:
        common_path = os.path.join(base, f"mp_rank_{tp_rank:02d}_{pp_rank:03d}")
    return os.path.join(common_path, "model_optim_rng.pt")


def get_checkpoint_tracker_filename(checkpoints_path):
    return os.path.join(checkpoints_path, "latest_checkpointed_iteration.txt")


def read_metadata(tracker_filename):
    with open(tracker_filename) as f:
        metastring = f.read().strip()
    release = metastring == "release"
    iteration = 0 if release else int(metastring)
    return iteration, release


def _rng_state():
    state = {
        "random_rng_state": random.getstate(),
        "np_rng_state": np.random.get_state(),
        "torch_rng_state": torch.get_rng_state(),
        "rng_tracker_states": get_cuda_rng_tracker().get_states(),
    }
    if torch.cuda.is_available():
        state["cuda_rng_state"] = torch.cuda.get_rng_state()
    return state


def _restore_rng_state(state):
    random.setstate(state["random_rng_state"])
    np.random.set_state(state["np_rng_state"])
    torch.set_rng_state(state["torch_rng_state"])
    if torch.cuda.is_available() and "cuda_rng_state" in state:
        torch.cuda.set_rng_state(state["cuda_rng_state"])
    get_cuda_rng_tracker().set_states(state["rng_tracker_states"])


_async_save_thread = None


def finalize_async_save(blocking: bool = True):
    """Join any in-flight background checkpoint write."""
    global _async_save_thread
    if _async_save_thread is not None and blocking:
        _async_save_thread.join()
        _async_save_thread = None


def save_local_checkpoint(iteration, model: List, optimizer,
                          opt_param_scheduler):
    """Non-persistent node-local checkpoint (reference
    nvidia_resiliency LocalCheckpointManager, training.py:702-726):
    a fast legacy-format save into local scratch, rotation depth 1, for
    quick in-place restart after a crash.  load_checkpoint prefers it
    when it is NEWER than the persistent checkpoint."""
    import shutil
    args = get_args()
    base = args.non_persistent_ckpt_dir or (
        os.path.join(args.save, "local_ckpt") if args.save else None)
    if base is None:
        return
    rank = dist.get_rank() if dist.is_initialized() else 0
    new_dir = os.path.join(base, f"iter_{iteration:07d}")
    os.makedirs(new_dir, exist_ok=True)
    sd = {"iteration": iteration, "checkpoint_version": 3.0}
    if len(model) == 1:
        sd["model"] = model[0].state_dict_for_save_checkpoint()
    else:
        for i, chunk in enumerate(model):
            sd[f"model{i}"] = chunk.state_dict_for_save_checkpoint()
    if optimizer is not None:
        sd["optimizer"] = optimizer.state_dict()
    if opt_param_scheduler is not None:
        sd["opt_param_scheduler"] = opt_param_scheduler.state_dict()
    sd["rng_state"] = _rng_state()
    torch.save(sd, os.path.join(new_dir, f"rank_{rank:05d}.pt"))
    if dist.is_initialized():
        dist.barrier()
    if rank == 0:
        with open(os.path.join(base, "latest"), "w") as f:
            f.write(str(iteration))
        # rotation depth 1: drop older local checkpoints
        for d in os.listdir(base):
            if d.startswith("iter_") and d != f"iter_{iteration:07d}":
                shutil.rmtree(os.path.join(base, d), ignore_errors=True)
    if dist.is_initialized():
        dist.barrier()


def load_local_checkpoint_if_newer(model: List, optimizer,
                                   opt_param_scheduler, persistent_iter):
    """Return the local iteration if a newer local checkpoint was
    loaded, else None."""
    args = get_args()
    base = args.non_persistent_ckpt_dir or (
        os.path.join(args.save, "local_ckpt") if args.save else None)
    if base is None or not os.path.exists(os.path.join(base, "latest")):
        return None
    with open(os.path.join(base, "latest")) as f:
        local_iter = int(f.read().strip())
    if local_iter <= persistent_iter:
        return None
    rank = dist.get_rank() if dist.is_initialized() else 0
    path = os.path.join(base, f"iter_{local_iter:07d}",
                        f"rank_{rank:05d}.pt")
    if not os.path.exists(path):
        return None
    sd = torch.load(path, map_location="cpu", weights_only=False)
    if len(model) == 1:
        model[0].load_state_dict(sd["model"], strict=True)
    else:
        for i, chunk in enumerate(model):
            chunk.load_state_dict(sd[f"model{i}"], strict=True)
    if optimizer is not None and "optimizer" in sd:
        optimizer.load_state_dict(sd["optimizer"])
    if opt_param_scheduler is not None and "opt_param_scheduler" in sd:
        opt_param_scheduler.load_state_dict(sd["opt_param_scheduler"])
    if sd.get("rng_state"):
        _restore_rng_state(sd["rng_state"])
    print(f"  loaded LOCAL (non-persistent) checkpoint at iteration "
          f"{local_iter}", flush=True)
    return local_iter


def save_checkpoint(iteration, model: List, optimizer, opt_param_scheduler,
                    num_floating_point_operations_so_far=0, checkpointing_context=None,
                    train_data_iterator=None, **kwargs):
    args = get_args()
    if args.save is None:
        return
    from . import ft_integration
    ft_integration.on_checkpointing_start()
    try:
        _save_checkpoint_impl(iteration, model, optimizer,
                              opt_param_scheduler,
                              num_floating_point_operations_so_far)
    finally:
        ft_integration.on_checkpointing_end()


def _save_checkpoint_impl(iteration, model, optimizer, opt_param_scheduler,
                          num_floating_point_operations_so_far):
    args = get_args()
    if getattr(args, "ckpt_format", "torch") == "torch_dist":
        return _save_checkpoint_torch_dist(
            iteration, model, optimizer, opt_param_scheduler,
            num_floating_point_operations_so_far)
    state_dict = {
        "args": vars(args).copy(),
        "checkpoint_version": 3.0,
        "iteration": iteration,
        "num_floating_point_operations_so_far": num_floating_point_operations_so_far,
    }
    # drop unpicklables from args copy
    state_dict["args"] = {k: v for k, v in state_dict["args"].items()
                          if isinstance(v, (int, float, str, bool, list, tuple,
                                            type(None)))}
    if len(model) == 1:
        state_dict["model"] = model[0].state_dict_for_save_checkpoint()
    else:
        for i, chunk in enumerate(model):
            state_dict[f"model{i}"] = chunk.state_dict_for_save_checkpoint()
    if optimizer is not None and not args.no_save_optim:
        state_dict["optimizer"] = optimizer.state_dict()
        if opt_param_scheduler is not None:
            state_dict["opt_param_scheduler"] = opt_param_scheduler.state_dict()
    if not args.no_save_rng:
        state_dict["rng_state"] = _rng_state()

    # dp rank 0 of every (tp, pp) writes; with the distributed optimizer
    # every dp rank holds a distinct shard -> append dp suffix
    dp_rank = parallel_state.get_data_parallel_rank()
    write = dp_rank == 0 or args.use_distributed_optimizer
    if write:
        name = get_checkpoint_name(args.save, iteration)
        if args.use_distributed_optimizer and dp_rank > 0:
            name = name.replace("model_optim_rng.pt",
                                f"optim_shard_dp{dp_rank:03d}.pt")
            state_dict = {"optimizer": state_dict.get("optimizer")}
        os.makedirs(os.path.dirname(name), exist_ok=True)
        if getattr(args, "async_save", False):
            # deep-copy tensors to CPU inline (cheap vs HBM), write in a
            # background thread so training resumes immediately
            import threading

            def _to_cpu(obj):
                if torch.is_tensor(obj):
                    return obj.detach().cpu().clone()
                if isinstance(obj, dict):
                    return {k: _to_cpu(v) for k, v in obj.items()}
                if isinstance(obj, (list, tuple)):
                    return type(obj)(_to_cpu(v) for v in obj)
                return obj

            snapshot = _to_cpu(state_dict)
            tracker = get_checkpoint_tracker_filename(args.save)

            def _write():
                torch.save(snapshot, name)
                with open(tracker, "w") as f:
                    f.write(str(iteration))

            global _async_save_thread
            finalize_async_save(blocking=True)
            _async_save_thread = threading.Thread(target=_write, daemon=False)
            _async_save_thread.start()
            return

        torch.save(state_dict, name)

    if dist.is_initialized():
        dist.barrier()
    if (not dist.is_initialized()) or dist.get_rank() == 0:
        with open(get_checkpoint_tracker_filename(args.save), "w") as f:
            f.write(str(iteration))
    if dist.is_initialized():
        dist.barrier()


def load_checkpoint(model: List, optimizer, opt_param_scheduler,
                    load_arg="load", strict=True, checkpointing_context=None,
                    skip_load_to_model_and_opt=False):
    args = get_args()
    load_dir = getattr(args, load_arg)
    if load_dir is None:
        return 0, 0
    tracker = get_checkpoint_tracker_filename(load_dir)
    if not os.path.isfile(tracker):
        if args.rank == 0:
            print(f"  no checkpoint tracker at {tracker}, starting fresh")
        return 0, 0
    iteration, release = read_metadata(tracker)
    # sharded (torch_dist) checkpoints: torch-DCP (.metadata) or the
    # round-1 legacy index.json layout
    base = get_checkpoint_name(load_dir, iteration, release,
                               return_base_dir=True)
    if os.path.exists(os.path.join(base, ".metadata")) or \
            os.path.exists(os.path.join(base, "index.json")):
        return _load_checkpoint_torch_dist(
            model, optimizer, opt_param_scheduler, base, iteration, strict)
    name = get_checkpoint_name(load_dir, iteration, release)
    state_dict = torch.load(name, map_location="cpu", weights_only=False)

    if len(model) == 1:
        model[0].load_state_dict(state_dict["model"], strict=strict)
    else:
        for i, chunk in enumerate(model):
            chunk.load_state_dict(state_dict[f"model{i}"], strict=strict)

    if optimizer is not None and not args.no_load_optim and not args.finetune:
        if args.use_distributed_optimizer and \
                parallel_state.get_data_parallel_rank() > 0:
            shard_name = name.replace(
                "model_optim_rng.pt",
                f"optim_shard_dp{parallel_state.get_data_parallel_rank():03d}.pt")
            opt_sd = torch.load(shard_name, map_location="cpu",
                                weights_only=False)["optimizer"]
        else:
            opt_sd = state_dict.get("optimizer")
        if opt_sd is not None:
            optimizer.load_state_dict(opt_sd)
        if opt_param_scheduler is not None and \
                "opt_param_scheduler" in state_dict:
            opt_param_scheduler.load_state_dict(
                state_dict["opt_param_scheduler"])
    if optimizer is not None and hasattr(optimizer, "reload_model_params") \
            and (args.no_load_optim or args.finetune):
        optimizer.reload_model_params()

    if not args.no_load_rng and not args.finetune and "rng_state" in state_dict:
        _restore_rng_state(state_dict["rng_state"])

    num_flop = state_dict.get("num_floating_point_operations_so_far", 0)
    if args.rank == 0:
        print(f"  loaded checkpoint from {load_dir} at iteration {iteration}")
    return (0 if args.finetune else iteration), num_flop


# ---------------------------------------------------------------------------
# torch_dist (sharded) format: model weights via dist_checkpointing with
# cross-topology resharding; optimizer shards + rng per-rank alongside.
# ---------------------------------------------------------------------------

def _globalize_layer_keys(sd, mod):
    """Rewrite local decoder layer indices to global layer numbers so PP
    stages contribute DISJOINT keys to the sharded index (stage 1's local
    layers.0 must not collide with stage 0's)."""
    import re
    block = getattr(mod, "decoder", None) or getattr(mod, "encoder", None)
    layers = getattr(block, "layers", None) if block is not None else None
    if layers is None:
        return sd

    def glob(m):
        local = int(m.group(1))
        layer = layers[local]
        g = getattr(layer, "layer_number", local + 1) - 1
        return f"layers.{g}."

    out = {}
    for key, st in sd.items():
        new_key = re.sub(r"layers\.(\d+)\.", glob, key)
        st.key = new_key
        out[new_key] = st
    return out


def _model_sharded_sd(model: List):
    from ..core.dist_checkpointing import module_sharded_state_dict
    sd = {}
    for chunk in model:
        mod = chunk
        while hasattr(mod, "module"):
            mod = mod.module
        # one namespace for every chunk: layer keys are globally unique
        # after _globalize_layer_keys, and embeddings / final norm / head
        # exist in exactly one chunk — so VPP checkpoints reshard to any
        # topology like plain PP ones
        chunk_sd = module_sharded_state_dict(mod, "model.")
        sd.update(_globalize_layer_keys(chunk_sd, mod))
    return sd


def _save_checkpoint_torch_dist(iteration, model, optimizer,
                                opt_param_scheduler, num_flop):
    from ..core.dist_checkpointing import save as dist_save
    args = get_args()
    base = get_checkpoint_name(args.save, iteration, return_base_dir=True)
    os.makedirs(base, exist_ok=True)
    common = {
        "args": {k: v for k, v in vars(args).items()
                 if isinstance(v, (int, float, str, bool, list, tuple,
                                   type(None)))},
        "iteration": iteration,
        "checkpoint_version": 3.0,
        "num_floating_point_operations_so_far": num_flop,
    }
    if opt_param_scheduler is not None:
        common["opt_param_scheduler"] = opt_param_scheduler.state_dict()
    if not args.no_save_rng:
        common["rng_state"] = _rng_state()
    single = (not dist.is_initialized()) or dist.get_world_size() == 1
    if getattr(args, "async_save", False) and single:
        # single-rank async: snapshot shards to CPU inline, write in a
        # background thread (multi-rank needs the barrier, stays sync)
        import threading
        sd = _model_sharded_sd(model)
        from ..core.dist_checkpointing.mapping import ShardedTensor
        cpu_sd = {k: ShardedTensor(st.key, st.data.detach().cpu().clone(),
                                   st.global_shape, st.global_offset,
                                   st.replica_id)
                  for k, st in sd.items()}
        opt_sd = (optimizer.state_dict()
                  if optimizer is not None and not args.no_save_optim
                  else None)
        tracker = get_checkpoint_tracker_filename(args.save)

        def _write():
            dist_save(cpu_sd, base, common_state=common)
            if opt_sd is not None:
                torch.save({"optimizer": opt_sd},
                           os.path.join(base, "optim_rank00000.pt"))
            with open(tracker, "w") as f:
                f.write(str(iteration))

        global _async_save_thread
        finalize_async_save(blocking=True)
        _async_save_thread = threading.Thread(target=_write, daemon=False)
        _async_save_thread.start()
        return
    dist_save(_model_sharded_sd(model), base, common_state=common)
    if optimizer is not None and not args.no_save_optim:
        rank = dist.get_rank() if dist.is_initialized() else 0
        torch.save({"optimizer": optimizer.state_dict()},
                   os.path.join(base, f"optim_rank{rank:05d}.pt"))
    if dist.is_initialized():
        dist.barrier()
    if (not dist.is_initialized()) or dist.get_rank() == 0:
        with open(get_checkpoint_tracker_filename(args.save), "w") as f:
            f.write(str(iteration))
    if dist.is_initialized():
        dist.barrier()


def _load_checkpoint_torch_dist(model, optimizer, opt_param_scheduler, base,
                                iteration, strict):
    from ..core.dist_checkpointing import load as dist_load, load_common
    args = get_args()
    dist_load(_model_sharded_sd(model), base, strict=strict)
    common = load_common(base)
    if optimizer is not None and not args.no_load_optim and not args.finetune:
        rank = dist.get_rank() if dist.is_initialized() else 0
        opt_path = os.path.join(base, f"optim_rank{rank:05d}.pt")
        if os.path.exists(opt_path):
            try:
                optimizer.load_state_dict(
                    torch.load(opt_path, map_location="cpu",
                               weights_only=False)["optimizer"])
            except (RuntimeError, ValueError, KeyError) as e:
                # cross-topology resume: model weights reshard through the
                # overlap-window loader, but optimizer shards are
                # per-topology — start the optimizer fresh
                if args.rank == 0:
                    print(f"  optimizer state does not match this topology "
                          f"({e}); reinitializing optimizer")
                if hasattr(optimizer, "reload_model_params"):
                    optimizer.reload_model_params()
        if opt_param_scheduler is not None and                 "opt_param_scheduler" in common:
            opt_param_scheduler.load_state_dict(common["opt_param_scheduler"])
    if optimizer is not None and hasattr(optimizer, "reload_model_params")             and (args.no_load_optim or args.finetune):
        optimizer.reload_model_params()
    if not args.no_load_rng and not args.finetune and "rng_state" in common:
        _restore_rng_state(common["rng_state"])
    if args.rank == 0:
        print(f"  loaded checkpoint (torch_dist) at iteration {iteration}")
    num_flop = common.get("num_floating_point_operations_so_far", 0)
    return (0 if args.finetune else iteration), num_flop
