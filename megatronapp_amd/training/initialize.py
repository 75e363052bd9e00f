"""Framework initialization (reference training/initialize.py:37).

initialize_megatron: parse args -> set globals -> init torch.distributed
(RCCL on GPU, gloo on CPU) -> initialize model parallel groups -> seed the
parallel RNG -> start the MegaScan tracer if --trace.
"""

from __future__ import annotations

import os
import random
from datetime import timedelta

import numpy as np
import torch
import torch.distributed as dist

from ..core import parallel_state
from ..core.num_microbatches_calculator import init_num_microbatches_calculator
from ..core.tensor_parallel.random import model_parallel_cuda_manual_seed
from .arguments import parse_args, validate_args
from .global_vars import set_global_variables


def initialize_megatron(extra_args_provider=None, args_defaults={},
                        ignore_unknown_args=False, allow_no_cuda=True,
                        skip_mpu_initialization=False,
                        parsed_args=None):
    args = parsed_args or parse_args(extra_args_provider, ignore_unknown_args)
    if getattr(args, "use_checkpoint_args", False) and args.load:
        _apply_checkpoint_args(args)
    validate_args(args, args_defaults)
    set_global_variables(args)

    if getattr(args, "deterministic_mode", False):
        # bitwise reproducibility: deterministic kernel selection plus a
        # pinned hipBLASLt workspace config (reference --deterministic-mode)
        torch.use_deterministic_algorithms(True, warn_only=True)
        os.environ.setdefault("CUBLAS_WORKSPACE_CONFIG", ":4096:8")
        os.environ.setdefault("HIPBLASLT_ALLOW_TF32", "0")
        if torch.cuda.is_available():
            torch.backends.cudnn.deterministic = True
            torch.backends.cudnn.benchmark = False

    _initialize_distributed(args)
    _set_random_seed(args.seed)

    init_num_microbatches_calculator(
        args.rank, args.rampup_batch_size, args.global_batch_size,
        args.micro_batch_size, args.data_parallel_size)

    if args.trace:
        from .trace import Tracer
        Tracer.initialize(trace_dir=args.trace_dir,
                          interval=args.trace_interval,
                          continuous_iters=args.continuous_trace_iterations,
                          granularity=args.trace_granularity,
                          max_iters=args.trace_max_iters)
    if getattr(args, "config_logger_dir", None):
        from ..core.config_logger import log_config_to_dir
        log_config_to_dir(args.config_logger_dir, rank=args.rank, args=args)
    from . import ft_integration
    ft_integration.setup(args)
    ft_integration.maybe_setup_simulated_fault(args)
    return args


def _load_comm_config(path):
    import yaml
    with open(path) as f:
        return yaml.safe_load(f) or {}


def _initialize_distributed(args):
    # per-group RCCL tuning (reference --nccl-communicator-config-path):
    # supported keys per group name (tp/dp/pp/ep/cp or 'default'):
    # NCCL_MIN_CTAS / NCCL_MAX_CTAS-style knobs exported before init so
    # RCCL picks them up; per-group ProcessGroupNCCL.Options are applied
    # for groups created after init through parallel_state.
    if getattr(args, "nccl_communicator_config_path", None):
        cfg = _load_comm_config(args.nccl_communicator_config_path)
        base = cfg.get("default", {})
        if "min_ctas" in base:
            os.environ.setdefault("NCCL_MIN_CTAS", str(base["min_ctas"]))
        if "max_ctas" in base:
            os.environ.setdefault("NCCL_MAX_CTAS", str(base["max_ctas"]))
        parallel_state.set_pg_comm_config(cfg)
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "6000")
        backend = args.distributed_backend
        if not torch.cuda.is_available():
            backend = "gloo"
        dist.init_process_group(
            backend=backend, world_size=args.world_size, rank=args.rank,
            timeout=timedelta(minutes=args.distributed_timeout_minutes))
    if torch.cuda.is_available():
        # modulo so oversubscribed test topologies (2 ranks, 1 GPU) run
        torch.cuda.set_device(args.local_rank % torch.cuda.device_count())

    if parallel_state.model_parallel_is_initialized():
        return
    if getattr(args, "forward_backward_disaggregating", False):
        from ..fbd.topology import initialize_model_parallel_fbd
        initialize_model_parallel_fbd(args)
        return
    parallel_state.initialize_model_parallel(
        tensor_model_parallel_size=args.tensor_model_parallel_size,
        pipeline_model_parallel_size=args.pipeline_model_parallel_size,
        virtual_pipeline_model_parallel_size=args.virtual_pipeline_model_parallel_size,
        context_parallel_size=args.context_parallel_size,
        expert_model_parallel_size=args.expert_model_parallel_size)


def _set_random_seed(seed: int, data_parallel_random_init=False):
    # PP-stage-offset seeds (reference initialize.py:399)
    seed = seed + 100 * parallel_state.get_pipeline_model_parallel_rank()
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        model_parallel_cuda_manual_seed(seed)
    else:
        model_parallel_cuda_manual_seed(seed)


def set_jit_fusion_options():
    """No-op on MI355X: fusion is done by our HIP kernels, not torch JIT."""


def write_args_to_tensorboard(args=None, writer=None):
    from .global_vars import get_args, get_tensorboard_writer
    args = args or get_args()
    writer = writer or get_tensorboard_writer()
    if writer is None:
        return
    for k, v in sorted(vars(args).items()):
        writer.add_text(k, str(v))


_CKPT_ARCH_ARGS = [
    "num_layers", "hidden_size", "ffn_hidden_size", "num_attention_heads",
    "num_query_groups", "group_query_attention", "kv_channels",
    "max_position_embeddings", "position_embedding_type", "normalization",
    "swiglu", "add_bias_linear", "untie_embeddings_and_output_weights",
    "vocab_size", "padded_vocab_size", "seq_length", "mtp_num_layers",
    "num_experts", "moe_router_topk"]


def _apply_checkpoint_args(args):
    """--use-checkpoint-args: architecture flags come from the checkpoint
    (reference checkpointing.load_args_from_checkpoint)."""
    import torch as _t
    from .checkpointing import (get_checkpoint_name,
                                get_checkpoint_tracker_filename,
                                read_metadata)
    tracker = get_checkpoint_tracker_filename(args.load)
    if not os.path.isfile(tracker):
        return
    iteration, release = read_metadata(tracker)
    base = get_checkpoint_name(args.load, iteration, release,
                               return_base_dir=True)
    saved = None
    if os.path.exists(os.path.join(base, "common.pt")):
        saved = _t.load(os.path.join(base, "common.pt"), map_location="cpu",
                        weights_only=False).get("args")
    else:
        legacy = os.path.join(base, "mp_rank_00", "model_optim_rng.pt")
        if os.path.exists(legacy):
            saved = _t.load(legacy, map_location="cpu",
                            weights_only=False).get("args")
    if not saved:
        return
    for key in _CKPT_ARCH_ARGS:
        if key in saved and saved[key] is not None:
            setattr(args, key, saved[key])
    if int(os.environ.get("RANK", "0")) == 0:
        # rank is set by validate_args later; use the env here
        print(f"  architecture args restored from checkpoint at "
              f"iteration {iteration}")
