"""Global singletons: args / tokenizer / timers / tracer / tb writer.

Reference: training/global_vars.py (:27-73)."""

from __future__ import annotations

from ..core.timers import Timers

_GLOBAL_ARGS = None
_GLOBAL_TOKENIZER = None
_GLOBAL_TIMERS = None
_GLOBAL_TENSORBOARD_WRITER = None
_GLOBAL_WANDB_WRITER = None
_GLOBAL_SIGNAL_HANDLER = None


def get_args():
    assert _GLOBAL_ARGS is not None, "args not initialized"
    return _GLOBAL_ARGS


def set_args(args):
    global _GLOBAL_ARGS
    _GLOBAL_ARGS = args


def get_tokenizer():
    return _GLOBAL_TOKENIZER


def set_tokenizer(tok):
    global _GLOBAL_TOKENIZER
    _GLOBAL_TOKENIZER = tok


def get_timers():
    return _GLOBAL_TIMERS


def get_tensorboard_writer():
    return _GLOBAL_TENSORBOARD_WRITER


def get_wandb_writer():
    return _GLOBAL_WANDB_WRITER


def get_tracer():
    from .trace import Tracer
    return Tracer.get()


def get_signal_handler():
    return _GLOBAL_SIGNAL_HANDLER


def set_global_variables(args, build_tokenizer=True):
    global _GLOBAL_TIMERS, _GLOBAL_TENSORBOARD_WRITER, _GLOBAL_SIGNAL_HANDLER
    set_args(args)
    _GLOBAL_TIMERS = Timers(args.timing_log_level, "minmax")
    if build_tokenizer:
        from .tokenizer import build_tokenizer as _bt
        set_tokenizer(_bt(args))
    if args.tensorboard_dir and args.rank == args.world_size - 1:
        try:
            from torch.utils.tensorboard import SummaryWriter
            _GLOBAL_TENSORBOARD_WRITER = SummaryWriter(
                log_dir=args.tensorboard_dir)
        except ImportError:
            # tensorboard is not installed in this image: fall back to a
            # JSONL scalar log with the same add_scalar surface so the
            # metrics are still captured under --tensorboard-dir
            _GLOBAL_TENSORBOARD_WRITER = _JsonlScalarWriter(
                args.tensorboard_dir)
    if getattr(args, "exit_signal_handler", False):
        from .dist_signal_handler import DistributedSignalHandler
        _GLOBAL_SIGNAL_HANDLER = DistributedSignalHandler().__enter__()


class _JsonlScalarWriter:
    """SummaryWriter-shaped JSONL fallback (scalars.jsonl per run)."""

    def __init__(self, log_dir):
        import os
        os.makedirs(log_dir, exist_ok=True)
        self._f = open(os.path.join(log_dir, "scalars.jsonl"), "a")

    def add_scalar(self, tag, value, step=None):
        import json
        self._f.write(json.dumps(
            {"tag": tag, "value": float(value), "step": step}) + "\n")
        self._f.flush()

    def flush(self):
        self._f.flush()

    def close(self):
        self._f.close()


def unset_global_variables():
    global _GLOBAL_ARGS, _GLOBAL_TOKENIZER, _GLOBAL_TIMERS
    global _GLOBAL_TENSORBOARD_WRITER, _GLOBAL_WANDB_WRITER
    _GLOBAL_ARGS = None
    _GLOBAL_TOKENIZER = None
    _GLOBAL_TIMERS = None
    _GLOBAL_TENSORBOARD_WRITER = None
    _GLOBAL_WANDB_WRITER = None
