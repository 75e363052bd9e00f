"""Core utilities (subset of reference core/utils.py:1734).

Includes: viewless-tensor helpers (pipeline deallocation correctness),
GlobalMemoryBuffer (reused communication scratch), CP batch slicing with
causal load balance (reference :1704), param-hash DP check (reference
check_param_hashes_across_dp_replicas).
"""

from __future__ import annotations

import operator
from functools import reduce
from typing import List, Optional

import torch
import torch.distributed as dist

from . import parallel_state


class MakeViewlessTensor(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inp, requires_grad):
        out = torch.empty((1,), dtype=inp.dtype, device=inp.device,
                          requires_grad=requires_grad)
        out.data = inp.data
        return out

    @staticmethod
    def backward(ctx, grad_output):
        return grad_output, None


def make_viewless_tensor(inp, requires_grad, keep_graph):
    if inp._base is None:
        return inp
    if keep_graph:
        return MakeViewlessTensor.apply(inp, requires_grad)
    out = torch.empty((1,), dtype=inp.dtype, device=inp.device,
                      requires_grad=requires_grad)
    out.data = inp.data
    return out


def assert_viewless_tensor(tensor, extra_msg=None):
    if isinstance(tensor, list):
        for t in tensor:
            assert_viewless_tensor(t)
        return tensor
    if not isinstance(tensor, torch.Tensor):
        return tensor
    assert tensor._base is None, f"viewed tensor where viewless expected: {extra_msg}"
    return tensor


def safely_set_viewless_tensor_data(tensor, new_data_tensor):
    assert_viewless_tensor(tensor)
    tensor.data = new_data_tensor


class GlobalMemoryBuffer:
    """Reused scratch for collectives (reference core/utils.py:377)."""

    def __init__(self):
        self.buffer = {}

    def get_tensor(self, tensor_shape, dtype, name):
        required_len = reduce(operator.mul, tensor_shape, 1)
        key = (name, dtype)
        buf = self.buffer.get(key)
        if buf is None or buf.numel() < required_len:
            device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
            buf = torch.empty(required_len, dtype=dtype, device=device,
                              requires_grad=False)
            self.buffer[key] = buf
        return buf[0:required_len].view(*tensor_shape)


_GLOBAL_MEMORY_BUFFER = GlobalMemoryBuffer()


def get_global_memory_buffer():
    return _GLOBAL_MEMORY_BUFFER


def get_batch_on_this_cp_rank(batch: dict) -> dict:
    """Causal-load-balanced context-parallel sequence slicing: CP rank i
    takes chunks i and 2*cp-1-i of 2*cp chunks (reference :1704)."""
    cp = parallel_state.get_context_parallel_world_size()
    if cp == 1:
        return batch
    cp_rank = parallel_state.get_context_parallel_rank()
    out = {}
    for key, val in batch.items():
        if val is None or not isinstance(val, torch.Tensor):
            out[key] = val
            continue
        seq_dim = 1 if key != "attention_mask" else 2
        if val.shape[seq_dim] % (2 * cp) != 0:
            out[key] = val
            continue
        v = val.view(*val.shape[0:seq_dim], 2 * cp,
                     val.shape[seq_dim] // (2 * cp), *val.shape[seq_dim + 1:])
        index = torch.tensor([cp_rank, (2 * cp - cp_rank - 1)],
                             device=val.device, dtype=torch.long)
        v = v.index_select(seq_dim, index)
        out[key] = v.view(*val.shape[0:seq_dim], -1, *val.shape[seq_dim + 2:])
    return out


def check_param_hashes_across_dp_replicas(model_chunks: List[torch.nn.Module],
                                          cross_check: bool = True) -> bool:
    """All-gather per-param hashes across DP and compare (silent-corruption
    detector; reference core/utils.py check_param_hashes_across_dp_replicas)."""
    dp_group = parallel_state.get_data_parallel_group()
    dp_world = dist.get_world_size(dp_group)
    if dp_world == 1:
        return True
    ok = True
    for chunk in model_chunks:
        for name, param in chunk.named_parameters():
            local = param.detach().float().sum().unsqueeze(0)
            gathered = [torch.empty_like(local) for _ in range(dp_world)]
            dist.all_gather(gathered, local, group=dp_group)
            ref = gathered[0]
            for g in gathered[1:]:
                if not torch.equal(ref, g):
                    ok = False
    return ok


def divide(a, b):
    assert a % b == 0
    return a // b


def get_model_config(model: torch.nn.Module):
    m = model
    while hasattr(m, "module"):
        if hasattr(m, "config"):
            break
        m = m.module
    return m.config


def log_single_rank(logger, level, msg, *args, rank: int = 0, **kwargs):
    if not dist.is_initialized() or dist.get_rank() == rank:
        logger.log(level, msg, *args, **kwargs)


def unwrap_model(model, module_instances=None):
    return_list = True
    if not isinstance(model, list):
        model = [model]
        return_list = False
    unwrapped = []
    for m in model:
        while hasattr(m, "module"):
            m = m.module
        unwrapped.append(m)
    if not return_list:
        return unwrapped[0]
    return unwrapped


def num_floating_point_operations(args, batch_size):
    """Approximate FLOPs per iteration (reference training/utils).

    6 * params * tokens for the dense stack + attention quadratic term.
    """
    h = args.hidden_size
    L = args.num_layers
    s = args.seq_length
    v = args.padded_vocab_size
    ffn = getattr(args, "ffn_hidden_size", 4 * h)
    gated = 3 if getattr(args, "swiglu", False) else 2
    per_layer = 4 * h * h + gated * h * ffn
    dense = L * per_layer + v * h
    attn_quad = L * 2 * s * h  # score+context matmuls per token
    return 6 * batch_size * s * (dense + attn_quad)
