"""Broadcast of the data batch within a TP group (reference data.py:69)."""

from __future__ import annotations

import torch
import torch.distributed as dist

from .. import parallel_state

_MAX_DATA_DIM = 5


def broadcast_data(keys, data, datatype):
    """Broadcast data dict from TP rank 0 to the whole TP group."""
    tp_world = parallel_state.get_tensor_model_parallel_world_size()
    if tp_world == 1:
        return {k: data[k].to(datatype) if data is not None else None for k in keys} \
            if data is not None else {}

    src = parallel_state.get_tensor_model_parallel_src_rank()
    group = parallel_state.get_tensor_model_parallel_group()
    rank = dist.get_rank()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    # pack sizes
    sizes = []
    if rank == src:
        for key in keys:
            t = data[key]
            assert t.dim() <= _MAX_DATA_DIM
            sizes.extend(list(t.size()) + [-1] * (_MAX_DATA_DIM - t.dim()))
    else:
        sizes = [0] * (len(keys) * _MAX_DATA_DIM)
    sizes_t = torch.tensor(sizes, dtype=torch.long, device=device)
    dist.broadcast(sizes_t, src, group=group)
    sizes = sizes_t.tolist()

    shapes, numels, total = {}, {}, 0
    for i, key in enumerate(keys):
        shape = [s for s in sizes[i * _MAX_DATA_DIM:(i + 1) * _MAX_DATA_DIM] if s >= 0]
        shapes[key] = shape
        n = 1
        for s in shape:
            n *= s
        numels[key] = n
        total += n

    if rank == src:
        flat = torch.cat([data[k].contiguous().view(-1).to(device, datatype)
                          for k in keys], dim=0)
    else:
        flat = torch.empty(total, dtype=datatype, device=device)
    dist.broadcast(flat, src, group=group)

    out, offset = {}, 0
    for key in keys:
        out[key] = flat[offset:offset + numels[key]].view(shapes[key])
        offset += numels[key]
    return out
