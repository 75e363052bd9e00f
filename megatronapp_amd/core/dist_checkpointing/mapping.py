"""ShardedTensor mapping (reference core/dist_checkpointing/mapping.py).

A ShardedTensor records where a rank's local tensor shard lives inside
the global parameter: key, global shape, offset and the sharded axis.
TP-sharded params derive their offset from ``param.partition_dim`` and
the TP rank; PP-sharded modules contribute disjoint keys (layer indices
are globalized by the caller's prefix); replicated params carry
``replica_id`` = dp rank so exactly one replica is persisted.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Optional, Tuple

import torch

from .. import parallel_state


@dataclass
class ShardedTensor:
    key: str
    data: torch.Tensor
    global_shape: Tuple[int, ...]
    global_offset: Tuple[int, ...]
    replica_id: int = 0

    @classmethod
    def from_rank_offsets(cls, key: str, data: torch.Tensor, *rank_offsets,
                          replica_id: int = 0):
        """rank_offsets: (axis, rank, world) triples (reference API)."""
        global_shape = list(data.shape)
        offset = [0] * data.dim()
        for axis, rank, world in rank_offsets:
            global_shape[axis] = data.shape[axis] * world
            offset[axis] = data.shape[axis] * rank
        return cls(key, data, tuple(global_shape), tuple(offset), replica_id)


def sharded_tensor_for_param(name: str, param: torch.Tensor) -> ShardedTensor:
    """Wrap a (possibly TP-sharded) parameter."""
    tp_rank = parallel_state.get_tensor_model_parallel_rank()
    tp_world = parallel_state.get_tensor_model_parallel_world_size()
    dp_rank = parallel_state.get_data_parallel_rank()
    is_tp = getattr(param, "tensor_model_parallel", False) and tp_world > 1
    if is_tp:
        axis = getattr(param, "partition_dim", 0)
        st = ShardedTensor.from_rank_offsets(
            name, param.data, (axis, tp_rank, tp_world), replica_id=dp_rank)
    else:
        # replicated across TP: tp rank 0 persists it
        replica = dp_rank * max(tp_world, 1) + tp_rank
        st = ShardedTensor(name, param.data, tuple(param.shape),
                           (0,) * max(param.dim(), 1), replica_id=replica)
    return st


def module_sharded_state_dict(module: torch.nn.Module, prefix: str = ""
                              ) -> Dict[str, ShardedTensor]:
    out: Dict[str, ShardedTensor] = {}
    for name, param in module.named_parameters():
        key = prefix + name
        out[key] = sharded_tensor_for_param(key, param)
    for name, buf in module.named_buffers():
        if buf is None:
            continue
        key = prefix + name
        out[key] = ShardedTensor(key, buf, tuple(buf.shape),
                                 (0,) * max(buf.dim(), 1),
                                 replica_id=parallel_state.get_data_parallel_rank())
    return out
