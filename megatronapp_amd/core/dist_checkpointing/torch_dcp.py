"""torch-DCP (torch.distributed.checkpoint) on-disk format for sharded
checkpoints — the reference's ``torch_dist`` layout (``__N_M.distcp``
data files + ``.metadata``), byte-compatible with upstream tooling
(reference core/dist_checkpointing/strategies/torch.py; the reference
saves non-sharded objects to common.pt the same way we do).

Our :class:`~..dist_checkpointing.mapping.ShardedTensor` carries a local
tensor plus (global_shape, global_offset, replica_id); these planners
translate that straight into DCP WriteItems/ReadItems, so saving goes
through torch's own FileSystemWriter (hence upstream-readable) and
loading resolves chunk intersections with torch's planner helpers
(hence cross-topology resharding for free).
"""

from __future__ import annotations

import dataclasses
import os
from typing import Any, Dict, List

import torch
import torch.distributed.checkpoint as dcp
from torch.distributed.checkpoint.default_planner import (
    create_default_global_save_plan)
from torch.distributed.checkpoint.metadata import (
    ChunkStorageMetadata, Metadata, MetadataIndex, TensorProperties,
    TensorStorageMetadata)
from torch.distributed.checkpoint.planner import (
    LoadPlan, LoadPlanner, ReadItem, SavePlan, SavePlanner, TensorWriteData,
    WriteItem, WriteItemType)
from torch.distributed.checkpoint.planner_helpers import (
    create_read_items_for_chunk_list)

from .mapping import ShardedTensor


class _MegatronSavePlanner(SavePlanner):
    """Plans WriteItems directly from our ShardedTensor shards."""

    def set_up_planner(self, state_dict, storage_meta=None,
                       is_coordinator=False):
        self.state_dict = state_dict
        self.is_coordinator = is_coordinator

    def create_local_plan(self) -> SavePlan:
        items: List[WriteItem] = []
        for fqn, st in self.state_dict.items():
            if st.replica_id != 0:
                continue  # fully-parallel dedup: one writer per shard
            t = st.data
            items.append(WriteItem(
                index=MetadataIndex(fqn, torch.Size(st.global_offset)),
                type=WriteItemType.SHARD,
                tensor_data=TensorWriteData(
                    chunk=ChunkStorageMetadata(
                        offsets=torch.Size(st.global_offset),
                        sizes=torch.Size(t.shape)),
                    properties=TensorProperties(dtype=t.dtype),
                    size=torch.Size(st.global_shape)),
            ))
        self.plan = SavePlan(items)
        return self.plan

    def create_global_plan(self, all_plans):
        plans, metadata = create_default_global_save_plan(
            all_plans, rewrite_index_hints=True)
        self.global_metadata = metadata
        return plans, metadata

    def finish_plan(self, new_plan: SavePlan) -> SavePlan:
        self.plan = new_plan
        return new_plan

    def resolve_data(self, write_item: WriteItem):
        st = self.state_dict[write_item.index.fqn]
        return st.data.detach().contiguous()


class _MegatronLoadPlanner(LoadPlanner):
    """Plans ReadItems for each local shard window; torch's chunk-list
    helper computes the intersections with whatever sharding the
    checkpoint was saved at (cross-topology reshard)."""

    def set_up_planner(self, state_dict, metadata=None,
                       is_coordinator=False):
        self.state_dict = state_dict
        self.metadata = metadata
        self.is_coordinator = is_coordinator

    def create_local_plan(self) -> LoadPlan:
        items: List[ReadItem] = []
        for fqn, st in self.state_dict.items():
            md = self.metadata.state_dict_metadata.get(fqn)
            if md is None:
                raise KeyError(f"{fqn} not present in checkpoint")
            chunk = ChunkStorageMetadata(
                offsets=torch.Size(st.global_offset),
                sizes=torch.Size(st.data.shape))
            items.extend(create_read_items_for_chunk_list(fqn, md, [chunk]))
        self.plan = LoadPlan(items)
        return self.plan

    def create_global_plan(self, global_plan):
        return global_plan

    def finish_plan(self, central_plan: LoadPlan) -> LoadPlan:
        return central_plan

    def resolve_tensor(self, read_item: ReadItem) -> torch.Tensor:
        st = self.state_dict[read_item.dest_index.fqn]
        t = st.data
        # dest_offsets are relative to the GLOBAL tensor; translate into
        # this shard's local window
        view = t
        for d, (off, length) in enumerate(
                zip(read_item.dest_offsets, read_item.lengths)):
            view = view.narrow(d, off, length)
        return view

    def commit_tensor(self, read_item: ReadItem, tensor: torch.Tensor):
        pass  # resolve_tensor returned a view into the live tensor

    def load_bytes(self, read_item, value):
        raise NotImplementedError("byte objects go to common.pt")

    def resolve_bytes(self, read_item):
        raise NotImplementedError("byte objects go to common.pt")


def save_dcp(sharded_state_dict: Dict[str, ShardedTensor], ckpt_dir: str,
             common_state: dict = None,
             process_group=None) -> None:
    """Write a torch-DCP checkpoint (``.metadata`` + ``__N_M.distcp``)."""
    os.makedirs(ckpt_dir, exist_ok=True)
    dcp.save(
        sharded_state_dict,
        storage_writer=dcp.FileSystemWriter(ckpt_dir),
        planner=_MegatronSavePlanner(),
        process_group=process_group,
    )
    if common_state is not None:
        rank = (torch.distributed.get_rank(process_group)
                if torch.distributed.is_initialized() else 0)
        if rank == 0:
            torch.save(common_state, os.path.join(ckpt_dir, "common.pt"))
        if torch.distributed.is_initialized():
            torch.distributed.barrier(group=process_group)


def load_dcp(sharded_state_dict: Dict[str, ShardedTensor], ckpt_dir: str,
             process_group=None) -> None:
    """Fill each ShardedTensor's local window from a torch-DCP
    checkpoint (ours or reference/upstream-produced)."""
    dcp.load(
        sharded_state_dict,
        storage_reader=dcp.FileSystemReader(ckpt_dir),
        planner=_MegatronLoadPlanner(),
        process_group=process_group,
    )


def is_dcp_checkpoint(ckpt_dir: str) -> bool:
    return os.path.exists(os.path.join(ckpt_dir, ".metadata"))


def load_dcp_consolidated(ckpt_dir: str) -> Dict[str, torch.Tensor]:
    """Read an entire DCP checkpoint into full (unsharded) tensors —
    the import path for reference-produced checkpoints on one process."""
    reader = dcp.FileSystemReader(ckpt_dir)
    md = reader.read_metadata()
    state: Dict[str, Any] = {}
    sharded: Dict[str, ShardedTensor] = {}
    for fqn, tmd in md.state_dict_metadata.items():
        if not isinstance(tmd, TensorStorageMetadata):
            continue
        t = torch.empty(tuple(tmd.size), dtype=tmd.properties.dtype)
        sharded[fqn] = ShardedTensor(
            key=fqn, data=t, global_shape=tuple(tmd.size),
            global_offset=(0,) * len(tmd.size), replica_id=0)
    load_dcp(sharded, ckpt_dir)
    for fqn, st in sharded.items():
        state[fqn] = st.data
    return state
