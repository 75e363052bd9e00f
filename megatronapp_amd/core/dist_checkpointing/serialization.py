"""Sharded checkpoint save/load with cross-topology resharding.

Reference: core/dist_checkpointing/serialization.py + strategies
(torch_dist default, fully_parallel dedup, resharding).

Default on-disk format (round 2): **torch-DCP** — ``.metadata`` +
``__N_M.distcp`` data files written through torch.distributed.checkpoint
(see torch_dcp.py), byte-compatible with the reference's ``torch_dist``
strategy and loadable by upstream tooling.  Non-sharded objects live in
``common.pt`` (same as the reference).

The round-1 private layout (``shards_rank*.pt`` + ``index.json``) is
still READ transparently for old checkpoints; new saves are DCP-only.

Save: every rank persists the shards whose ``replica_id == 0`` (the
fully-parallel dedup — each unique shard written exactly once).
Load: each rank fills its local (offset, shape) window from whatever
sharding the checkpoint was saved at (cross-topology reshard).
"""

from __future__ import annotations

import json
import os
from typing import Dict

import torch
import torch.distributed as dist

from .mapping import ShardedTensor


def _rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def save(sharded_state_dict: Dict[str, ShardedTensor], ckpt_dir: str,
         common_state: dict = None) -> None:
    """Write a torch-DCP sharded checkpoint (upstream-compatible)."""
    from .torch_dcp import save_dcp
    save_dcp(sharded_state_dict, ckpt_dir, common_state=common_state)


def _save_legacy(sharded_state_dict: Dict[str, ShardedTensor],
                 ckpt_dir: str, common_state: dict = None) -> None:
    os.makedirs(ckpt_dir, exist_ok=True)
    rank = _rank()
    mine = {k: st for k, st in sharded_state_dict.items()
            if st.replica_id == 0}
    fname = f"shards_rank{rank:05d}.pt"
    payload = {k: {"offset": st.global_offset,
                   "global_shape": st.global_shape,
                   "tensor": st.data.detach().cpu()}
               for k, st in mine.items()}
    if payload:
        torch.save(payload, os.path.join(ckpt_dir, fname))

    # build the global index on rank 0
    entry = [(k, {"file": fname, "offset": list(st.global_offset),
                  "shape": list(st.data.shape),
                  "global_shape": list(st.global_shape)})
             for k, st in mine.items()]
    if dist.is_initialized() and dist.get_world_size() > 1:
        gathered = [None] * dist.get_world_size() if rank == 0 else None
        dist.gather_object(entry, gathered, dst=0)
        entries = [e for lst in (gathered or []) for e in lst]
    else:
        entries = entry
    if rank == 0:
        index: Dict[str, list] = {}
        for k, meta in entries:
            index.setdefault(k, []).append(meta)
        with open(os.path.join(ckpt_dir, "index.json"), "w") as f:
            json.dump(index, f)
        if common_state is not None:
            torch.save(common_state, os.path.join(ckpt_dir, "common.pt"))
    if dist.is_initialized():
        dist.barrier()


def load_common(ckpt_dir: str) -> dict:
    path = os.path.join(ckpt_dir, "common.pt")
    if os.path.exists(path):
        return torch.load(path, map_location="cpu", weights_only=False)
    return {}


def load(sharded_state_dict: Dict[str, ShardedTensor], ckpt_dir: str,
         strict: bool = True) -> Dict[str, "torch.Tensor"]:
    """Fill each local shard in-place (from a torch-DCP checkpoint, or a
    round-1 legacy index.json one) and return {key: filled tensor}
    (reference API: feedable to ``module.load_state_dict``)."""
    from .torch_dcp import is_dcp_checkpoint, load_dcp
    if is_dcp_checkpoint(ckpt_dir):
        sd = sharded_state_dict
        if not strict:
            import torch.distributed.checkpoint as _dcp
            md = _dcp.FileSystemReader(ckpt_dir).read_metadata()
            sd = {k: st for k, st in sd.items()
                  if k in md.state_dict_metadata}
        load_dcp(sd, ckpt_dir)
        return {k: st.data for k, st in sd.items()}
    return _load_legacy(sharded_state_dict, ckpt_dir, strict)


def _load_legacy(sharded_state_dict: Dict[str, ShardedTensor],
                 ckpt_dir: str, strict: bool = True):
    with open(os.path.join(ckpt_dir, "index.json")) as f:
        index = json.load(f)
    cache: Dict[str, dict] = {}

    def piece(fname):
        if fname not in cache:
            cache[fname] = torch.load(os.path.join(ckpt_dir, fname),
                                      map_location="cpu", weights_only=False)
        return cache[fname]

    for key, st in sharded_state_dict.items():
        metas = index.get(key)
        if metas is None:
            if strict:
                raise KeyError(f"checkpoint missing key {key}")
            continue
        want_off = list(st.global_offset)
        want_shape = list(st.data.shape)
        filled = 0
        for meta in metas:
            have_off = meta["offset"]
            have_shape = meta["shape"]
            # intersection window per dim
            lo = [max(a, b) for a, b in zip(want_off, have_off)]
            hi = [min(a + s, b + t) for a, s, b, t in
                  zip(want_off, want_shape, have_off, have_shape)]
            if any(h <= l for l, h in zip(lo, hi)):
                continue
            src = piece(meta["file"])[key]["tensor"]
            src_slices = tuple(slice(l - o, h - o)
                               for l, h, o in zip(lo, hi, have_off))
            dst_slices = tuple(slice(l - o, h - o)
                               for l, h, o in zip(lo, hi, want_off))
            st.data[dst_slices].copy_(src[src_slices].to(st.data.dtype))
            n = 1
            for l, h in zip(lo, hi):
                n *= h - l
            filled += n
        if strict and filled < st.data.numel():
            raise RuntimeError(
                f"{key}: only {filled}/{st.data.numel()} elements found "
                f"in checkpoint (topology mismatch?)")
    return {k: st.data for k, st in sharded_state_dict.items()}
