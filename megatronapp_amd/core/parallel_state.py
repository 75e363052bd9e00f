"""Process-group topology for DP / TP / PP / CP / EP parallelism.

MI355X-native equivalent of the reference's ``megatron/core/parallel_state.py``
(initialize_model_parallel: parallel_state.py:1272, RankGenerator
:1170, getters :2083-2933 — see SURVEY.md §1 L2).  One process per GPU;
collectives ride RCCL ("nccl" backend on ROCm) over xGMI, "gloo" on CPU.

Rank layout follows the canonical Megatron order ``tp-cp-ep-dp-pp``: the
tensor-parallel dimension varies fastest (neighbouring ranks, i.e. the
tightest xGMI neighbourhood, carry the latency-bound TP collectives), the
pipeline dimension slowest (PP p2p crosses nodes last).
"""

from __future__ import annotations

import itertools
from typing import List, Optional

import torch
import torch.distributed as dist

# ---------------------------------------------------------------------------
# Module state
# ---------------------------------------------------------------------------

_TENSOR_MODEL_PARALLEL_GROUP = None
_PIPELINE_MODEL_PARALLEL_GROUP = None
# Direction-split duplicates of the PP group: activations (toward next)
# ride the FWD communicator, gradients (toward prev) the BWD one.  At
# PP=2 next==prev, so without the split forward and backward traffic
# between one rank pair aliases onto one channel and isend/irecv order
# matching can pair an activation send with a gradient recv (the
# reference's dual-group trick, p2p_communication.py:202-216).
_PIPELINE_FWD_GROUP = None
_PIPELINE_BWD_GROUP = None
_MODEL_PARALLEL_GROUP = None
_DATA_PARALLEL_GROUP = None
_DATA_PARALLEL_GROUP_GLOO = None
_DATA_PARALLEL_GROUP_WITH_CP = None
_DATA_PARALLEL_GROUP_WITH_CP_GLOO = None
_CONTEXT_PARALLEL_GROUP = None
_TENSOR_AND_CONTEXT_PARALLEL_GROUP = None
_EXPERT_MODEL_PARALLEL_GROUP = None
_EXPERT_TENSOR_PARALLEL_GROUP = None
_EXPERT_DATA_PARALLEL_GROUP = None
_EMBEDDING_GROUP = None
_POSITION_EMBEDDING_GROUP = None

_TENSOR_MODEL_PARALLEL_WORLD_SIZE = None
_PIPELINE_MODEL_PARALLEL_WORLD_SIZE = None
_CONTEXT_PARALLEL_WORLD_SIZE = None
_EXPERT_MODEL_PARALLEL_WORLD_SIZE = None

_TENSOR_MODEL_PARALLEL_RANK = None
_PIPELINE_MODEL_PARALLEL_RANK = None

_PIPELINE_GLOBAL_RANKS = None
_CONTEXT_PARALLEL_GLOBAL_RANKS = None
_DATA_PARALLEL_GLOBAL_RANKS = None
_DATA_PARALLEL_GLOBAL_RANKS_WITH_CP = None
_TENSOR_MODEL_PARALLEL_GLOBAL_RANKS = None
_EMBEDDING_GLOBAL_RANKS = None
_POSITION_EMBEDDING_GLOBAL_RANKS = None

_VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK = None
_VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE = None

# MegaFBD forward/backward-disaggregation state (see megatronapp_amd/fbd).
_FORWARD_BACKWARD_PARALLEL_GROUP = None
_IS_FORWARD_STAGE = None
_FBD_DUAL_RANK = None


class RankGenerator:
    """Enumerates rank groups on the (pp, dp, cp, tp) grid.

    ``order`` lists dimensions fastest-varying first, e.g. "tp-cp-dp-pp"
    (the reference's semantics for order 'tp-cp-ep-dp-pp',
    parallel_state.py:1170).  ``get_ranks(token)`` returns, for the
    dimensions named in ``token``, every group of global ranks that share
    the coordinates of all *other* dimensions.
    """

    def __init__(self, tp: int, dp: int, pp: int, cp: int = 1,
                 order: str = "tp-cp-dp-pp") -> None:
        self.sizes = {"tp": tp, "cp": cp, "dp": dp, "pp": pp}
        self.order = order.split("-")
        assert set(self.order) == set(self.sizes), (self.order, self.sizes)
        self.world_size = tp * dp * pp * cp
        # stride of each dim in the global rank number
        self.strides = {}
        s = 1
        for name in self.order:
            self.strides[name] = s
            s *= self.sizes[name]

    def global_rank(self, **coords) -> int:
        return sum(coords[n] * self.strides[n] for n in self.order)

    def get_ranks(self, token: str) -> List[List[int]]:
        """token like "tp", "dp", "tp-pp", "dp-cp"; dims in token are grouped."""
        group_dims = token.split("-")
        other_dims = [n for n in self.order if n not in group_dims]
        # order group dims fastest-first for deterministic in-group ordering
        group_dims_sorted = [n for n in self.order if n in group_dims]
        groups = []
        for other in itertools.product(
                *[range(self.sizes[n]) for n in reversed(other_dims)]):
            other_coords = dict(zip(reversed(other_dims), other))
            ranks = []
            for g in itertools.product(
                    *[range(self.sizes[n]) for n in reversed(group_dims_sorted)]):
                coords = dict(zip(reversed(group_dims_sorted), g))
                coords.update(other_coords)
                ranks.append(self.global_rank(**coords))
            groups.append(ranks)
        return groups


# per-group RCCL tuning from --nccl-communicator-config-path
_PG_COMM_CONFIG = {}


def set_pg_comm_config(cfg: dict):
    global _PG_COMM_CONFIG
    _PG_COMM_CONFIG = cfg or {}


def _pg_options(kind):
    """ProcessGroupNCCL.Options for a named group kind (tp/dp/pp/cp/ep),
    honoring min_ctas/max_ctas/cga_cluster_size from the comm config."""
    cfg = (_PG_COMM_CONFIG or {}).get(kind)
    if not cfg:
        return None
    try:
        opts = dist.ProcessGroupNCCL.Options()
        nccl_cfg = opts.config
        if "min_ctas" in cfg:
            nccl_cfg.min_ctas = int(cfg["min_ctas"])
        if "max_ctas" in cfg:
            nccl_cfg.max_ctas = int(cfg["max_ctas"])
        if "cga_cluster_size" in cfg:
            nccl_cfg.cga_cluster_size = int(cfg["cga_cluster_size"])
        return opts
    except Exception:
        return None


def _new_group(ranks, backend=None, gloo=False, kind=None):
    """Create a group; every rank must call this in the same order."""
    if gloo:
        if dist.is_gloo_available():
            return dist.new_group(ranks, backend="gloo")
        return None
    opts = _pg_options(kind) if kind else None
    if opts is not None:
        return dist.new_group(ranks, backend=backend, pg_options=opts)
    return dist.new_group(ranks, backend=backend)


def initialize_model_parallel(
    tensor_model_parallel_size: int = 1,
    pipeline_model_parallel_size: int = 1,
    virtual_pipeline_model_parallel_size: Optional[int] = None,
    context_parallel_size: int = 1,
    expert_model_parallel_size: int = 1,
    order: str = "tp-cp-dp-pp",
    create_gloo_process_groups: bool = True,
) -> None:
    """Build every process group from the global torch.distributed world.

    Mirrors the reference API (parallel_state.py:1272) with an
    MI355X-native group construction: groups are plain RCCL communicators
    (no SHARP / NCCL-config plumbing — RCCL tunes ring/tree per group
    size over the 7-link xGMI mesh on its own).
    """
    assert dist.is_initialized(), "torch.distributed must be initialized first"
    world_size = dist.get_world_size()
    rank = dist.get_rank()

    tp = tensor_model_parallel_size
    pp = pipeline_model_parallel_size
    cp = context_parallel_size
    ep = expert_model_parallel_size
    total_model = tp * pp * cp
    assert world_size % total_model == 0, (
        f"world size {world_size} not divisible by tp*pp*cp = {total_model}")
    dp = world_size // total_model
    assert dp % ep == 0, f"dp {dp} not divisible by ep {ep}"

    if virtual_pipeline_model_parallel_size is not None:
        assert pp > 1, "interleaved schedule requires pipeline parallelism"
        global _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK
        global _VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE
        _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK = 0
        _VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE = (
            virtual_pipeline_model_parallel_size)

    gen = RankGenerator(tp=tp, dp=dp, pp=pp, cp=cp, order=order)

    global _TENSOR_MODEL_PARALLEL_GROUP, _TENSOR_MODEL_PARALLEL_GLOBAL_RANKS
    global _PIPELINE_MODEL_PARALLEL_GROUP, _PIPELINE_GLOBAL_RANKS
    global _MODEL_PARALLEL_GROUP
    global _DATA_PARALLEL_GROUP, _DATA_PARALLEL_GROUP_GLOO
    global _DATA_PARALLEL_GROUP_WITH_CP, _DATA_PARALLEL_GROUP_WITH_CP_GLOO
    global _DATA_PARALLEL_GLOBAL_RANKS, _DATA_PARALLEL_GLOBAL_RANKS_WITH_CP
    global _CONTEXT_PARALLEL_GROUP, _CONTEXT_PARALLEL_GLOBAL_RANKS
    global _TENSOR_AND_CONTEXT_PARALLEL_GROUP
    global _EXPERT_MODEL_PARALLEL_GROUP, _EXPERT_TENSOR_PARALLEL_GROUP
    global _EXPERT_DATA_PARALLEL_GROUP
    global _EMBEDDING_GROUP, _EMBEDDING_GLOBAL_RANKS
    global _POSITION_EMBEDDING_GROUP, _POSITION_EMBEDDING_GLOBAL_RANKS

    assert _TENSOR_MODEL_PARALLEL_GROUP is None, \
        "model parallel already initialized (call destroy_model_parallel first)"

    for ranks in gen.get_ranks("tp"):
        group = _new_group(ranks, kind="tp")
        if rank in ranks:
            _TENSOR_MODEL_PARALLEL_GROUP = group
            _TENSOR_MODEL_PARALLEL_GLOBAL_RANKS = ranks

    global _PIPELINE_FWD_GROUP, _PIPELINE_BWD_GROUP
    for ranks in gen.get_ranks("pp"):
        group = _new_group(ranks, kind="pp")
        fwd_group = _new_group(ranks, kind="pp")
        bwd_group = _new_group(ranks, kind="pp")
        if rank in ranks:
            _PIPELINE_MODEL_PARALLEL_GROUP = group
            _PIPELINE_FWD_GROUP = fwd_group
            _PIPELINE_BWD_GROUP = bwd_group
            _PIPELINE_GLOBAL_RANKS = ranks
        # Embedding group: first and last stage of each pipeline (tied
        # word embeddings grad all-reduce, finalize_model_grads.py:120).
        if len(ranks) > 1:
            emb_ranks = [ranks[0], ranks[-1]]
            pos_ranks = [ranks[0]]
        else:
            emb_ranks = list(ranks)
            pos_ranks = list(ranks)
        group = _new_group(emb_ranks)
        if rank in emb_ranks:
            _EMBEDDING_GROUP = group
            _EMBEDDING_GLOBAL_RANKS = emb_ranks
        group = _new_group(pos_ranks)
        if rank in pos_ranks:
            _POSITION_EMBEDDING_GROUP = group
            _POSITION_EMBEDDING_GLOBAL_RANKS = pos_ranks

    for ranks in gen.get_ranks("dp"):
        group = _new_group(ranks, kind="dp")
        group_gloo = _new_group(ranks, gloo=True) if create_gloo_process_groups else None
        if rank in ranks:
            _DATA_PARALLEL_GROUP = group
            _DATA_PARALLEL_GROUP_GLOO = group_gloo
            _DATA_PARALLEL_GLOBAL_RANKS = ranks

    for ranks in gen.get_ranks("dp-cp"):
        group = _new_group(ranks, kind="dp")
        group_gloo = _new_group(ranks, gloo=True) if create_gloo_process_groups else None
        if rank in ranks:
            _DATA_PARALLEL_GROUP_WITH_CP = group
            _DATA_PARALLEL_GROUP_WITH_CP_GLOO = group_gloo
            _DATA_PARALLEL_GLOBAL_RANKS_WITH_CP = ranks

    for ranks in gen.get_ranks("cp"):
        group = _new_group(ranks, kind="cp")
        if rank in ranks:
            _CONTEXT_PARALLEL_GROUP = group
            _CONTEXT_PARALLEL_GLOBAL_RANKS = ranks

    for ranks in gen.get_ranks("tp-cp"):
        group = _new_group(ranks)
        if rank in ranks:
            _TENSOR_AND_CONTEXT_PARALLEL_GROUP = group

    for ranks in gen.get_ranks("tp-pp"):
        group = _new_group(ranks)
        if rank in ranks:
            _MODEL_PARALLEL_GROUP = group

    # Expert parallelism: split each dp-cp group into ep-sized slices.
    # Experts shard across ep consecutive dp ranks; the remaining dp/ep
    # replicas form the expert-data-parallel group.
    global _EXPERT_MODEL_PARALLEL_WORLD_SIZE
    _EXPERT_MODEL_PARALLEL_WORLD_SIZE = ep
    for ranks in gen.get_ranks("dp-cp"):
        n = len(ranks)
        for i in range(0, n, ep):
            ep_ranks = ranks[i:i + ep]
            group = _new_group(ep_ranks)
            if rank in ep_ranks:
                _EXPERT_MODEL_PARALLEL_GROUP = group
        for i in range(ep):
            edp_ranks = ranks[i::ep]
            group = _new_group(edp_ranks)
            if rank in edp_ranks:
                _EXPERT_DATA_PARALLEL_GROUP = group
    # Expert-tensor-parallel group: reuse TP group (expert_tp == tp).
    _EXPERT_TENSOR_PARALLEL_GROUP = _TENSOR_MODEL_PARALLEL_GROUP

    global _TENSOR_MODEL_PARALLEL_WORLD_SIZE, _TENSOR_MODEL_PARALLEL_RANK
    global _PIPELINE_MODEL_PARALLEL_WORLD_SIZE, _PIPELINE_MODEL_PARALLEL_RANK
    global _CONTEXT_PARALLEL_WORLD_SIZE
    _TENSOR_MODEL_PARALLEL_WORLD_SIZE = tp
    _PIPELINE_MODEL_PARALLEL_WORLD_SIZE = pp
    _CONTEXT_PARALLEL_WORLD_SIZE = cp
    _TENSOR_MODEL_PARALLEL_RANK = None
    _PIPELINE_MODEL_PARALLEL_RANK = None


def model_parallel_is_initialized() -> bool:
    return _TENSOR_MODEL_PARALLEL_GROUP is not None


def destroy_model_parallel() -> None:
    """Reset all module state (used by tests to re-initialize topologies)."""
    g = globals()
    for name in list(g):
        # every piece of topology state is an _UPPER_CASE module global
        if name.startswith("_") and name[1:].replace("_", "").isupper():
            g[name] = None


# ---------------------------------------------------------------------------
# Getters (reference API surface)
# ---------------------------------------------------------------------------

def get_tensor_model_parallel_group(check_initialized=True):
    if check_initialized:
        assert _TENSOR_MODEL_PARALLEL_GROUP is not None
    return _TENSOR_MODEL_PARALLEL_GROUP


def get_pipeline_model_parallel_group():
    assert _PIPELINE_MODEL_PARALLEL_GROUP is not None
    return _PIPELINE_MODEL_PARALLEL_GROUP


def get_pipeline_forward_group():
    return _PIPELINE_FWD_GROUP or _PIPELINE_MODEL_PARALLEL_GROUP


def get_pipeline_backward_group():
    return _PIPELINE_BWD_GROUP or _PIPELINE_MODEL_PARALLEL_GROUP


def get_model_parallel_group():
    assert _MODEL_PARALLEL_GROUP is not None
    return _MODEL_PARALLEL_GROUP


def get_data_parallel_group(with_context_parallel: bool = False):
    if with_context_parallel:
        assert _DATA_PARALLEL_GROUP_WITH_CP is not None
        return _DATA_PARALLEL_GROUP_WITH_CP
    assert _DATA_PARALLEL_GROUP is not None
    return _DATA_PARALLEL_GROUP


def get_data_parallel_group_gloo(with_context_parallel: bool = False):
    if with_context_parallel:
        return _DATA_PARALLEL_GROUP_WITH_CP_GLOO
    return _DATA_PARALLEL_GROUP_GLOO


def get_context_parallel_group(check_initialized=True):
    if check_initialized:
        assert _CONTEXT_PARALLEL_GROUP is not None
    return _CONTEXT_PARALLEL_GROUP


def get_context_parallel_global_ranks():
    return _CONTEXT_PARALLEL_GLOBAL_RANKS


def get_tensor_and_context_parallel_group():
    return _TENSOR_AND_CONTEXT_PARALLEL_GROUP


def get_expert_model_parallel_group():
    return _EXPERT_MODEL_PARALLEL_GROUP


def get_expert_tensor_parallel_group():
    return _EXPERT_TENSOR_PARALLEL_GROUP


def get_expert_data_parallel_group():
    return _EXPERT_DATA_PARALLEL_GROUP


def get_embedding_group():
    return _EMBEDDING_GROUP


def get_position_embedding_group():
    return _POSITION_EMBEDDING_GROUP


def get_tensor_model_parallel_world_size():
    global _TENSOR_MODEL_PARALLEL_WORLD_SIZE
    if _TENSOR_MODEL_PARALLEL_WORLD_SIZE is None:
        _TENSOR_MODEL_PARALLEL_WORLD_SIZE = dist.get_world_size(
            group=get_tensor_model_parallel_group())
    return _TENSOR_MODEL_PARALLEL_WORLD_SIZE


def get_tensor_model_parallel_rank():
    global _TENSOR_MODEL_PARALLEL_RANK
    if _TENSOR_MODEL_PARALLEL_RANK is None:
        _TENSOR_MODEL_PARALLEL_RANK = dist.get_rank(
            group=get_tensor_model_parallel_group())
    return _TENSOR_MODEL_PARALLEL_RANK


def get_tensor_model_parallel_src_rank():
    assert _TENSOR_MODEL_PARALLEL_GLOBAL_RANKS is not None
    return _TENSOR_MODEL_PARALLEL_GLOBAL_RANKS[0]


def get_tensor_model_parallel_global_ranks():
    return _TENSOR_MODEL_PARALLEL_GLOBAL_RANKS


def get_pipeline_model_parallel_world_size():
    global _PIPELINE_MODEL_PARALLEL_WORLD_SIZE
    if _PIPELINE_MODEL_PARALLEL_WORLD_SIZE is None:
        _PIPELINE_MODEL_PARALLEL_WORLD_SIZE = dist.get_world_size(
            group=get_pipeline_model_parallel_group())
    return _PIPELINE_MODEL_PARALLEL_WORLD_SIZE


def get_pipeline_model_parallel_rank():
    global _PIPELINE_MODEL_PARALLEL_RANK
    if _PIPELINE_MODEL_PARALLEL_RANK is None:
        _PIPELINE_MODEL_PARALLEL_RANK = dist.get_rank(
            group=get_pipeline_model_parallel_group())
    return _PIPELINE_MODEL_PARALLEL_RANK


def get_pipeline_model_parallel_first_rank():
    return _PIPELINE_GLOBAL_RANKS[0]


def get_pipeline_model_parallel_last_rank():
    return _PIPELINE_GLOBAL_RANKS[-1]


def get_pipeline_model_parallel_next_rank():
    ranks = _PIPELINE_GLOBAL_RANKS
    i = get_pipeline_model_parallel_rank()
    return ranks[(i + 1) % len(ranks)]


def get_pipeline_model_parallel_prev_rank():
    ranks = _PIPELINE_GLOBAL_RANKS
    i = get_pipeline_model_parallel_rank()
    return ranks[(i - 1) % len(ranks)]


def is_pipeline_first_stage(ignore_virtual: bool = False):
    if not ignore_virtual and get_virtual_pipeline_model_parallel_world_size():
        if get_virtual_pipeline_model_parallel_rank() != 0:
            return False
    return get_pipeline_model_parallel_rank() == 0


def is_pipeline_last_stage(ignore_virtual: bool = False):
    if not ignore_virtual and get_virtual_pipeline_model_parallel_world_size():
        vpp = get_virtual_pipeline_model_parallel_world_size()
        if get_virtual_pipeline_model_parallel_rank() != vpp - 1:
            return False
    return (get_pipeline_model_parallel_rank() ==
            get_pipeline_model_parallel_world_size() - 1)


def is_rank_in_embedding_group(ignore_virtual: bool = False):
    if _EMBEDDING_GLOBAL_RANKS is None:
        return False
    rank = dist.get_rank()
    if ignore_virtual:
        return rank in _EMBEDDING_GLOBAL_RANKS
    if rank in _EMBEDDING_GLOBAL_RANKS:
        if rank == _EMBEDDING_GLOBAL_RANKS[0]:
            return is_pipeline_first_stage(ignore_virtual=False)
        if rank == _EMBEDDING_GLOBAL_RANKS[-1]:
            return is_pipeline_last_stage(ignore_virtual=False)
        return True
    return False


def get_virtual_pipeline_model_parallel_rank():
    return _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK


def set_virtual_pipeline_model_parallel_rank(rank):
    global _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK
    _VIRTUAL_PIPELINE_MODEL_PARALLEL_RANK = rank


def get_virtual_pipeline_model_parallel_world_size():
    return _VIRTUAL_PIPELINE_MODEL_PARALLEL_WORLD_SIZE


def get_context_parallel_world_size():
    if _CONTEXT_PARALLEL_GROUP is None:
        return 1
    return dist.get_world_size(group=_CONTEXT_PARALLEL_GROUP)


def get_context_parallel_rank():
    if _CONTEXT_PARALLEL_GROUP is None:
        return 0
    return dist.get_rank(group=_CONTEXT_PARALLEL_GROUP)


def get_data_parallel_world_size(with_context_parallel: bool = False):
    return dist.get_world_size(
        group=get_data_parallel_group(with_context_parallel))


def get_data_parallel_rank(with_context_parallel: bool = False):
    return dist.get_rank(group=get_data_parallel_group(with_context_parallel))


def get_data_parallel_src_rank(with_context_parallel: bool = False):
    ranks = (_DATA_PARALLEL_GLOBAL_RANKS_WITH_CP if with_context_parallel
             else _DATA_PARALLEL_GLOBAL_RANKS)
    return ranks[0]


def get_expert_model_parallel_world_size():
    if _EXPERT_MODEL_PARALLEL_GROUP is None:
        return 1
    return dist.get_world_size(group=_EXPERT_MODEL_PARALLEL_GROUP)


def get_expert_model_parallel_rank():
    if _EXPERT_MODEL_PARALLEL_GROUP is None:
        return 0
    return dist.get_rank(group=_EXPERT_MODEL_PARALLEL_GROUP)


def get_expert_data_parallel_world_size():
    if _EXPERT_DATA_PARALLEL_GROUP is None:
        return 1
    return dist.get_world_size(group=_EXPERT_DATA_PARALLEL_GROUP)


def get_expert_data_parallel_rank():
    if _EXPERT_DATA_PARALLEL_GROUP is None:
        return 0
    return dist.get_rank(group=_EXPERT_DATA_PARALLEL_GROUP)


# --- MegaFBD hooks (populated by megatronapp_amd.fbd when enabled) -----------

def get_forward_backward_parallel_group():
    return _FORWARD_BACKWARD_PARALLEL_GROUP


def is_forward_stage():
    """True on forward-only ranks under forward/backward disaggregation."""
    if _IS_FORWARD_STAGE is None:
        return False
    return _IS_FORWARD_STAGE


def get_forward_backward_parallel_dual_rank():
    return _FBD_DUAL_RANK
