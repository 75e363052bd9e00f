"""MegaFBD process-group topology.

Layout (reference parallel_state.py:444-520): the world interleaves
forward and backward blocks of TP size —

  [ fwd(tp) | bwd(tp) | fwd(tp) | bwd(tp) | ... ]

so global rank r is a FORWARD instance iff (r // tp) is even.  Each
forward rank r pairs with its dual backward rank r + tp.  Logical
(Megatron) topology is built over the per-kind half-worlds: TP groups
within a block, PP groups across same-kind blocks, half-DP groups among
same-kind replicas (reference get_half_data_parallel_group :2191).
"""

from __future__ import annotations

import torch.distributed as dist

from ..core import parallel_state


def initialize_model_parallel_fbd(args):
    """Build FBD groups and install them into core.parallel_state."""
    world = dist.get_world_size()
    rank = dist.get_rank()
    tp = args.tensor_model_parallel_size
    pp = args.pipeline_model_parallel_size
    assert world % (2 * tp) == 0, (
        f"FBD needs world divisible by 2*tp, got {world} / {2 * tp}")
    logical_world = world // 2
    assert logical_world % (tp * pp) == 0
    dp = logical_world // (tp * pp)

    is_forward = (rank // tp) % 2 == 0
    dual = rank + tp if is_forward else rank - tp

    # map: logical rank l -> (fwd global, bwd global)
    def fwd_global(l):
        block = l // tp
        return block * 2 * tp + (l % tp)

    def bwd_global(l):
        return fwd_global(l) + tp

    # my logical rank
    block = rank // (2 * tp)
    logical = block * tp + (rank % tp)

    # logical grid (order tp-dp-pp like core RankGenerator)
    gen = parallel_state.RankGenerator(tp=tp, dp=dp, pp=pp, cp=1)

    ps = parallel_state

    def build(groups_logical, kind):
        """Create torch groups for each logical group, per kind."""
        mine = None
        for ranks_l in groups_logical:
            for mapper in (fwd_global, bwd_global):
                ranks_g = [mapper(l) for l in ranks_l]
                g = dist.new_group(ranks_g)
                if rank in ranks_g:
                    mine = (g, ranks_g)
        return mine

    # TP groups
    g, ranks_g = build(gen.get_ranks("tp"), "tp")
    ps._TENSOR_MODEL_PARALLEL_GROUP = g
    ps._TENSOR_MODEL_PARALLEL_GLOBAL_RANKS = ranks_g
    ps._TENSOR_MODEL_PARALLEL_WORLD_SIZE = tp

    # PP groups (+ direction-split duplicates + embedding groups)
    mine_pp = None
    for ranks_l in gen.get_ranks("pp"):
        for mapper in (fwd_global, bwd_global):
            ranks_g = [mapper(l) for l in ranks_l]
            g = dist.new_group(ranks_g)
            gf = dist.new_group(ranks_g)
            gb = dist.new_group(ranks_g)
            if rank in ranks_g:
                mine_pp = (g, gf, gb, ranks_g)
            emb = [ranks_g[0], ranks_g[-1]] if len(ranks_g) > 1 else list(ranks_g)
            ge = dist.new_group(emb)
            if rank in emb:
                ps._EMBEDDING_GROUP = ge
                ps._EMBEDDING_GLOBAL_RANKS = emb
    ps._PIPELINE_MODEL_PARALLEL_GROUP = mine_pp[0]
    ps._PIPELINE_FWD_GROUP = mine_pp[1]
    ps._PIPELINE_BWD_GROUP = mine_pp[2]
    ps._PIPELINE_GLOBAL_RANKS = mine_pp[3]
    ps._PIPELINE_MODEL_PARALLEL_WORLD_SIZE = pp

    # half-DP groups (per kind) — the FBD "half data parallel" groups
    g, ranks_g = build(gen.get_ranks("dp"), "dp")
    ps._DATA_PARALLEL_GROUP = g
    ps._DATA_PARALLEL_GLOBAL_RANKS = ranks_g
    ps._DATA_PARALLEL_GROUP_WITH_CP = g
    ps._DATA_PARALLEL_GLOBAL_RANKS_WITH_CP = ranks_g
    if dist.is_gloo_available():
        g2, _ = build(gen.get_ranks("dp"), "dp_gloo")
        ps._DATA_PARALLEL_GROUP_GLOO = g2

    # model-parallel group (tp x pp, per kind)
    g, _ = build(gen.get_ranks("tp-pp"), "mp")
    ps._MODEL_PARALLEL_GROUP = g

    # context/expert groups: trivial under FBD round-1
    g, ranks_g = build([[l] for l in range(logical_world)], "self")
    ps._CONTEXT_PARALLEL_GROUP = g
    ps._CONTEXT_PARALLEL_GLOBAL_RANKS = ranks_g
    ps._CONTEXT_PARALLEL_WORLD_SIZE = 1

    # dual pair groups (fwd i, bwd i)
    for l in range(logical_world):
        pair = [fwd_global(l), bwd_global(l)]
        g = dist.new_group(pair)
        if rank in pair:
            ps._FORWARD_BACKWARD_PARALLEL_GROUP = g
    ps._IS_FORWARD_STAGE = is_forward
    ps._FBD_DUAL_RANK = dual
    return is_forward, dual, logical
