"""Context-parallel attention (native — the reference delegates CP comm to
TransformerEngine, extensions/transformer_engine.py:667-692; SURVEY.md
§5.7 requires a native rebuild).

Sequence chunking follows core.utils.get_batch_on_this_cp_rank: CP rank i
holds global chunks (i, 2*cp-1-i) of 2*cp chunks — the causal
load-balancing split.  Two comm modes, selected by config.cp_comm_type:

* "allgather": K/V all-gathered over the CP group (autograd: backward is
  the reduce-scatter of dK/dV), reordered to global sequence order, local
  Q attends with an explicit causal mask built from global positions.
* "a2a" (Ulysses): q/k/v all-to-all'd head<->sequence over the CP group so
  each rank holds the FULL sequence for nh/cp heads; normal causal
  attention after chunk reorder; inverse a2a on the context output.

* "ring": flash-style ring attention — K/V rotate around the CP ring
  (p2p over xGMI) while each rank folds the visiting block into its
  running (O, logsumexp) online; backward re-rotates K/V, recomputes the
  block softmax from the saved lse and circulates dK/dV accumulators one
  full lap so every contribution lands back on its owner.  O(s_local)
  activation memory — the long-context mode.
"""

from __future__ import annotations

import torch

from .. import parallel_state
from ..enums import AttnMaskType
from ..tensor_parallel.mappings import all_to_all
from ..transformer_config import TransformerConfig


class _AllGatherAlongFirstDimCP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        world = torch.distributed.get_world_size(group)
        ctx.world = world
        if world == 1:
            return x
        out = torch.empty((x.shape[0] * world,) + tuple(x.shape[1:]),
                          dtype=x.dtype, device=x.device)
        torch.distributed.all_gather_into_tensor(out, x.contiguous(),
                                                 group=group)
        return out

    @staticmethod
    def backward(ctx, grad):
        if ctx.world == 1:
            return grad, None
        shape = (grad.shape[0] // ctx.world,) + tuple(grad.shape[1:])
        out = torch.empty(shape, dtype=grad.dtype, device=grad.device)
        torch.distributed.reduce_scatter_tensor(out, grad.contiguous(),
                                                group=ctx.group)
        return out, None


def _chunk_order(cp: int):
    """Global chunk ids in gathered (rank-major) order, and the permutation
    taking gathered order -> global order."""
    gathered = []
    for r in range(cp):
        gathered += [r, 2 * cp - 1 - r]
    perm = sorted(range(2 * cp), key=lambda i: gathered[i])
    return gathered, perm


def _reorder_to_global(x: torch.Tensor, cp: int):
    """x: [2*cp*L, ...] in gathered order -> global sequence order."""
    _, perm = _chunk_order(cp)
    chunks = x.chunk(2 * cp, dim=0)
    return torch.cat([chunks[i] for i in perm], dim=0)


def _reorder_from_global(x: torch.Tensor, cp: int):
    gathered, _ = _chunk_order(cp)
    chunks = x.chunk(2 * cp, dim=0)
    return torch.cat([chunks[g] for g in gathered], dim=0)


def _local_global_positions(cp: int, cp_rank: int, s_local: int,
                            device) -> torch.Tensor:
    """Global position of each local row (two chunks: rank, 2cp-1-rank)."""
    half = s_local // 2
    c0, c1 = cp_rank, 2 * cp - 1 - cp_rank
    return torch.cat([
        torch.arange(c0 * half, (c0 + 1) * half, device=device),
        torch.arange(c1 * half, (c1 + 1) * half, device=device)])


class ContextParallelAttention(torch.nn.Module):
    """Wraps a core-attention module with CP communication."""

    def __init__(self, core_attention, config: TransformerConfig):
        super().__init__()
        self.core = core_attention
        self.config = config

    def forward(self, query, key, value, attention_mask=None,
                attn_mask_type=None, attention_bias=None,
                packed_seq_params=None):
        cp = parallel_state.get_context_parallel_world_size()
        if cp == 1:
            return self.core(query, key, value, attention_mask,
                             attn_mask_type, attention_bias,
                             packed_seq_params)
        group = parallel_state.get_context_parallel_group()
        cp_rank = parallel_state.get_context_parallel_rank()
        mode = self.config.cp_comm_type

        if mode == "a2a":
            return self._ulysses(query, key, value, group, cp, cp_rank)
        if mode == "ring":
            return self._ring(query, key, value, group, cp, cp_rank)
        return self._allgather(query, key, value, group, cp, cp_rank)

    # ----------------------------------------------------------- allgather
    def _allgather(self, q, k, v, group, cp, cp_rank):
        sq_local = q.shape[0]
        k_full = _reorder_to_global(
            _AllGatherAlongFirstDimCP.apply(k, group), cp)
        v_full = _reorder_to_global(
            _AllGatherAlongFirstDimCP.apply(v, group), cp)
        sk = k_full.shape[0]
        q_pos = _local_global_positions(cp, cp_rank, sq_local, q.device)
        kv_pos = torch.arange(sk, device=q.device)
        # mask True = masked out (row attends to kv_pos <= q_pos)
        mask = kv_pos.unsqueeze(0) > q_pos.unsqueeze(1)   # [sq_local, sk]
        mask = mask.view(1, 1, sq_local, sk)
        return self.core(q, k_full, v_full, attention_mask=mask,
                         attn_mask_type=AttnMaskType.padding)

    # ------------------------------------------------------------------ ring
    def _ring(self, q, k, v, group, cp, cp_rank):
        import math
        sl, b, nh, hd = q.shape
        assert k.shape[2] == nh, "ring CP requires num_query_groups == heads"
        scale = 1.0 / math.sqrt(hd)
        out = _RingAttention.apply(q.contiguous(), k.contiguous(),
                                   v.contiguous(), group, cp, cp_rank, scale)
        return out.reshape(sl, b, nh * hd)

    # -------------------------------------------------------------- ulysses
    def _ulysses(self, q, k, v, group, cp, cp_rank):
        # [s_local, b, nh, hd] -> full seq, nh/cp heads
        sq_local, b, nh, hd = q.shape
        assert nh % cp == 0, "Ulysses CP needs heads divisible by cp"

        def scatter_heads(x):
            s_l, b_, nh_, hd_ = x.shape
            # split heads into cp groups; a2a exchanges (head-group, seq)
            xs = x.reshape(s_l, b_, cp, nh_ // cp, hd_).permute(
                2, 0, 1, 3, 4).reshape(cp * s_l, b_, nh_ // cp, hd_)
            out = all_to_all(group, xs.contiguous())
            return _reorder_to_global(out, cp)

        def gather_heads(x_full):
            # [s_full, b, nh/cp, hd] -> [s_local, b, nh, hd]
            x_full = _reorder_from_global(x_full, cp)
            out = all_to_all(group, x_full.contiguous())
            s_tot = out.shape[0]
            s_l = s_tot // cp
            return out.reshape(cp, s_l, b, nh // cp, hd).permute(
                1, 2, 0, 3, 4).reshape(s_l, b, nh, hd)

        q_full = scatter_heads(q)
        ng = k.shape[2]
        assert ng % cp == 0, "Ulysses CP needs kv groups divisible by cp"
        k_full = scatter_heads(k)
        v_full = scatter_heads(v)
        ctx = self.core(q_full, k_full, v_full, attention_mask=None,
                        attn_mask_type=AttnMaskType.causal)
        # ctx: [s_full, b, (nh/cp)*hd] -> back to local seq, all heads
        s_full = ctx.shape[0]
        ctx = ctx.reshape(s_full, b, nh // cp, hd)
        out = gather_heads(ctx)
        return out.reshape(sq_local, b, nh * hd)


# ------------------------------------------------------------------- ring
def _ring_exchange(t: torch.Tensor, group, cp: int, rank: int):
    """One hop: send t to (rank+1) % cp, return the tensor received from
    (rank-1) % cp.  isend/irecv so odd ring sizes cannot deadlock."""
    ranks = torch.distributed.get_process_group_ranks(group)
    nxt = ranks[(rank + 1) % cp]
    prv = ranks[(rank - 1) % cp]
    recv = torch.empty_like(t)
    reqs = [torch.distributed.P2POp(torch.distributed.isend, t.contiguous(),
                                    nxt, group=group),
            torch.distributed.P2POp(torch.distributed.irecv, recv, prv,
                                    group=group)]
    for r in torch.distributed.batch_isend_irecv(reqs):
        r.wait()
    return recv


def _block_mask(cp, q_rank, kv_rank, sl, device):
    """True = masked; causal between this rank's global q positions and
    the visiting rank's global kv positions."""
    q_pos = _local_global_positions(cp, q_rank, sl, device)
    kv_pos = _local_global_positions(cp, kv_rank, sl, device)
    return kv_pos.unsqueeze(0) > q_pos.unsqueeze(1)      # [sl, sl]


class _RingAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, cp, rank, scale):
        sl, b, nh, hd = q.shape
        qf = q.float().permute(1, 2, 0, 3)                # [b, nh, sl, hd]
        o = torch.zeros_like(qf)
        lse = torch.full((b, nh, sl), float("-inf"), device=q.device)
        kv = torch.stack([k.float(), v.float()])          # travels the ring
        for step in range(cp):
            src = (rank - step) % cp
            kf = kv[0].permute(1, 2, 0, 3)                # [b, nh, sl, hd]
            vf = kv[1].permute(1, 2, 0, 3)
            scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
            scores = scores.masked_fill(
                _block_mask(cp, rank, src, sl, q.device), float("-inf"))
            lse_p = torch.logsumexp(scores, dim=-1)       # [b, nh, sl]
            p = torch.exp(scores - lse_p.unsqueeze(-1)).nan_to_num(0.0)
            o_p = torch.matmul(p, vf)
            new_lse = torch.logaddexp(lse, lse_p)
            a = torch.exp(lse - new_lse).nan_to_num(0.0).unsqueeze(-1)
            c = torch.exp(lse_p - new_lse).nan_to_num(0.0).unsqueeze(-1)
            o = o * a + o_p * c
            lse = new_lse
            if step + 1 < cp:
                kv = _ring_exchange(kv, group, cp, rank)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.meta = (group, cp, rank, scale)
        return o.permute(2, 0, 1, 3).to(q.dtype)          # [sl, b, nh, hd]

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o, lse = ctx.saved_tensors
        group, cp, rank, scale = ctx.meta
        sl, b, nh, hd = q.shape
        qf = q.float().permute(1, 2, 0, 3)
        dof = dout.float().permute(1, 2, 0, 3)            # [b, nh, sl, hd]
        drow = (dof * o).sum(-1, keepdim=True)            # [b, nh, sl, 1]
        dq = torch.zeros_like(qf)
        # kv and its grad accumulators travel one full lap together, so
        # each rank's contribution rides home to the owner
        kvd = torch.stack([k.float(), v.float(),
                           torch.zeros(sl, b, nh, hd, device=q.device),
                           torch.zeros(sl, b, nh, hd, device=q.device)])
        for step in range(cp):
            src = (rank - step) % cp
            kf = kvd[0].permute(1, 2, 0, 3)
            vf = kvd[1].permute(1, 2, 0, 3)
            scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
            scores = scores.masked_fill(
                _block_mask(cp, rank, src, sl, q.device), float("-inf"))
            p = torch.exp(scores - lse.unsqueeze(-1)).nan_to_num(0.0)
            dp = torch.matmul(dof, vf.transpose(-1, -2))
            ds = p * (dp - drow) * scale                  # [b, nh, sl, sl]
            dq += torch.matmul(ds, kf)
            dk_p = torch.matmul(ds.transpose(-1, -2), qf) # [b, nh, sl_k, hd]
            dv_p = torch.matmul(p.transpose(-1, -2), dof)
            kvd[2] += dk_p.permute(2, 0, 1, 3)
            kvd[3] += dv_p.permute(2, 0, 1, 3)
            kvd = _ring_exchange(kvd, group, cp, rank)
        # after cp hops the stack is home: kvd[2]/kvd[3] hold the full sums
        dq = dq.permute(2, 0, 1, 3).to(q.dtype)
        return (dq, kvd[2].to(k.dtype), kvd[3].to(v.dtype),
                None, None, None, None)
