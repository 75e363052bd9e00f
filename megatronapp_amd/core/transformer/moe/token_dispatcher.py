"""MoE token dispatchers (reference moe/token_dispatcher.py).

* MoEAlltoAllTokenDispatcher (:248): tokens permuted by expert, exchanged
  with one RCCL all-to-all over the EP group (variable splits), processed
  by local experts, returned by the inverse all-to-all.  On MI355X the
  a2a rides point-to-point xGMI links — each (src,dst) pair moves its
  slice concurrently, which is exactly the fabric's strength.
* MoEAllGatherTokenDispatcher (:114): all-gather every rank's tokens and
  mask locally — cheaper at tiny EP/token counts.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ... import parallel_state
from ...tensor_parallel.mappings import all_to_all
from ...transformer_config import TransformerConfig


def permute(tokens: torch.Tensor, indices: torch.Tensor):
    """Sort tokens (replicated topk times) by expert id.
    tokens [n, h], indices [n, topk] -> (permuted [n*topk, h], sort_idx)."""
    topk = indices.shape[1]
    flat = indices.reshape(-1)
    sort_idx = torch.argsort(flat, stable=True)
    tok_idx = sort_idx // topk
    return tokens.index_select(0, tok_idx), sort_idx


class _FusedCombineFn(torch.autograd.Function):
    """unpermute + topk-weighted sum as ONE kernel each way
    (ops/csrc/moe.hip) — the eager version is 3 full passes over
    [n*topk, h]."""

    @staticmethod
    def forward(ctx, permuted, sort_idx, probs, n_tokens):
        from .... import ops as _ops
        inv_pos = torch.empty_like(sort_idx)
        inv_pos[sort_idx] = torch.arange(sort_idx.numel(),
                                         device=sort_idx.device)
        probs_f = probs.reshape(-1).float().contiguous()
        ctx.save_for_backward(permuted, sort_idx, probs_f)
        ctx.topk = probs.shape[1]
        ctx.probs_dtype = probs.dtype
        return _ops.get_ops().moe_combine_fwd(
            permuted.contiguous(), inv_pos.contiguous(), probs_f, n_tokens)

    @staticmethod
    def backward(ctx, dout):
        from .... import ops as _ops
        permuted, sort_idx, probs_f = ctx.saved_tensors
        dpermuted, dprobs = _ops.get_ops().moe_combine_bwd(
            dout, permuted.contiguous(), sort_idx.contiguous(), probs_f,
            ctx.topk)
        dprobs = dprobs.reshape(-1, ctx.topk).to(ctx.probs_dtype)
        return dpermuted, None, dprobs, None


def unpermute(permuted: torch.Tensor, sort_idx: torch.Tensor,
              probs: torch.Tensor, n_tokens: int):
    """Inverse of permute + weighted combine over topk copies."""
    from .... import ops as _ops
    if (permuted.is_cuda and permuted.dtype == torch.bfloat16
            and permuted.shape[-1] % 8 == 0 and _ops.have_ops()
            and hasattr(_ops.get_ops(), "moe_combine_fwd")):
        return _FusedCombineFn.apply(permuted, sort_idx, probs, n_tokens)
    topk = probs.shape[1]
    h = permuted.shape[-1]
    # out-of-place index_copy keeps the autograd graph to `permuted`
    unsorted = torch.zeros_like(permuted).index_copy(0, sort_idx, permuted)
    unsorted = unsorted.reshape(n_tokens, topk, h)
    return (unsorted * probs.unsqueeze(-1).to(unsorted.dtype)).sum(dim=1)


class MoEAlltoAllTokenDispatcher:
    def __init__(self, num_local_experts: int, local_expert_indices,
                 config: TransformerConfig):
        self.config = config
        self.num_local_experts = num_local_experts
        self.num_experts = config.num_moe_experts
        self.ep_group = parallel_state.get_expert_model_parallel_group()
        self.ep_size = parallel_state.get_expert_model_parallel_world_size()

    def dispatch(self, tokens: torch.Tensor, probs: torch.Tensor,
                 indices: torch.Tensor):
        """Returns (expert_inputs [m, h], tokens_per_local_expert,
        state for combine)."""
        n = tokens.shape[0]
        permuted, sort_idx = permute(tokens, indices)
        counts = torch.bincount(indices.reshape(-1),
                                minlength=self.num_experts)
        if self.ep_size == 1:
            state = (sort_idx, probs, n, None, None, None)
            tpe = counts.reshape(self.num_local_experts)
            return permuted, tpe, state

        # splits by destination rank (experts are contiguous per rank)
        per_rank = counts.reshape(self.ep_size, self.num_local_experts).sum(-1)
        input_splits = per_rank.tolist()
        # exchange counts so we know recv splits
        recv_counts = torch.empty_like(counts)
        torch.distributed.all_to_all_single(
            recv_counts.reshape(self.ep_size, self.num_local_experts),
            counts.reshape(self.ep_size, self.num_local_experts),
            group=self.ep_group)
        output_splits = recv_counts.reshape(
            self.ep_size, self.num_local_experts).sum(-1).tolist()

        exchanged = all_to_all(self.ep_group, permuted, output_splits,
                               input_splits)
        # tokens arrive grouped by source rank, each group sorted by local
        # expert; re-sort to local-expert-major order
        src_expert = torch.repeat_interleave(
            torch.arange(self.ep_size * self.num_local_experts,
                         device=tokens.device) % self.num_local_experts,
            recv_counts.reshape(-1))
        resort = torch.argsort(src_expert, stable=True)
        exchanged = exchanged.index_select(0, resort)
        tokens_per_local_expert = recv_counts.reshape(
            self.ep_size, self.num_local_experts).sum(0)
        state = (sort_idx, probs, n, input_splits, output_splits, resort)
        return exchanged, tokens_per_local_expert, state

    def combine(self, expert_output: torch.Tensor, state):
        sort_idx, probs, n, input_splits, output_splits, resort = state
        if self.ep_size > 1:
            unsorted = torch.zeros_like(expert_output).index_copy(
                0, resort, expert_output)
            expert_output = all_to_all(self.ep_group, unsorted, input_splits,
                                       output_splits)
        return unpermute(expert_output, sort_idx, probs, n)


class MoEAllGatherTokenDispatcher:
    """All-gather variant (reference :114): every rank gathers the EP
    group's tokens, runs its LOCAL experts over the tokens routed to
    them, scatters the partial outputs back into the permuted buffer and
    all-reduces over EP before un-permuting its own token range."""

    def __init__(self, num_local_experts: int, local_expert_indices,
                 config: TransformerConfig):
        self.config = config
        self.num_local_experts = num_local_experts
        self.local_expert_indices = list(local_expert_indices)
        self.ep_group = parallel_state.get_expert_model_parallel_group()
        self.ep_size = parallel_state.get_expert_model_parallel_world_size()

    def dispatch(self, tokens, probs, indices):
        n = tokens.shape[0]
        if self.ep_size > 1:
            gathered = [torch.empty_like(tokens) for _ in range(self.ep_size)]
            torch.distributed.all_gather(gathered, tokens.contiguous(),
                                         group=self.ep_group)
            all_tokens = torch.cat(gathered, dim=0)
            gi = [torch.empty_like(indices) for _ in range(self.ep_size)]
            torch.distributed.all_gather(gi, indices.contiguous(),
                                         group=self.ep_group)
            all_indices = torch.cat(gi, dim=0)
            gp = [torch.empty_like(probs) for _ in range(self.ep_size)]
            torch.distributed.all_gather(gp, probs.contiguous(),
                                         group=self.ep_group)
            all_probs = torch.cat(gp, dim=0)
        else:
            all_tokens, all_indices, all_probs = tokens, indices, probs
        permuted, sort_idx = permute(all_tokens, all_indices)
        counts = torch.bincount(all_indices.reshape(-1),
                                minlength=self.config.num_moe_experts)
        lo = int(counts[:self.local_expert_indices[0]].sum())
        local_counts = counts[self.local_expert_indices[0]:
                              self.local_expert_indices[-1] + 1]
        hi = lo + int(local_counts.sum())
        state = (sort_idx, all_probs, all_indices.shape[0], n,
                 permuted.shape, lo, hi, permuted.dtype, permuted.device)
        return permuted[lo:hi], local_counts, state

    def combine(self, expert_output, state):
        (sort_idx, all_probs, n_all, n, full_shape, lo, hi, dtype,
         device) = state
        pre = torch.zeros((lo, full_shape[1]), dtype=expert_output.dtype,
                          device=device)
        post = torch.zeros((full_shape[0] - hi, full_shape[1]),
                           dtype=expert_output.dtype, device=device)
        full = torch.cat([pre, expert_output, post], dim=0)
        combined = unpermute(full, sort_idx, all_probs, n_all)
        if self.ep_size > 1:
            torch.distributed.all_reduce(combined, group=self.ep_group)
            rank = parallel_state.get_expert_model_parallel_rank()
            combined = combined[rank * n:(rank + 1) * n]
        return combined
