"""Vocab-parallel cross entropy (reference cross_entropy.py:17-232).

Logits stay sharded [s, b, v/tp]; two small TP all-reduces (per-token max,
per-token sum-exp + target-logit) replace gathering the full vocab —
at vocab 51200 and TP 4 this saves 3/4 of the logit traffic.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from .. import parallel_state
from .utils import VocabUtility
from ... import ops as _ops


class _VocabParallelCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, vocab_parallel_logits, target, label_smoothing=0.0):
        tp_group = parallel_state.get_tensor_model_parallel_group()
        tp_world = parallel_state.get_tensor_model_parallel_world_size()
        tp_rank = parallel_state.get_tensor_model_parallel_rank()

        if (vocab_parallel_logits.is_cuda
                and vocab_parallel_logits.dtype == torch.bfloat16
                and label_smoothing == 0.0 and _ops.have_ops()
                and hasattr(_ops.get_ops(), "ce_fwd")
                and vocab_parallel_logits.size(-1) % 8 == 0):
            # fused path: the bf16 logits are the only [N, V] tensor ever
            # read or written — no fp32 softmax materialization
            ops = _ops.get_ops()
            Vp = vocab_parallel_logits.size(-1)
            logits_2d = vocab_parallel_logits.reshape(-1, Vp).contiguous()
            vocab_start, vocab_end = \
                VocabUtility.vocab_range_from_per_partition_vocab_size(
                    Vp, tp_rank, tp_world)
            tgt = target.reshape(-1).to(torch.int32) - vocab_start
            tgt = torch.where((target.reshape(-1) >= vocab_start)
                              & (target.reshape(-1) < vocab_end), tgt,
                              torch.full_like(tgt, -1))
            tgt = tgt.contiguous()
            rowmax = ops.ce_rowmax(logits_2d)
            if tp_world > 1:
                dist.all_reduce(rowmax, op=dist.ReduceOp.MAX, group=tp_group)
            sumexp, predicted = ops.ce_fwd(logits_2d, rowmax, tgt)
            if tp_world > 1:
                dist.all_reduce(sumexp, group=tp_group)
                dist.all_reduce(predicted, group=tp_group)
            loss = (torch.log(sumexp) - predicted).view_as(
                target).to(torch.float32)
            ctx.fused = True
            ctx.label_smoothing = 0.0
            ctx.logits_shape = vocab_parallel_logits.shape
            ctx.save_for_backward(logits_2d, rowmax, sumexp, tgt)
            return loss
        ctx.fused = False

        logits_max = torch.max(vocab_parallel_logits, dim=-1)[0]
        if tp_world > 1:
            dist.all_reduce(logits_max, op=dist.ReduceOp.MAX, group=tp_group)
        vocab_parallel_logits = vocab_parallel_logits - logits_max.unsqueeze(-1)

        partition_vocab_size = vocab_parallel_logits.size(-1)
        vocab_start, vocab_end = VocabUtility.vocab_range_from_per_partition_vocab_size(
            partition_vocab_size, tp_rank, tp_world)

        target_mask = (target < vocab_start) | (target >= vocab_end)
        masked_target = target.clone() - vocab_start
        masked_target[target_mask] = 0

        logits_2d = vocab_parallel_logits.view(-1, partition_vocab_size)
        masked_target_1d = masked_target.view(-1)
        arange_1d = torch.arange(logits_2d.size(0), device=logits_2d.device)
        predicted_logits_1d = logits_2d[arange_1d, masked_target_1d].clone()
        predicted_logits = predicted_logits_1d.view_as(target)
        predicted_logits[target_mask] = 0.0
        if tp_world > 1:
            dist.all_reduce(predicted_logits, group=tp_group)

        exp_logits = torch.exp(vocab_parallel_logits.float())
        sum_exp_logits = exp_logits.sum(dim=-1)
        if tp_world > 1:
            dist.all_reduce(sum_exp_logits, group=tp_group)

        loss = torch.log(sum_exp_logits) - predicted_logits

        exp_logits = exp_logits.div_(sum_exp_logits.unsqueeze(-1))
        ctx.label_smoothing = label_smoothing
        ctx.vocab_size = partition_vocab_size * tp_world
        if label_smoothing > 0:
            # smoothed loss: (1-eps)*nll + eps/K * sum(-log p)
            eps = label_smoothing
            K = ctx.vocab_size
            log_probs = torch.log(exp_logits + 1e-20)
            mean_log_probs = log_probs.mean(dim=-1)
            if tp_world > 1:
                dist.all_reduce(mean_log_probs, group=tp_group)
                mean_log_probs = mean_log_probs / tp_world
            loss = (1.0 - eps) * loss - eps * mean_log_probs
        ctx.input_dtype = vocab_parallel_logits.dtype
        ctx.save_for_backward(exp_logits, target_mask, masked_target_1d)
        return loss

    @staticmethod
    def backward(ctx, grad_output):
        if ctx.fused:
            logits_2d, rowmax, sumexp, tgt = ctx.saved_tensors
            dlogits = _ops.get_ops().ce_bwd(
                logits_2d, rowmax, sumexp, tgt,
                grad_output.reshape(-1).float())
            return dlogits.view(ctx.logits_shape), None, None
        softmax, target_mask, masked_target_1d = ctx.saved_tensors
        grad_input = softmax
        partition_vocab_size = softmax.size(-1)
        grad_2d = grad_input.view(-1, partition_vocab_size)
        arange_1d = torch.arange(grad_2d.size(0), device=grad_2d.device)
        softmax_update = 1.0 - target_mask.view(-1).float()
        if ctx.label_smoothing > 0:
            eps = ctx.label_smoothing
            K = ctx.vocab_size
            grad_2d[arange_1d, masked_target_1d] -= (1.0 - eps) * softmax_update
            grad_2d -= eps / K
        else:
            grad_2d[arange_1d, masked_target_1d] -= softmax_update
        grad_input = grad_input * grad_output.unsqueeze(-1)
        return grad_input.to(ctx.input_dtype), None, None


def vocab_parallel_cross_entropy(vocab_parallel_logits, target,
                                 label_smoothing=0.0):
    return _VocabParallelCrossEntropy.apply(
        vocab_parallel_logits, target, label_smoothing)
