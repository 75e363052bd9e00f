"""Theoretical per-GPU memory footprint for a training configuration.

Reference: training/theoretical_memory_usage.py:1-287.  Same model: weights
and optimizer state from the parameter count of the most-loaded pipeline
shard, activations from the selective-recompute formula of Korthikanti et
al. (arXiv:2205.05198, Table 2).  Sized against MI355X's 288 GB HBM3E —
`report_theoretical_memory` also prints the fraction of a single MI355X the
config needs, which is the number that decides TP/PP degree on this part.
"""

from __future__ import annotations

import math

MiB = 1024 * 1024
MI355X_HBM_BYTES = 288 * 1024 ** 3


def compute_weight_and_optimizer_memory(args, verbose=False):
    """Bytes of weights + gradients + optimizer state on the busiest GPU."""
    h = args.hidden_size
    kv_channels = args.kv_channels or h // args.num_attention_heads
    q_proj = kv_channels * args.num_attention_heads
    groups = (args.num_query_groups
              if getattr(args, "group_query_attention", False)
              and args.num_query_groups else args.num_attention_heads)
    gated = 1.5 if getattr(args, "swiglu", False) else 1.0
    ffn = args.ffn_hidden_size or 4 * h

    # attention: QKV (q full + 2 kv groups) + output projection
    attn_params = (q_proj * h                      # Q
                   + 2 * (q_proj // args.num_attention_heads) * groups * h  # KV
                   + q_proj * h)                   # proj
    # dense mlp + the two layernorms
    dense_layer = attn_params + 2 * h * (ffn * gated + 2)

    num_experts = getattr(args, "num_experts", None) or 0
    if num_experts:
        moe_ffn = getattr(args, "moe_ffn_hidden_size", None) or ffn
        moe_layer = attn_params + 2 * h * (moe_ffn * num_experts * gated + 2)
        n_moe = args.num_layers  # every layer MoE unless a pattern narrows it
        n_dense = 0
    else:
        moe_layer = 0
        n_moe = 0
        n_dense = args.num_layers

    block = dense_layer * n_dense + moe_layer * n_moe + 2 * h  # + final norm
    vocab = getattr(args, "padded_vocab_size", None) or args.vocab_size
    embedding = h * vocab
    n_embed = 2 * embedding if getattr(
        args, "untie_embeddings_and_output_weights", False) else embedding
    total = block + n_embed

    pp = args.pipeline_model_parallel_size
    tp = args.tensor_model_parallel_size
    most_loaded = (block / pp + embedding) / tp
    if getattr(args, "untie_embeddings_and_output_weights", False) and pp == 1:
        most_loaded += embedding / tp

    if verbose:
        print(f"  total parameters: {total / 1e9:.2f} B "
              f"(block {block / 1e9:.2f} B, embeddings {n_embed / 1e9:.2f} B)")
        print(f"  parameters on most-loaded shard: {most_loaded / 1e9:.4f} B")

    # bf16 param + bf16 grad copy in buffer (2+2) plus fp32 master/m/v (12);
    # the distributed optimizer shards the fp32 state across DP
    dp = max(getattr(args, "data_parallel_size", 1) or 1, 1)
    bytes_per_param = (4 + 12 / dp
                       if getattr(args, "use_distributed_optimizer", False)
                       else 16)
    # main_grad is fp32 in this framework (HIP wgrad accumulates fp32): +4
    bytes_per_param += 2
    return most_loaded * bytes_per_param


def compute_activation_memory(args, num_microbatches=None, verbose=False):
    """Bytes of activations on pipeline stage 0 (selective recompute)."""
    h = args.hidden_size
    s = args.seq_length
    b = args.micro_batch_size
    ffn = args.ffn_hidden_size or 4 * h
    pp = args.pipeline_model_parallel_size
    tp = args.tensor_model_parallel_size
    vocab = getattr(args, "padded_vocab_size", None) or args.vocab_size

    per_layer = s * b * h * (18 + 4 * (ffn / h))
    if verbose:
        print(f"  activation per layer: {per_layer / tp / MiB:.1f} MiB")
    act = per_layer * args.num_layers
    # embedding input ids (int64) + embedding dropout, pp microbatches deep
    act += 8 * s * b * pp + s * b * h * pp

    vpp = getattr(args, "virtual_pipeline_model_parallel_size", None)
    if vpp:
        penalty = 1 + (pp - 1) / (pp * vpp)
        act *= penalty
        if verbose:
            print(f"  interleaved-schedule penalty: {penalty:.2f}")
    elif pp > 1 and num_microbatches is not None:
        act *= min(1.0, num_microbatches / pp)

    if pp == 1:
        # logits + cross-entropy intermediates
        act += s * b * h * 4 * (1 + vocab / h)
    return act / tp


def report_theoretical_memory(args, num_microbatches=None, verbose=False):
    weights_opt = compute_weight_and_optimizer_memory(args, verbose=verbose)
    act = compute_activation_memory(args, num_microbatches, verbose=verbose)
    total = weights_opt + act
    print(f"Theoretical memory: weights+optimizer {weights_opt / MiB:.0f} MiB, "
          f"activations {act / MiB:.0f} MiB, total {total / MiB:.0f} MiB "
          f"({100 * total / MI355X_HBM_BYTES:.1f}% of one MI355X's 288 GB)")
    return total
