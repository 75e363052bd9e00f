#!/usr/bin/env python3
"""Flagship training-step benchmark (driver contract — see repo docs).

Measures whole-job training throughput (tokens/s) for the BASELINE.json
headline config: GPT-3 1.3B, PP=4 at N>=4 (dp elsewhere), bf16, synthetic
data, random-init weights.

  python bench.py --gpus N --steps K --warmup W
  (N>1 is launched by the driver via torch.distributed.run, 1 rank/GPU)
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch
import torch.distributed as dist

# arch="gpt" -> learned positions, GeLU, LayerNorm, biases, tied embeddings
# arch="llama" -> rope, SwiGLU, RMSNorm, no biases, untied embeddings (+GQA)
MODELS = {
    "gpt3-1.3b": dict(num_layers=24, hidden_size=2048, num_attention_heads=16,
                      ffn_hidden_size=8192, seq_length=2048, vocab_size=51200,
                      arch="gpt"),
    "gpt3-345m": dict(num_layers=24, hidden_size=1024, num_attention_heads=16,
                      ffn_hidden_size=4096, seq_length=2048, vocab_size=51200,
                      arch="gpt"),
    "gpt-tiny": dict(num_layers=4, hidden_size=256, num_attention_heads=4,
                     ffn_hidden_size=1024, seq_length=512, vocab_size=8192,
                     arch="gpt"),
    "llama3-8b": dict(num_layers=32, hidden_size=4096,
                      num_attention_heads=32, num_query_groups=8,
                      ffn_hidden_size=14336, seq_length=2048,
                      vocab_size=128256, arch="llama"),
    "llama-1b": dict(num_layers=16, hidden_size=2048,
                     num_attention_heads=32, num_query_groups=8,
                     ffn_hidden_size=8192, seq_length=2048,
                     vocab_size=32000, arch="llama"),
    "mixtral-8x1b": dict(num_layers=16, hidden_size=2048,
                         num_attention_heads=32, num_query_groups=8,
                         ffn_hidden_size=8192, seq_length=2048,
                         vocab_size=32000, arch="llama",
                         num_moe_experts=8, moe_router_topk=2),
}

BASELINE_TOKENS_PER_S = 16 * 2048 / 0.722  # BASELINE.md row 2 (4x RTX4090)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--model", default="gpt3-1.3b", choices=list(MODELS))
    p.add_argument("--micro-batch-size", type=int, default=None,
                   help="default: fattest microbatch the topology allows "
                        "(16 at PP=1; 4 with a deep-enough pipeline)")
    p.add_argument("--global-batch-size", type=int, default=None,
                   help="default: 16 * n_gpus (weak scaling)")
    p.add_argument("--pp", type=int, default=None,
                   help="pipeline parallel size (default min(gpus,4) for >=4 GPUs)")
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--trace", action="store_true", help="enable MegaScan tracing")
    p.add_argument("--precision-aware-optimizer", action="store_true",
                   help="bf16 Adam exp_avg/exp_avg_sq (fp32 math in-kernel)")
    p.add_argument("--moe-sequential-experts", action="store_true",
                   help="use SequentialMLP instead of the grouped-GEMM "
                        "experts (A/B debugging)")
    p.add_argument("--torch-profile", default=None, metavar="OUT",
                   help="run 2 extra steps under torch.profiler (with python "
                        "stacks) after the timed loop and write the table to "
                        "OUT (kernel attribution, not part of the metric)")
    p.add_argument("--trace-dir", default="trace_out")
    p.add_argument("--attention", default="flash", choices=["flash", "fused"])
    p.add_argument("--fp8", action="store_true",
                   help="fp8 (e4m3) forward+dgrad+wgrad GEMMs (wgrad "
                        "accumulates fp32; NOT the headline dtype)")
    p.add_argument("--seq-length", type=int, default=None,
                   help="override the model's sequence length")
    p.add_argument("--no-overlap-grad-reduce", action="store_true")
    return p.parse_args()


def main():
    args = parse_args()
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    on_gpu = torch.cuda.is_available()

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29399")
    backend = "nccl" if on_gpu else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend, rank=rank,
                                world_size=world_size)
    if on_gpu:
        # modulo so oversubscribed runs (4 ranks on 1 GPU) stay valid
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
    device = (torch.device("cuda", local_rank % torch.cuda.device_count())
              if on_gpu else torch.device("cpu"))

    n = world_size
    if args.pp is None:
        pp = min(n, 4) if n >= 4 else 1
    else:
        pp = args.pp
    tp = args.tp
    dp = n // (pp * tp)
    assert dp * pp * tp == n, f"world {n} != dp{dp}*pp{pp}*tp{tp}"

    spec = MODELS[args.model]
    seq = args.seq_length or spec["seq_length"]
    vocab = spec["vocab_size"]
    gbs = args.global_batch_size or 16 * n
    mbs = args.micro_batch_size
    if mbs is None:
        # fatter microbatches feed fatter GEMMs (+15% measured at PP=1);
        # with a pipeline keep enough microbatches to bound the bubble
        spd = gbs // dp  # samples per dp rank
        if pp == 1:
            # 8B-class models (h >= 4096) OOM 288 GB at mbs 16
            # (weights+optimizer ~150 GB + ~4 GB activations/layer)
            cap = 8 if spec["hidden_size"] >= 4096 else 16
            mbs = min(cap, spd)
        elif spd % 4 == 0 and spd // 4 >= pp:
            mbs = 4
        else:
            mbs = 2
    num_microbatches = gbs // (mbs * dp)
    assert num_microbatches * mbs * dp == gbs

    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.distributed import (
        DistributedDataParallel, DistributedDataParallelConfig)
    from megatronapp_amd.core.distributed.finalize_model_grads import (
        finalize_model_grads)
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.optimizer import (
        OptimizerConfig, get_megatron_optimizer)
    from megatronapp_amd.core.pipeline_parallel import get_forward_backward_func
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    from megatronapp_amd.core.transformer_config import TransformerConfig

    parallel_state.initialize_model_parallel(
        tensor_model_parallel_size=tp, pipeline_model_parallel_size=pp)
    model_parallel_cuda_manual_seed(1234)
    torch.manual_seed(1234)

    bf16 = on_gpu
    llama_style = spec.get("arch") == "llama"
    config = TransformerConfig(
        num_layers=spec["num_layers"], hidden_size=spec["hidden_size"],
        num_attention_heads=spec["num_attention_heads"],
        ffn_hidden_size=spec["ffn_hidden_size"],
        tensor_model_parallel_size=tp, pipeline_model_parallel_size=pp,
        bf16=bf16, params_dtype=torch.bfloat16 if bf16 else torch.float32,
        pipeline_dtype=torch.bfloat16 if bf16 else torch.float32,
        hidden_dropout=0.0, attention_dropout=0.0,
        position_embedding_type=("learned_absolute" if llama_style is False
                                 else "rope"),
        normalization="LayerNorm" if not llama_style else "RMSNorm",
        activation_func="gelu" if not llama_style else "silu",
        gated_linear_unit=llama_style,
        add_bias_linear=not llama_style, masked_softmax_fusion=True,
        num_query_groups=spec.get("num_query_groups"),
        num_moe_experts=spec.get("num_moe_experts"),
        moe_router_topk=spec.get("moe_router_topk", 2),
        moe_router_load_balancing_type="aux_loss",
        moe_aux_loss_coeff=0.01 if spec.get("num_moe_experts") else 0.0,
        finalize_model_grads_func=finalize_model_grads,
        sequence_parallel=(tp > 1))

    pre = parallel_state.is_pipeline_first_stage()
    post = parallel_state.is_pipeline_last_stage()
    ctx = torch.device(device) if on_gpu else torch.device("cpu")
    with ctx:
        model = GPTModel(
            config=config,
            transformer_layer_spec=get_gpt_layer_local_spec(
                normalization=config.normalization,
                num_experts=spec.get("num_moe_experts"),
                moe_grouped_gemm=not args.moe_sequential_experts,
                use_flash=(args.attention == "flash")),
            vocab_size=vocab, max_sequence_length=seq,
            position_embedding_type=config.position_embedding_type,
            pre_process=pre, post_process=post,
            share_embeddings_and_output_weights=not llama_style)
    ddp_config = DistributedDataParallelConfig(
        overlap_grad_reduce=not args.no_overlap_grad_reduce,
        use_distributed_optimizer=(dp > 1),
        grad_reduce_in_fp32=True)
    if args.fp8 and on_gpu:
        from megatronapp_amd.core.fp8 import enable_fp8_training
        n8 = enable_fp8_training(model)
        if rank == 0:
            print(f"fp8 training: {n8} linears flagged")
    model = DistributedDataParallel(config, ddp_config, model)
    optimizer = get_megatron_optimizer(
        OptimizerConfig(lr=1e-4, weight_decay=0.1, clip_grad=1.0, bf16=bf16,
                        use_distributed_optimizer=(dp > 1),
                        overlap_param_gather=(dp > 1),
                        use_precision_aware_optimizer=args.precision_aware_optimizer,
                        exp_avg_dtype="bf16" if args.precision_aware_optimizer else "fp32",
                        exp_avg_sq_dtype="bf16" if args.precision_aware_optimizer else "fp32"),
        [model])

    tracer = None
    if args.trace:
        from megatronapp_amd.training.trace import Tracer
        tracer = Tracer.initialize(trace_dir=args.trace_dir, interval=1,
                                   continuous_iters=args.steps,
                                   granularity="full")

    # synthetic fixed batch (regenerated views per microbatch index)
    g = torch.Generator(device="cpu").manual_seed(4321 + rank)
    tokens_all = torch.randint(0, vocab, (mbs * num_microbatches, seq + 1),
                               generator=g).to(device)
    position_ids = torch.arange(seq, device=device).unsqueeze(0).expand(mbs, -1)
    mb_counter = {"i": 0}

    def forward_step(data_iterator, m):
        i = mb_counter["i"] % num_microbatches
        mb_counter["i"] += 1
        tok = tokens_all[i * mbs:(i + 1) * mbs]
        batch = {"tokens": tok[:, :-1], "labels": tok[:, 1:]}

        def loss_func(out):
            loss = out.float().mean()
            return loss, {"lm loss": loss.detach()}

        return m(batch["tokens"], position_ids,
                 labels=batch["labels"]), loss_func

    fb = get_forward_backward_func()

    def one_step(it):
        if tracer is not None:
            tracer.iteration_begin(it)
        if hasattr(optimizer, "finish_param_sync"):
            optimizer.finish_param_sync()   # overlapped ZeRO-1 gather
        model.zero_grad_buffer()
        optimizer.zero_grad()
        out = fb(forward_step_func=forward_step, data_iterator=None,
                 model=model, num_microbatches=num_microbatches,
                 seq_length=seq, micro_batch_size=mbs, forward_only=False)
        ok, grad_norm, _ = optimizer.step()
        if tracer is not None:
            tracer.iteration_end()
        return out

    for it in range(args.warmup):
        one_step(it)

    dist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for it in range(args.steps):
        one_step(args.warmup + it)
    dist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    if tracer is not None:
        tracer.shutdown()

    if args.torch_profile and rank == 0 and on_gpu:
        from torch.profiler import ProfilerActivity, profile
        cfg = torch._C._profiler._ExperimentalConfig(verbose=True)
        with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                     with_stack=True, experimental_config=cfg) as prof:
            for it in range(2):
                one_step(args.warmup + args.steps + it)
        with open(args.torch_profile, "w") as f:
            f.write(prof.key_averages(group_by_stack_n=6).table(
                sort_by="self_cuda_time_total", row_limit=60,
                max_src_column_width=160))
        prof.export_stacks(args.torch_profile + ".stacks",
                           "self_cuda_time_total")

    # max elapsed over ranks
    t = torch.tensor([elapsed], dtype=torch.float64, device=device
                     if backend == "nccl" else "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = t.item()

    tokens_per_step = gbs * seq
    tokens_per_s = tokens_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        par = []
        if dp > 1:
            par.append(f"dp{dp}")
        if tp > 1:
            par.append(f"tp{tp}")
        if pp > 1:
            par.append(f"pp{pp}")
        result = {
            "metric": "tokens/sec/node",
            "value": round(tokens_per_s, 1),
            "unit": "tokens/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(tokens_per_s / BASELINE_TOKENS_PER_S, 3),
            "dtype": ("fp8" if args.fp8 and on_gpu else
                      "bf16" if bf16 else "fp32"),
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": gbs,
                "seq_len": seq,
                "micro_batch": mbs,
                "parallelism": "-".join(par) or "dp1",
                "trace": bool(args.trace),
            },
        }
        print(json.dumps(result), flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
