"""Decode throughput: eager KV-cache loop vs hipGraph-captured step.

python scripts/bench_decode.py [--model gpt3-1.3b] [--batch 8] [--new 128]
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _set_fp8(model, enabled):
    """Toggle the serving fast path by hiding/restoring fp8 attrs."""
    for mod in model.modules():
        w = getattr(mod, "weight", None)
        if w is None:
            continue
        if enabled and hasattr(w, "_fp8_data_stash"):
            w.fp8_data, w.fp8_scale = w._fp8_data_stash
            del w._fp8_data_stash
        elif not enabled and hasattr(w, "fp8_data"):
            w._fp8_data_stash = (w.fp8_data, w.fp8_scale)
            del w.fp8_data
            del w.fp8_scale


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="gpt3-1.3b")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--prompt", type=int, default=128)
    ap.add_argument("--new", type=int, default=128)
    ap.add_argument("--fp8", action="store_true",
                    help="also time the fp8 (e4m3 _scaled_mm) weights path")
    args = ap.parse_args()

    import bench
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    from megatronapp_amd.core.inference.static_engine import (
        get_inference_engine)
    from megatronapp_amd.core.inference.sampling_params import SamplingParams
    from megatronapp_amd.training.tokenizer import NullTokenizer

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29377")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    torch.distributed.init_process_group("nccl", rank=0, world_size=1)
    torch.cuda.set_device(0)
    parallel_state.initialize_model_parallel()
    model_parallel_cuda_manual_seed(1)

    spec = bench.MODELS[args.model]
    cfg = TransformerConfig(
        num_layers=spec["num_layers"], hidden_size=spec["hidden_size"],
        num_attention_heads=spec["num_attention_heads"],
        ffn_hidden_size=spec["ffn_hidden_size"], hidden_dropout=0.0,
        attention_dropout=0.0, bf16=True, params_dtype=torch.bfloat16,
        masked_softmax_fusion=True)
    with torch.device("cuda"):
        model = GPTModel(config=cfg,
                         transformer_layer_spec=get_gpt_layer_local_spec(
                             use_flash=False),
                         vocab_size=spec["vocab_size"],
                         max_sequence_length=args.prompt + args.new,
                         pre_process=True, post_process=True).eval()

    tok = NullTokenizer(spec["vocab_size"] - 1)
    prompts = [" ".join(str((i * 7 + j) % 500) for j in range(args.prompt))
               for i in range(args.batch)]
    sp = SamplingParams(num_tokens_to_generate=args.new, top_k=1)

    modes = [(False, False, "eager"), (True, False, "hipgraph")]
    if args.fp8:
        from megatronapp_amd.inference.fp8 import quantize_model_fp8
        n = quantize_model_fp8(model)
        print(f"fp8: quantized {n} linears (e4m3, per-row scales)")
        modes.append((False, True, "eager+fp8"))
        modes.append((True, True, "hipgraph+fp8"))
    ref_text = None
    for use_graphs, use_fp8, label in modes:
        _set_fp8(model, use_fp8)
        engine = get_inference_engine(model, tok, max_batch_size=args.batch)
        engine.controller.use_hip_graphs = use_graphs
        engine.generate(prompts[:2], SamplingParams(num_tokens_to_generate=8,
                                                    top_k=1))  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = engine.generate(prompts, sp)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        tps = args.batch * args.new / dt
        txt = out[0].generated_text
        match = "" if ref_text is None else \
            f" match={'Y' if txt == ref_text else 'n'}"
        if ref_text is None:
            ref_text = txt
        print(f"{label:13s}: {dt:.2f}s  {tps:8.1f} tokens/s "
              f"({dt / args.new * 1000:.2f} ms/step){match} "
              f"sample={txt[:40]!r}")


if __name__ == "__main__":
    main()
