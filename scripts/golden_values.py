#!/usr/bin/env python3
"""Golden-value loss-curve harness (reference tests/functional_tests/:
model_config.yaml + golden_values_*.json compared within tolerance).

Runs N deterministic fp32 CPU steps of scaled-down versions of the
BASELINE config families (same architectural features — the harness
exists to catch broken backwards like round 1's residual-grad bug, and
feature coverage matters, size doesn't):

  gpt      LayerNorm + GeLU + learned positions + tied embeddings
  llama    RMSNorm + SwiGLU + RoPE + GQA + untied
  mixtral  llama + top-2 MoE router + aux loss
  fbdgrid  gpt at PP-ready layer count (single-rank no-pipelining here)

``python scripts/golden_values.py --write`` regenerates
tests/golden/golden_values_cpu.json; the pytest in
tests/test_golden_values.py compares a fresh run against it.
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

GOLDEN_PATH = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tests", "golden",
    "golden_values_cpu.json")

CONFIGS = {
    "gpt": dict(arch="gpt", num_layers=3, hidden=64, heads=4, ffn=128,
                seq=64, vocab=256),
    "llama": dict(arch="llama", num_layers=3, hidden=64, heads=4, groups=2,
                  ffn=128, seq=64, vocab=256),
    "mixtral": dict(arch="llama", num_layers=2, hidden=64, heads=4, groups=2,
                    ffn=128, seq=64, vocab=256, experts=4, topk=2),
    "fbdgrid": dict(arch="gpt", num_layers=4, hidden=48, heads=4, ffn=96,
                    seq=32, vocab=128),
}
STEPS = 16
LR = 3e-3


def run_config(name: str, spec: dict) -> list:
    import torch.distributed as dist
    from megatronapp_amd.core import parallel_state
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig

    torch.manual_seed(1234)
    llama = spec["arch"] == "llama"
    cfg = TransformerConfig(
        num_layers=spec["num_layers"], hidden_size=spec["hidden"],
        num_attention_heads=spec["heads"],
        num_query_groups=spec.get("groups", spec["heads"]),
        ffn_hidden_size=spec["ffn"], hidden_dropout=0.0,
        attention_dropout=0.0, params_dtype=torch.float32,
        normalization="RMSNorm" if llama else "LayerNorm",
        gated_linear_unit=llama,
        activation_func="silu" if llama else "gelu",
        add_bias_linear=not llama,
        position_embedding_type="rope" if llama else "learned_absolute",
        num_moe_experts=spec.get("experts"),
        moe_router_topk=spec.get("topk", 2),
        moe_router_load_balancing_type="aux_loss",
        moe_aux_loss_coeff=0.01 if spec.get("experts") else 0.0)
    model = GPTModel(
        config=cfg,
        transformer_layer_spec=get_gpt_layer_local_spec(
            normalization=cfg.normalization,
            num_experts=spec.get("experts")),
        vocab_size=spec["vocab"], max_sequence_length=spec["seq"],
        position_embedding_type=cfg.position_embedding_type,
        pre_process=True, post_process=True,
        share_embeddings_and_output_weights=not llama)
    opt = torch.optim.AdamW(model.parameters(), lr=LR, weight_decay=0.01)

    g = torch.Generator().manual_seed(99)
    seq, vocab, b = spec["seq"], spec["vocab"], 4
    # ONE fixed batch, memorized over the steps: the loss must FALL well
    # below the ln(vocab) entropy floor, so a silently broken backward
    # (round 1's residual-grad bug) reads as a flat curve, not noise
    tokens = torch.randint(0, vocab, (b, seq + 1), generator=g)
    inp, lbl = tokens[:, :-1], tokens[:, 1:]
    pos = torch.arange(seq).unsqueeze(0).expand(b, seq)
    losses = []
    for step in range(STEPS):
        out = model(inp, pos, None, labels=lbl)
        loss = out.float().mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(round(float(loss.detach()), 6))
    assert losses[-1] < losses[0] - 0.05, (
        f"{name}: no learning ({losses[0]} -> {losses[-1]})")
    return losses


def compute_all() -> dict:
    import torch.distributed as dist
    from megatronapp_amd.core import parallel_state
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29461")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=0, world_size=1)
    if parallel_state._TENSOR_MODEL_PARALLEL_GROUP is None:
        parallel_state.initialize_model_parallel()
    return {name: run_config(name, spec) for name, spec in CONFIGS.items()}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--write", action="store_true")
    args = ap.parse_args()
    got = compute_all()
    if args.write:
        os.makedirs(os.path.dirname(GOLDEN_PATH), exist_ok=True)
        with open(GOLDEN_PATH, "w") as f:
            json.dump(got, f, indent=1)
        print(f"wrote {GOLDEN_PATH}")
        for k, v in got.items():
            print(f"  {k}: {v[0]} -> {v[-1]}")
    else:
        ref = json.load(open(GOLDEN_PATH))
        bad = []
        for k, v in got.items():
            for i, (a, b) in enumerate(zip(v, ref[k])):
                if abs(a - b) > 2e-3 * max(1.0, abs(b)):
                    bad.append((k, i, a, b))
        print("MISMATCH:" if bad else "GOLDEN OK", bad[:10])
        sys.exit(1 if bad else 0)


if __name__ == "__main__":
    main()
