#!/usr/bin/env python3
"""Standalone flash-attention kernel microbenchmark (bench shapes)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from megatronapp_amd import ops

def main():
    torch.cuda.set_device(0)
    sq, b, nh, d = 2048, 2, 16, 128
    scale = d ** -0.5
    q = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q); v = torch.randn_like(q)
    o, lse = ops.get_ops().attn_fwd(q, k, v, scale, True)
    do = torch.randn_like(q)
    flops_fwd = 2 * 2 * b * nh * sq * sq * d / 2  # causal half
    reps = 20
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(reps):
        o, lse = ops.get_ops().attn_fwd(q, k, v, scale, True)
    torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / reps
    print(f"attn_fwd: {dt*1e6:.0f} us  {flops_fwd/dt/1e12:.0f} TF")
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(reps):
        dq, dk, dv = ops.get_ops().attn_bwd(do, q, k, v, o, lse, scale, True)
    torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / reps
    print(f"attn_bwd: {dt*1e6:.0f} us  {2.5*flops_fwd/dt/1e12:.0f} TF-equiv")
    if hasattr(ops.get_ops(), "attn_fwd_t"):
        ot, _ = ops.get_ops().attn_fwd_t(q, k, v, scale, True)
        err = (ot.float() - o.float()).abs().max()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(reps):
            ops.get_ops().attn_fwd_t(q, k, v, scale, True)
        torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / reps
        print(f"attn_fwd_t (transposed-S): {dt*1e6:.0f} us  "
              f"{flops_fwd/dt/1e12:.0f} TF  maxerr-vs-default {float(err):.4f}")
    for lvl, name in ((1, "noPV"), (2, "noSM"), (3, "loader-only")):
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(reps):
            ops.get_ops().attn_fwd_ablate(q, k, v, scale, lvl)
        torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / reps
        print(f"attn_fwd ablate {name}: {dt*1e6:.0f} us")

if __name__ == "__main__":
    main()
