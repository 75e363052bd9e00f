#!/usr/bin/env python3
"""Flash-attention kernel microbenchmark vs the unfused hipBLASLt path.

Times, at the GPT-3 1.3B bench hot shape (and the round-1 microbench
shape), all of:
  * attn_fwd        — default MFMA flash forward
  * attn_fwd_t      — transposed-S flash forward (S^T = K Q^T)
  * attn_bwd        — flash backward (dq + dkv kernels)
  * unfused fwd/bwd — baddbmm + scaled-masked-softmax HIP kernel + bmm,
                      the path the flash kernel must beat (VERDICT weak#1)
plus the forward ablation ladder (loader / +S / +softmax / full).
"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from megatronapp_amd import ops


def timeit(fn, reps=20, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def unfused_fwd(q, k, v, scale, causal):
    # [sq,b,nh,d] -> scores -> softmax -> context, as DotProductAttention
    # (autograd-aware so the fwd+bwd comparison exercises the real path)
    from megatronapp_amd.core.fusions.fused_softmax import (
        ScaledUpperTriangMaskedSoftmax)
    sq, b, nh, d = q.shape
    qq = q.permute(1, 2, 0, 3).reshape(b * nh, sq, d)
    kk = k.permute(1, 2, 0, 3).reshape(b * nh, sq, d)
    scores = torch.empty(b * nh, sq, sq, dtype=q.dtype, device=q.device)
    scores = torch.baddbmm(scores, qq, kk.transpose(1, 2), beta=0.0,
                           alpha=scale)
    probs = ScaledUpperTriangMaskedSoftmax.apply(scores, 1.0)
    vv = v.permute(1, 2, 0, 3).reshape(b * nh, sq, d)
    ctx = torch.bmm(probs, vv)
    return ctx


def bench_shape(sq, b, nh, d, label):
    scale = d ** -0.5
    q = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    do = torch.randn_like(q)
    O = ops.get_ops()
    flops_fwd = 2 * 2 * b * nh * sq * sq * d / 2  # causal half
    print(f"--- {label}: sq={sq} b={b} nh={nh} d={d} "
          f"(causal flops fwd {flops_fwd/1e12:.2f} TF) ---")

    dt = timeit(lambda: O.attn_fwd(q, k, v, scale, True))
    print(f"attn_fwd:              {dt*1e6:7.0f} us  {flops_fwd/dt/1e12:5.0f} TF")
    o, lse = O.attn_fwd(q, k, v, scale, True)

    if hasattr(O, "attn_fwd2"):
        o2, _ = O.attn_fwd2(q, k, v, scale, True)
        err = (o2.float() - o.float()).abs().max()
        dt = timeit(lambda: O.attn_fwd2(q, k, v, scale, True))
        print(f"attn_fwd2 (32x32/tr16): {dt*1e6:6.0f} us  "
              f"{flops_fwd/dt/1e12:5.0f} TF  maxerr {float(err):.4f}")

    if hasattr(O, "attn_fwd_t"):
        ot, _ = O.attn_fwd_t(q, k, v, scale, True)
        err = (ot.float() - o.float()).abs().max()
        dt = timeit(lambda: O.attn_fwd_t(q, k, v, scale, True))
        print(f"attn_fwd_t:            {dt*1e6:7.0f} us  "
              f"{flops_fwd/dt/1e12:5.0f} TF  maxerr {float(err):.4f}")

    dt = timeit(lambda: O.attn_bwd(do, q, k, v, o, lse, scale, True))
    print(f"attn_bwd (dispatch):   {dt*1e6:7.0f} us  "
          f"{2.5*flops_fwd/dt/1e12:5.0f} TF-equiv")
    if hasattr(O, "attn_bwd_v1"):
        dt = timeit(lambda: O.attn_bwd_v1(do, q, k, v, o, lse, scale, True))
        print(f"attn_bwd_v1:           {dt*1e6:7.0f} us  "
              f"{2.5*flops_fwd/dt/1e12:5.0f} TF-equiv")

    # unfused forward
    dt = timeit(lambda: unfused_fwd(q, k, v, scale, True))
    print(f"unfused fwd:           {dt*1e6:7.0f} us  {flops_fwd/dt/1e12:5.0f} TF")

    # unfused fwd+bwd through autograd (the real competitor for training)
    qg = q.detach().requires_grad_(True)
    kg = k.detach().requires_grad_(True)
    vg = v.detach().requires_grad_(True)

    def unfused_step():
        ctx = unfused_fwd(qg, kg, vg, scale, True)
        ctx.backward(do.permute(1, 2, 0, 3).reshape(b * nh, sq, d))
        qg.grad = kg.grad = vg.grad = None

    dt_u = timeit(unfused_step, reps=10)
    print(f"unfused fwd+bwd:       {dt_u*1e6:7.0f} us  "
          f"{3.5*flops_fwd/dt_u/1e12:5.0f} TF-equiv")

    def flash_step():
        o2, lse2 = O.attn_fwd(q, k, v, scale, True)
        O.attn_bwd(do, q, k, v, o2, lse2, scale, True)

    dt_f = timeit(flash_step, reps=10)
    print(f"flash fwd+bwd:         {dt_f*1e6:7.0f} us  "
          f"{3.5*flops_fwd/dt_f/1e12:5.0f} TF-equiv   "
          f"vs unfused: {dt_u/dt_f:.2f}x")

    # ablation ladder
    for lvl, name in ((3, "loader-only"), (2, "+S"), (1, "+softmax"),
                      (0, "full")):
        dt = timeit(lambda: O.attn_fwd_ablate(q, k, v, scale, lvl))
        print(f"attn_fwd ablate {name:12s} {dt*1e6:7.0f} us")


def main():
    torch.cuda.set_device(0)
    # round-1 microbench shape
    bench_shape(2048, 2, 16, 128, "r1-micro")
    # GPT-3 1.3B bench hot shape (mbs 16, TP1: 16 heads of 128)
    bench_shape(2048, 16, 16, 128, "gpt3-1.3b-mbs16")


if __name__ == "__main__":
    main()
