"""Microbench for the backward elementwise/reduction kernels at the
gpt3-1.3b bench shapes (mbs16: rows = 32768).

  python tools/bench_eltwise.py          # on an MI355X via gpurun
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from megatronapp_amd import ops


def timeit(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(True)
    t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(iters):
        fn()
    t1.record()
    torch.cuda.synchronize()
    return t0.elapsed_time(t1) / iters * 1000  # us


def main():
    assert ops.have_ops()
    o = ops.get_ops()
    R = 32768
    torch.manual_seed(0)

    print("== colsum_accum (bias grads; bound = R*F*2 / 8TB/s) ==")
    for F in (2048, 6144, 8192):
        dy = torch.randn(R, F, device="cuda", dtype=torch.bfloat16)
        out = torch.zeros(F, device="cuda", dtype=torch.float32)
        us = timeit(lambda: o.colsum_accum(dy, out))
        bound = R * F * 2 / 8e12 * 1e6
        print(f"  F={F:5d}: {us:7.1f} us   (HBM bound {bound:5.1f} us)")

    print("== bias_gelu_bwd [32768, 8192] (bound ~ 3 passes) ==")
    F = 8192
    dy = torch.randn(R, F, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(R, F, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(F, device="cuda", dtype=torch.bfloat16)
    us = timeit(lambda: o.bias_gelu_bwd(dy, x, b))
    print(f"  {us:7.1f} us   (HBM bound {R * F * 6 / 8e12 * 1e6:5.1f} us)")

    print("== layernorm_bwd [32768, 2048] + dres (bound ~ 4 bf16 passes) ==")
    H = 2048
    x2 = torch.randn(R, H, device="cuda", dtype=torch.bfloat16)
    dy2 = torch.randn(R, H, device="cuda", dtype=torch.bfloat16)
    w2 = torch.randn(H, device="cuda", dtype=torch.bfloat16)
    b2 = torch.randn(H, device="cuda", dtype=torch.bfloat16)
    dres = torch.randn(R, H, device="cuda", dtype=torch.bfloat16)
    y2, mean, invstd = o.layernorm_fwd(x2, w2, b2, 1e-5)
    us = timeit(lambda: o.layernorm_bwd(dy2, x2, w2, mean, invstd,
                                    dres=dres))
    print(f"  {us:7.1f} us   (HBM bound {R * H * 8 / 8e12 * 1e6:5.1f} us)")

    print("== rmsnorm_bwd [32768, 2048] + dres ==")
    yr, invrms = o.rmsnorm_fwd(x2, w2, 1e-5)
    us = timeit(lambda: o.rmsnorm_bwd(dy2, x2, w2, invrms, dres=dres))
    print(f"  {us:7.1f} us   (HBM bound {R * H * 8 / 8e12 * 1e6:5.1f} us)")

    print("== qkv wgrad: separate (wgrad_accum + colsum) vs BGRADB ==")
    out_f, in_f = 6144, 2048
    g = torch.randn(R, out_f, device="cuda", dtype=torch.bfloat16)
    xin = torch.randn(R, in_f, device="cuda", dtype=torch.bfloat16)
    mg = torch.zeros(out_f, in_f, device="cuda", dtype=torch.float32)
    db = torch.zeros(out_f, device="cuda", dtype=torch.float32)
    us_sep = timeit(lambda: (o.wgrad_accum(g, xin, mg),
                             o.colsum_accum(g, db)))
    ok = o.wgrad_accum_bgrad(g, xin, mg, db)
    print(f"  bgrad epilogue supported: {ok}")
    if ok:
        us_fused = timeit(lambda: o.wgrad_accum_bgrad(g, xin, mg, db))
        print(f"  separate {us_sep:7.1f} us   fused {us_fused:7.1f} us   "
              f"({us_sep / us_fused:.2f}x)")
    else:
        print(f"  separate {us_sep:7.1f} us   (fused unavailable)")


if __name__ == "__main__":
    if "--adam" in sys.argv:
        bench_adam()
    else:
        main()


def bench_adam():
    assert ops.have_ops()
    o = ops.get_ops()
    n = 500_000_000
    p = torch.randn(n, device="cuda")
    g = torch.randn(n, device="cuda")
    for st_dt, label in ((torch.float32, "fp32"), (torch.bfloat16, "bf16")):
        m = torch.zeros(n, device="cuda", dtype=st_dt)
        v = torch.zeros(n, device="cuda", dtype=st_dt)
        us = timeit(lambda: o.adamw_flat(p, g, m, v, 1e-3, 0.9, 0.999,
                                         1e-8, 0.01, 2), iters=10)
        bw = n * (16 + (8 if st_dt == torch.float32 else 4) * 2) / (us / 1e6) / 1e12
        print(f"  adamw_flat {label} states: {us:8.1f} us  ({bw:5.2f} TB/s)")

