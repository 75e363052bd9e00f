#!/usr/bin/env python3
"""Probe torch._grouped_mm on MI355X: correctness vs per-expert loop,
autograd support, and speed at mixtral-8x1b bench shapes."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def main():
    torch.cuda.set_device(0)
    E, h, f = 8, 2048, 4096  # mixtral-8x1b-ish fc1 (gated: 2f out)
    tokens = 16 * 2048 * 2 // E * E  # divisible
    torch.manual_seed(7)
    x = torch.randn(tokens, h, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(E, h, f, device="cuda", dtype=torch.bfloat16) * 0.02
    # uneven segment sizes
    sizes = torch.tensor([tokens // E + (256 if e % 2 == 0 else -256)
                         for e in range(E)], device="cuda")
    sizes[-1] += tokens - int(sizes.sum())
    offs = torch.cumsum(sizes, 0).to(torch.int32)
    print("sizes:", sizes.tolist())

    try:
        y = torch._grouped_mm(x, w, offs=offs)
        print("fwd OK:", y.shape, y.dtype)
    except Exception as e:
        print("grouped_mm FAILED:", repr(e))
        return

    # correctness vs loop
    y_ref = torch.empty_like(y)
    start = 0
    for e in range(E):
        n = int(sizes[e])
        y_ref[start:start + n] = x[start:start + n] @ w[e]
        start += n
    err = (y.float() - y_ref.float()).abs().max() / y_ref.float().abs().max()
    print("rel maxerr vs loop:", float(err))

    # autograd
    xg = x.detach().requires_grad_(True)
    wg = w.detach().requires_grad_(True)
    try:
        out = torch._grouped_mm(xg, wg, offs=offs)
        out.sum().backward()
        print("bwd OK: dx", xg.grad.shape, "dw", wg.grad.shape)
    except Exception as e:
        print("autograd FAILED:", repr(e))

    # timing
    def t(fn, reps=20):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / reps

    flops = 2 * tokens * h * f
    dt_g = t(lambda: torch._grouped_mm(x, w, offs=offs))
    print(f"grouped_mm: {dt_g*1e6:.0f} us  {flops/dt_g/1e12:.0f} TF")

    def loop():
        outs = []
        start = 0
        for e in range(E):
            n = int(sizes[e])
            outs.append(x[start:start + n] @ w[e])
            start += n
        return torch.cat(outs)

    dt_l = t(loop)
    print(f"python loop: {dt_l*1e6:.0f} us  {flops/dt_l/1e12:.0f} TF  "
          f"grouped speedup {dt_l/dt_g:.2f}x")


if __name__ == "__main__":
    main()
