#!/usr/bin/env python3
"""Probe fp8 GEMM support on MI355X: torch._scaled_mm with e4m3fn
(OCP — gfx950 uses OCP fp8, not the MI300 fnuz variant), correctness
and speed vs bf16 mm at a decode-projection shape."""
import time
import torch


def t(fn, reps=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main():
    torch.cuda.set_device(0)
    for dt in (torch.float8_e4m3fn, torch.float8_e5m2):
        try:
            a = torch.randn(16, 16, device="cuda").to(dt)
            print(dt, "tensor create OK")
        except Exception as e:
            print(dt, "create FAILED:", repr(e))

    M, K, N = 4096, 2048, 8192  # MLP fc1-ish
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.02
    ref = x @ w.t()
    dt_bf16 = t(lambda: x @ w.t())
    flops = 2 * M * N * K
    print(f"bf16 mm: {dt_bf16*1e6:.0f} us {flops/dt_bf16/1e12:.0f} TF")

    try:
        xs = x.abs().amax() / 448.0
        ws = w.abs().amax() / 448.0
        x8 = (x / xs).to(torch.float8_e4m3fn)
        w8 = (w / ws).to(torch.float8_e4m3fn)
        y = torch._scaled_mm(x8, w8.t(), scale_a=xs.float(),
                             scale_b=ws.float(), out_dtype=torch.bfloat16)
        err = (y.float() - ref.float()).abs().max() / ref.float().abs().max()
        print("scaled_mm tensorwise OK, rel maxerr", float(err))
        dt8 = t(lambda: torch._scaled_mm(x8, w8.t(), scale_a=xs.float(),
                                         scale_b=ws.float(),
                                         out_dtype=torch.bfloat16))
        print(f"fp8 scaled_mm: {dt8*1e6:.0f} us {flops/dt8/1e12:.0f} TF  "
              f"{dt_bf16/dt8:.2f}x vs bf16")
    except Exception as e:
        print("scaled_mm FAILED:", repr(e))

    # rowwise scaling variant
    try:
        xs = x.abs().amax(dim=1, keepdim=True) / 448.0
        ws = w.abs().amax(dim=1, keepdim=True) / 448.0
        x8 = (x / xs).to(torch.float8_e4m3fn)
        w8 = (w / ws).to(torch.float8_e4m3fn)
        y = torch._scaled_mm(x8, w8.t(), scale_a=xs.float(),
                             scale_b=ws.t().float(),
                             out_dtype=torch.bfloat16)
        err = (y.float() - ref.float()).abs().max() / ref.float().abs().max()
        print("scaled_mm rowwise OK, rel maxerr", float(err))
        dt8 = t(lambda: torch._scaled_mm(x8, w8.t(), scale_a=xs.float(),
                                         scale_b=ws.t().float(),
                                         out_dtype=torch.bfloat16))
        print(f"fp8 rowwise: {dt8*1e6:.0f} us {flops/dt8/1e12:.0f} TF")
    except Exception as e:
        print("rowwise scaled_mm FAILED:", repr(e))


if __name__ == "__main__":
    main()
