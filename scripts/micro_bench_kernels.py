import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
import megatronapp_amd.ops as O
ops = O.get_ops()
def t(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6
# softmax shapes: [b*np, sq, sk] = [32, 2048, 2048] bf16
x = torch.randn(256, 2048, 2048, device="cuda", dtype=torch.bfloat16)
y = ops.scaled_upper_triang_masked_softmax_fwd(x, 0.08)
dy = torch.randn_like(x)
us = t(lambda: ops.scaled_upper_triang_masked_softmax_fwd(x, 0.08))
print(f"softmax causal fwd: {us:.1f} us  ({(x.numel()*2*1.5)/us/1e3:.0f} GB/s eff)")
us = t(lambda: ops.scaled_upper_triang_masked_softmax_bwd(dy, y, 0.08))
print(f"softmax causal bwd: {us:.1f} us")
us = t(lambda: ops.scaled_softmax_bwd(dy, y, 0.08))
print(f"softmax generic bwd: {us:.1f} us")
# colsum: [4096, 2048] bf16 -> [2048] f32
g = torch.randn(4096, 2048, device="cuda", dtype=torch.bfloat16)
acc = torch.zeros(2048, device="cuda", dtype=torch.float32)
us = t(lambda: ops.colsum_accum(g, acc))
print(f"colsum 4096x2048: {us:.1f} us ({g.numel()*2/us/1e3:.0f} GB/s)")
g3 = torch.randn(32768, 2048, device="cuda", dtype=torch.bfloat16)
acc3 = torch.zeros(2048, device="cuda", dtype=torch.float32)
us = t(lambda: ops.colsum_accum(g3, acc3))
print(f"colsum 32768x2048: {us:.1f} us ({g3.numel()*2/us/1e3:.0f} GB/s)")
g2 = torch.randn(4096, 8192, device="cuda", dtype=torch.bfloat16)
acc2 = torch.zeros(8192, device="cuda", dtype=torch.float32)
us = t(lambda: ops.colsum_accum(g2, acc2))
print(f"colsum 4096x8192: {us:.1f} us ({g2.numel()*2/us/1e3:.0f} GB/s)")
# bias gelu fwd/bwd on mbs16 shapes
gx = torch.randn(32768, 8192, device="cuda", dtype=torch.bfloat16)
gb = torch.randn(8192, device="cuda", dtype=torch.bfloat16)
gd = torch.randn_like(gx)
us = t(lambda: ops.bias_gelu_fwd(gx, gb), iters=10)
print(f"bias_gelu_fwd 32768x8192: {us:.1f} us ({gx.numel()*2*2/us/1e3:.0f} GB/s)")
us = t(lambda: ops.bias_gelu_bwd(gd, gx, gb), iters=10)
print(f"bias_gelu_bwd 32768x8192: {us:.1f} us ({gx.numel()*2*3/us/1e3:.0f} GB/s)")
# bias_add_residual
xa = torch.randn(2048, 2, 2048, device="cuda", dtype=torch.bfloat16)
ra = torch.randn_like(xa); ba = torch.randn(2048, device="cuda", dtype=torch.bfloat16)
us = t(lambda: ops.bias_add_residual(xa, ba, ra))
print(f"bias_add_residual: {us:.1f} us ({xa.numel()*2*3/us/1e3:.0f} GB/s)")
us = t(lambda: (xa + ba + ra))
print(f"eager x+b+r:       {us:.1f} us")

# hot fc GEMM shapes: torch.matmul baseline (its own hipBLASLt heuristics)
for (M, K, N) in ((32768, 2048, 8192), (32768, 8192, 2048),
                  (32768, 2048, 6144)):
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    us = t(lambda: torch.matmul(a, w.t()), iters=20)
    fl = 2 * M * K * N / us / 1e6
    print(f"torch.mm {M}x{K}x{N}: {us:.1f} us ({fl:.0f} TFLOP/s... unit: 1e12/s)")
    ref = torch.matmul(a.float(), w.t().float())
    o2 = ops.gemm_nt(a, w)
    err = (o2.float() - ref).abs().max() / ref.abs().max()
    us = t(lambda: ops.gemm_nt(a, w), iters=20)
    fl = 2 * M * K * N / us / 1e6
    print(f"gemm_nt  {M}x{K}x{N}: {us:.1f} us ({fl:.0f}) relerr={float(err):.4f}")
    g = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    us = t(lambda: torch.matmul(g, w), iters=20)
    print(f"torch.nn {M}x{N}x{K}: {us:.1f} us")
    refnn = torch.matmul(g.float(), w.float())
    o3 = ops.gemm_nn(g, w)
    errnn = (o3.float() - refnn).abs().max() / refnn.abs().max()
    us = t(lambda: ops.gemm_nn(g, w), iters=20)
    print(f"gemm_nn  {M}x{N}x{K}: {us:.1f} us relerr={float(errnn):.4f}")
