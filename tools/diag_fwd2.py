#!/usr/bin/env python3
"""Diagnose attn_fwd2 numerics: small shapes, structured inputs, print
error patterns (which q rows / d cols are wrong) to localize the bug."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from megatronapp_amd import ops


def ref(q, k, v, scale, causal):
    sq, b, nh, d = q.shape
    s = torch.einsum("qbhd,kbhd->bhqk", q.float(), k.float()) * scale
    if causal:
        sk = k.shape[0]
        mask = torch.triu(torch.ones(sq, sk, dtype=torch.bool,
                                     device=q.device), 1 + (sk - sq))
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return torch.einsum("bhqk,kbhd->qbhd", p, v.float())


def run(sq, b, nh, causal, mode):
    d = 128
    torch.manual_seed(3)
    if mode == "randn":
        q = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
        k = torch.randn_like(q)
        v = torch.randn_like(q)
    elif mode == "v_iota":
        # softmax-neutral scores, V identifies d columns
        q = torch.zeros(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
        k = torch.zeros_like(q)
        v = torch.arange(d, device="cuda", dtype=torch.bfloat16)
        v = v.expand(sq, b, nh, d).contiguous()
    elif mode == "v_kv":
        # V identifies kv rows: V[kv, :, :, :] = kv/64
        q = torch.zeros(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
        k = torch.zeros_like(q)
        v = (torch.arange(sq, device="cuda", dtype=torch.float32) / 64.0)
        v = v.view(sq, 1, 1, 1).expand(sq, b, nh, d).to(torch.bfloat16)
        v = v.contiguous()
    scale = d ** -0.5
    o, lse = ops.get_ops().attn_fwd2(q, k, v, scale, causal)
    o_ref = ref(q, k, v, scale, causal)
    err = (o.float() - o_ref).abs()
    print(f"sq={sq} b={b} nh={nh} causal={causal} mode={mode}: "
          f"maxerr={err.max().item():.4f} nan={torch.isnan(o).sum().item()}")
    if err.max() > 0.05 or torch.isnan(o).any():
        bad = (err > 0.05) | torch.isnan(o.float())
        qrows = bad.any(dim=(1, 2, 3)).nonzero().flatten()
        dcols = bad.any(dim=(0, 1, 2)).nonzero().flatten()
        print("  bad q rows:", qrows[:20].tolist(), "..." , len(qrows))
        print("  bad d cols:", dcols[:20].tolist(), "...", len(dcols))
        i = qrows[0].item()
        print(f"  q{i} got   :", o[i, 0, 0, :8].float().tolist())
        print(f"  q{i} expect:", o_ref[i, 0, 0, :8].tolist())


def main():
    torch.cuda.set_device(0)
    for mode in ("v_iota", "v_kv", "randn"):
        run(256, 1, 1, False, mode)
    run(256, 1, 1, True, "randn")
    run(512, 1, 1, False, "randn")
    run(2048, 2, 4, True, "randn")


if __name__ == "__main__":
    main()
