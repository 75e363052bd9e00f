#!/usr/bin/env python3
"""Minimal fwd2/bwd2 run for PMC counter collection."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from megatronapp_amd import ops

torch.cuda.set_device(0)
sq, b, nh, d = 2048, 16, 16, 128
scale = d ** -0.5
q = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
do = torch.randn_like(q)
O = ops.get_ops()
for _ in range(3):
    o, lse = O.attn_fwd2(q, k, v, scale, True)
    O.attn_bwd(do, q, k, v, o, lse, scale, True)
torch.cuda.synchronize()
print("pmc run done")
