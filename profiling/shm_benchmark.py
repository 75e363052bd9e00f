#!/usr/bin/env python3
"""MegaDPP shm-channel bandwidth microbenchmark (reference
profiling/shm_benchmark.cpp + shm_benchmark_test.py).

Spawns a sender and receiver process and measures put/get throughput of
the POSIX-shm tagged mailbox for a sweep of tensor sizes; with a GPU
present the tensors live on device so the hipMemcpy staging is included.

  python profiling/shm_benchmark.py [--sizes-mb 1 4 16 64]
"""

import argparse
import multiprocessing as mp
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def _proc(rank, sizes_mb, iters, q, use_device=True):
    import torch
    from megatronapp_amd.dpp.transport import _load
    c = _load()
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    if dev == "cuda":
        torch.cuda.set_device(0)
    base = 30 if use_device else 70
    for idx, size_mb in enumerate(sizes_mb):
        n = size_mb * 1024 * 1024 // 4
        slot = n * 4
        src, dst = base + 2 * idx, base + 1 + 2 * idx  # one channel per size
        if rank == 0:
            c.init_channel("fwd", src, dst, slot, 2, False, use_device)
            t = torch.ones(n, dtype=torch.float32, device=dev)
            t0 = time.perf_counter()
            for i in range(iters):
                c.put_tensor("fwd", src, dst, 0, i, t)
            dt = time.perf_counter() - t0
            q.put((size_mb, "put", size_mb * iters / dt))
        else:
            c.init_channel("fwd", src, dst, slot, 2, True, use_device)
            out = torch.empty(n, dtype=torch.float32, device=dev)
            t0 = time.perf_counter()
            for i in range(iters):
                c.get_tensor("fwd", src, dst, 0, i, out)
            dt = time.perf_counter() - t0
            q.put((size_mb, "get", size_mb * iters / dt))
    if rank == 1:
        c.clean_channels()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sizes-mb", nargs="+", type=int, default=[1, 4, 16, 64])
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    from megatronapp_amd.dpp.transport import build_dpp_extension
    build_dpp_extension()  # build once here; children only load
    ctx = mp.get_context("spawn")
    import torch
    modes = [(True, "device (hipIpc D2D payload)"),
             (False, "host (D2H/shm/H2D staging)")] \
        if torch.cuda.is_available() else [(False, "host shm")]
    for use_device, label in modes:
        q = ctx.Queue()
        procs = [ctx.Process(target=_proc,
                             args=(r, args.sizes_mb, args.iters, q,
                                   use_device))
                 for r in (1, 0)]
        for p in procs:
            p.start()
        results = [q.get(timeout=300)
                   for _ in range(2 * len(args.sizes_mb))]
        for p in procs:
            p.join(timeout=60)
        print(f"-- {label} --")
        for size_mb, op, mbps in sorted(results):
            print(f"{size_mb:4d} MiB {op}: {mbps / 1024:.2f} GiB/s")


if __name__ == "__main__":
    main()
