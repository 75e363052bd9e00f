"""MegaDPP tests: shm channel, send-ordering policy, end-to-end pipeline."""

import os
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _ensure_built():
    so = os.path.join(REPO, "megatronapp_amd", "dpp", "_C_dpp.so")
    if not os.path.exists(so):
        from megatronapp_amd.dpp.transport import build_dpp_extension
        build_dpp_extension()


def test_send_queue_policies():
    from megatronapp_amd.dpp.transport import _SendQueue
    items = [(1, 0), (0, 2), (0, 1), (1, 1)]  # (chunk, mb)

    q = _SendQueue("depth_first")
    for c, m in items:
        q.push((c, m, "fwd", 0, 1, None))
    order = [(q.pop()[0], q.pop.__self__ and None) for _ in range(0)]
    got = [tuple(q.pop()[:2]) for _ in range(4)]
    assert got == [(0, 1), (0, 2), (1, 0), (1, 1)]  # chunk-major

    q = _SendQueue("breadth_first")
    for c, m in items:
        q.push((c, m, "fwd", 0, 1, None))
    got = [tuple(q.pop()[:2]) for _ in range(4)]
    assert got == [(1, 0), (0, 1), (1, 1), (0, 2)]  # microbatch-major

    q = _SendQueue("greedy")
    for c, m in items:
        q.push((c, m, "fwd", 0, 1, None))
    got = [tuple(q.pop()[:2]) for _ in range(4)]
    assert got == items  # FIFO


def _channel_proc(rank):
    """Two processes exchange tagged tensors out of order."""
    _ensure_built()
    sys.path.insert(0, REPO)
    from megatronapp_amd.dpp.transport import _load
    c = _load()

    slot = 1024 * 4
    if rank == 0:
        c.init_channel("fwd", 0, 1, slot, 4, False)  # sender opens
        # send OUT OF ORDER: mb 2, then 0, then 1
        for mb in (2, 0, 1):
            t = torch.full((1024,), float(mb), dtype=torch.float32)
            c.put_tensor("fwd", 0, 1, 0, mb, t)
    else:
        c.init_channel("fwd", 0, 1, slot, 4, True)   # receiver creates
        # receive IN ORDER despite the sender's order
        for mb in (0, 1, 2):
            out = torch.empty(1024, dtype=torch.float32)
            c.get_tensor("fwd", 0, 1, 0, mb, out)
            assert torch.all(out == float(mb)), (mb, out[:4])
        c.clean_channels()


def test_shm_channel_out_of_order_exchange():
    _ensure_built()
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    recv = ctx.Process(target=_channel_proc, args=(1,))
    send = ctx.Process(target=_channel_proc, args=(0,))
    recv.start()
    send.start()
    recv.join(timeout=120)
    send.join(timeout=120)
    assert recv.exitcode == 0 and send.exitcode == 0


ARGS = [
    "--num-layers", "4", "--hidden-size", "64", "--num-attention-heads", "4",
    "--seq-length", "32", "--micro-batch-size", "2", "--global-batch-size",
    "8", "--pipeline-model-parallel-size", "2", "--mock-data",
    "--train-iters", "4", "--lr", "1e-3", "--log-interval", "1",
    "--vocab-size", "128", "--eval-iters", "0", "--hidden-dropout", "0",
    "--attention-dropout", "0",
]


@pytest.mark.parametrize("extra", [["--use-dpp"],
                                   ["--use-dpp",
                                    "--num-layers-per-virtual-pipeline-stage",
                                    "1"]])
def test_pretrain_with_dpp(extra, tmp_path):
    _ensure_built()
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29661",
         os.path.join(REPO, "pretrain_gpt.py")] + ARGS + extra,
        capture_output=True, text=True, cwd=REPO, timeout=420)
    assert out.returncode == 0, out.stderr[-4000:]
    assert "lm loss" in out.stdout


def _channel_proc_cuda(rank):
    """Same exchange with CUDA tensors: exercises the hipMemcpy staging
    in/out of the shm slot on both sides."""
    _ensure_built()
    sys.path.insert(0, REPO)
    import torch
    from megatronapp_amd.dpp.transport import _load
    c = _load()
    torch.cuda.set_device(0)
    slot = 1024 * 4
    if rank == 0:
        c.init_channel("fwd", 10, 11, slot, 4, False)
        for mb in (2, 0, 1):
            t = torch.full((1024,), float(mb) + 0.5, dtype=torch.float32,
                           device="cuda")
            c.put_tensor("fwd", 10, 11, 0, mb, t)
        torch.cuda.synchronize()
    else:
        c.init_channel("fwd", 10, 11, slot, 4, True)
        for mb in (0, 1, 2):
            out = torch.empty(1024, dtype=torch.float32, device="cuda")
            c.get_tensor("fwd", 10, 11, 0, mb, out)
            torch.cuda.synchronize()
            assert torch.all(out == float(mb) + 0.5).item(), (mb, out[:4])
        c.clean_channels()


@pytest.mark.gpu
def test_shm_channel_cuda_tensors():
    _ensure_built()
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    recv = ctx.Process(target=_channel_proc_cuda, args=(1,))
    send = ctx.Process(target=_channel_proc_cuda, args=(0,))
    recv.start(); send.start()
    recv.join(timeout=180); send.join(timeout=180)
    assert recv.exitcode == 0 and send.exitcode == 0
