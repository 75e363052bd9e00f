"""Driver-contract guard for bench.py.

The round-end harness runs `python bench.py --gpus N ...` (N>1 via
torch.distributed.run) and parses ONE JSON line from rank 0.  These
tests pin that contract on CPU/gloo so a refactor cannot silently break
the scaling benchmark: required keys, aggregate-over-ranks metric, and
a clean multi-rank (dp and pp) launch.
"""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


def _clean_env():
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT", "LOCAL_WORLD_SIZE", "GROUP_RANK"):
        env.pop(k, None)
    return env


def _json_line(out: str) -> dict:
    for line in out.splitlines():
        line = line.strip()
        if line.startswith("{") and '"metric"' in line:
            return json.loads(line)
    raise AssertionError(f"no bench JSON line in output:\n{out[-2000:]}")


def test_bench_single_rank_contract():
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--model", "gpt-tiny", "--seq-length", "64",
         "--global-batch-size", "4"],
        cwd=REPO, env=_clean_env(), capture_output=True, text=True,
        timeout=420)
    assert r.returncode == 0, r.stderr[-2000:]
    j = _json_line(r.stdout)
    for k in REQUIRED:
        assert k in j, f"missing key {k}"
    assert j["n_gpus"] == 1 and j["steps"] == 2 and j["warmup"] == 1
    assert j["value"] > 0 and j["ms_per_step"] > 0
    assert j["config"]["model"] == "gpt-tiny"


@pytest.mark.parametrize("nproc,par", [(2, "dp2"), (4, "pp4")])
def test_bench_multi_rank_contract(nproc, par):
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
         "--master-port", str(29490 + nproc), "bench.py",
         "--gpus", str(nproc), "--steps", "2", "--warmup", "1",
         "--model", "gpt-tiny", "--seq-length", "64",
         "--global-batch-size", "8"],
        cwd=REPO, env=_clean_env(), capture_output=True, text=True,
        timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    j = _json_line(r.stdout)
    assert j["n_gpus"] == nproc
    assert j["config"]["parallelism"] == par
    assert j["value"] > 0
