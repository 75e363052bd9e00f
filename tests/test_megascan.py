"""MegaScan tracer + aggregator + detector tests (CPU)."""

import json
import os
import subprocess
import sys

import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _make_trace(trace_dir, ranks=2, iters=3, slow_rank=None):
    """Synthesize per-rank trace files in the wire format."""
    for r in range(ranks):
        records = []
        scale = 2.0 if r == slow_rank else 1.0
        for it in range(iters):
            info = dict(dp_rk=r, pp_rk=0, tp_rk=0, g_rk=r, dev=r)
            records.append({"name": "iteration", "ph": "B", "rel_ts": 0,
                            "iteration": it, "pad_before": 1000, **info})
            t = 0
            for name, dur, grp in (
                    ("forward", int(4_000_000 * scale), None),
                    ("backward", int(8_000_000 * scale), None),
                    ("loss", int(1_000_000 / scale), None),
                    ("allreduce", int(2_000_000 / scale), list(range(ranks)))):
                rec_b = {"name": name, "ph": "B", "rel_ts": t, **info}
                t += dur
                rec_e = {"name": name, "ph": "E", "rel_ts": t, **info}
                if grp:
                    rec_e["group"] = grp
                    rec_e["data"] = 1 << 20
                records.extend([rec_b, rec_e])
            records.append({"name": "iteration", "ph": "E", "rel_ts": t,
                            "iteration": it, "duration_wall": t + 500,
                            "duration_cuda": t, **info})
        fname = f"benchmark-data-{r}-pipeline-0-tensor-0.json"
        with open(os.path.join(trace_dir, fname), "w") as f:
            json.dump(records, f)


def test_tracer_produces_wire_format(tmp_path):
    """Tracer on CPU produces the benchmark-data file with B/E pairs."""
    from megatronapp_amd.training.trace import Tracer
    from .utils import initialize_model_parallel, destroy

    initialize_model_parallel()
    tracer = Tracer.initialize(trace_dir=str(tmp_path), interval=1,
                               continuous_iters=2)
    for it in range(2):
        tracer.iteration_begin(it)
        with tracer.scope("forward"):
            torch.randn(64, 64) @ torch.randn(64, 64)
        with tracer.scope("backward"):
            pass
        with tracer.scope("allreduce", data=4096, group=[0]):
            pass
        tracer.iteration_end()
    tracer.shutdown()

    path = tmp_path / "benchmark-data-0-pipeline-0-tensor-0.json"
    assert path.exists()
    records = json.load(open(path))
    names = [r["name"] for r in records]
    assert names.count("iteration") == 4  # 2 iters x B/E
    assert "forward" in names and "allreduce" in names
    fb = [r for r in records if r["name"] == "forward" and r["ph"] == "B"][0]
    assert {"rel_ts", "dp_rk", "pp_rk", "tp_rk", "g_rk"} <= set(fb)
    ar = [r for r in records if r["name"] == "allreduce" and r["ph"] == "E"][0]
    assert ar["group"] == [0] and ar["data"] == 4096
    destroy()


def test_windowed_activation(tmp_path):
    from megatronapp_amd.training.trace import Tracer
    tracer = Tracer(trace_dir=str(tmp_path), interval=5, continuous_iters=2)
    active = [tracer._window_active(i) for i in range(10)]
    assert active == [True, True, False, False, False,
                      True, True, False, False, False]


def test_aggregate_and_detect(tmp_path):
    _make_trace(str(tmp_path), ranks=4, iters=6, slow_rank=2)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "scripts", "aggregate.py"),
         "--trace-dir", str(tmp_path), "--detect"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    bench = json.load(open(tmp_path / "benchmark.json"))
    events = bench["traceEvents"]
    xs = [e for e in events if e.get("ph") == "X"]
    assert len(xs) == 4 * 6 * 5  # ranks x iters x (4 events + iteration)
    # color map applied
    assert any(e.get("cname") for e in xs)
    # collective linking
    ar = [e for e in xs if e["name"] == "allreduce"]
    assert all("related_sync_op" in e["args"] for e in ar)
    # detection: rank 2 was made slow (its loss/allreduce finish fastest —
    # they wait least — and its backward is slowest)
    report = open(tmp_path / "abnormal.txt").read()
    assert "rank 2" in report, report


def test_base_granularity_filters():
    from megatronapp_amd.training.trace import Tracer, BASE_TRACING_EVENTS
    tracer = Tracer(granularity="base")
    assert tracer._keep("forward")
    assert tracer._keep("allreduce")
    assert not tracer._keep("transformer_layer")
    assert not tracer._keep("_reduce")
