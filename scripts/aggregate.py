#!/usr/bin/env python3
"""MegaScan trace aggregation: per-rank JSON traces -> one Chrome trace
(+ slow-GPU detection).

Reference behaviours replicated (scripts/aggregate.py):
* collect ``benchmark-data-{dp}-pipeline-{pp}-tensor-{tp}.json`` files;
* split records into iterations on the iteration B/E markers;
* stitch a global timeline from per-iteration ``pad_before`` /
  ``duration_wall`` wall-clock info (no global clock needed);
* emit Chrome Trace Format: B/E pairs collapsed to "X" events with dur,
  pid = global rank, color by event name, process metadata;
* link collective events sharing (name, sorted group ranks) into
  ``related_sync_op`` id lists, and reconcile p2p send/recv pairs
  (dependency() / amendP2P() of reference scripts/dependency.py);
* slow-rank detection (try_detect): stage 1 compares each event class
  across DP peers — for wait-dominated events ("loss", "allreduce") the
  FASTEST outlier is the suspect (it waited least => everyone waited on
  it); for compute events ("backward", "forward") the SLOWEST outlier is;
  stage 2 confirms suspects within collective sync groups and writes
  ``abnormal.txt``.

Usage: python scripts/aggregate.py --trace-dir trace_output \
           [--output benchmark.json] [--detect]
"""

from __future__ import annotations

import argparse
import glob
import json
import os
import re
from collections import defaultdict

COLOR_MAP = {
    "forward": "good",
    "backward": "bad",
    "loss": "terrible",
    "optimizer": "yellow",
    "allreduce": "olive",
    "grad-sync": "olive",
    "grad-sync-dp": "olive",
    "grad-sync-embedding": "olive",
    "grad-sync-layernorm": "olive",
    "recv-forward": "thread_state_runnable",
    "recv-backward": "thread_state_iowait",
    "send-forward": "thread_state_running",
    "send-backward": "thread_state_unknown",
    "send-forward-recv-backward": "rail_response",
    "send-backward-recv-forward": "rail_animation",
    "exchange-next": "cq_build_running",
    "exchange-prev": "cq_build_passed",
    "attention": "rail_load",
    "mlp": "rail_idle",
    "decoder": "generic_work",
    "transformer_layer": "good",
}

WAIT_DOMINATED_EVENTS = {"loss", "allreduce", "grad-sync", "grad-sync-dp"}
COMPUTE_EVENTS = {"backward", "forward"}

FNAME_RE = re.compile(
    r"benchmark-data-(\d+)-pipeline-(\d+)-tensor-(\d+)\.json")


def collect_benchmark_files(trace_dir):
    out = []
    for path in sorted(glob.glob(os.path.join(
            trace_dir, "benchmark-data-*-pipeline-*-tensor-*.json"))):
        m = FNAME_RE.search(os.path.basename(path))
        if m:
            out.append((path, tuple(int(x) for x in m.groups())))
    return out


def read_benchmark_file(path):
    """Split a rank's record stream into iterations."""
    with open(path) as f:
        records = json.load(f)
    iterations = []
    current = None
    for rec in records:
        if rec.get("name") == "iteration" and rec.get("ph") == "B":
            current = {"begin": rec, "events": [], "end": None}
        elif rec.get("name") == "iteration" and rec.get("ph") == "E":
            if current is not None:
                current["end"] = rec
                iterations.append(current)
                current = None
        elif current is not None:
            current["events"].append(rec)
    return iterations


def aggregate_benchmark_data(per_rank_iterations):
    """Assign each iteration a global start offset per rank by stitching
    pad_before gaps + previous iteration durations (reference :142-239)."""
    all_events = []
    for (dp, pp, tp), iterations in per_rank_iterations.items():
        wall = 0
        for it in iterations:
            wall += it["begin"].get("pad_before", 0)
            base = wall
            g_rk = it["begin"].get("g_rk", 0)
            it_idx = it["begin"].get("iteration", 0)
            all_events.append(dict(it["begin"], abs_ts=base, iteration=it_idx))
            for ev in it["events"]:
                all_events.append(dict(ev, abs_ts=base + ev.get("rel_ts", 0),
                                       iteration=it_idx))
            end = it["end"] or {}
            dur = end.get("duration_wall", end.get("duration_cuda", 0))
            all_events.append(dict(end, abs_ts=base + dur, iteration=it_idx))
            wall += dur
    all_events.sort(key=lambda e: e["abs_ts"])
    return all_events


def benchmark_to_chrome_trace(events):
    """Collapse B/E into X events (Chrome Trace Format)."""
    trace = []
    open_stack = defaultdict(list)   # (g_rk, name) -> [begin event]
    idx = 0
    for ev in events:
        key = (ev.get("g_rk", 0), ev.get("name"))
        if ev.get("ph") == "B":
            open_stack[key].append(ev)
        elif ev.get("ph") == "E":
            if open_stack[key]:
                b = open_stack[key].pop()
                x = {
                    "name": ev["name"], "ph": "X",
                    "ts": b["abs_ts"] / 1000.0,       # ns -> us
                    "dur": max(ev["abs_ts"] - b["abs_ts"], 0) / 1000.0,
                    "pid": ev.get("g_rk", 0), "tid": 0,
                    "args": {k: v for k, v in ev.items()
                             if k in ("group", "data", "bandwidth",
                                      "iteration", "dp_rk", "pp_rk", "tp_rk")},
                    "id": idx,
                }
                idx += 1
                cname = COLOR_MAP.get(ev["name"])
                if cname:
                    x["cname"] = cname
                trace.append(x)
        elif ev.get("ph") == "i":
            trace.append({"name": ev["name"], "ph": "i",
                          "ts": ev["abs_ts"] / 1000.0,
                          "pid": ev.get("g_rk", 0), "tid": 0, "s": "t"})
    # process metadata
    ranks = sorted({e["pid"] for e in trace})
    for r in ranks:
        trace.append({"name": "process_name", "ph": "M", "pid": r,
                      "args": {"name": f"rank {r}"}})
        trace.append({"name": "process_sort_index", "ph": "M", "pid": r,
                      "args": {"sort_index": r}})
    return trace


def dependency(trace):
    """Link collective events sharing (name, sorted group ranks, iteration)
    into related_sync_op id lists (reference dependency.py:26-51)."""
    groups = defaultdict(list)
    for x in trace:
        if x.get("ph") != "X":
            continue
        grp = x.get("args", {}).get("group")
        if grp:
            key = (x["name"], tuple(sorted(grp)),
                   x.get("args", {}).get("iteration"))
            groups[key].append(x)
    for key, members in groups.items():
        # members from different pids that represent the same collective:
        # pair them in per-pid call order
        by_pid = defaultdict(list)
        for m in members:
            by_pid[m["pid"]].append(m)
        for lst in by_pid.values():
            lst.sort(key=lambda x: x["ts"])
        n_calls = min(len(v) for v in by_pid.values())
        for i in range(n_calls):
            related = [by_pid[p][i] for p in sorted(by_pid)]
            ids = " ".join(str(m["id"]) for m in related)
            for m in related:
                m["args"]["related_sync_op"] = ids
    return trace


def amendP2P(trace):
    """Truncate p2p send/recv pairs to the min duration and reconcile
    bandwidth (reference dependency.py:54-86)."""
    for x in trace:
        if x.get("ph") != "X":
            continue
        rel = x.get("args", {}).get("related_sync_op")
        if not rel or not x["name"].startswith(("send-", "recv-", "exchange-")):
            continue
        ids = [int(i) for i in rel.split()]
        peers = [t for t in trace if t.get("id") in ids]
        if len(peers) < 2:
            continue
        min_dur = min(p["dur"] for p in peers)
        for p in peers:
            if p["dur"] > min_dur:
                p["ts"] += p["dur"] - min_dur
                p["dur"] = min_dur
            data = p.get("args", {}).get("data")
            if data and min_dur > 0:
                p["args"]["bandwidth"] = data * 8.0 / (min_dur * 1000.0)
    return trace


# --------------------------------------------------------------------------
# Slow-rank detection (reference try_detect :399-489)
# --------------------------------------------------------------------------

def try_detect(trace, fast_threshold=0.9, slow_threshold=1.1,
               suspect_limit=5):
    """Stage 1: per (iteration, event-class, pp, tp) compare across DP
    peers.  Returns suspect counts per global rank."""
    suspects = defaultdict(int)
    buckets = defaultdict(list)
    for x in trace:
        if x.get("ph") != "X":
            continue
        name = x["name"]
        cls = None
        if name in WAIT_DOMINATED_EVENTS or name == "_reduce":
            cls = "wait"
        elif name in COMPUTE_EVENTS:
            cls = "compute"
        if cls is None:
            continue
        a = x.get("args", {})
        key = (a.get("iteration"), name, a.get("pp_rk"), a.get("tp_rk"), cls)
        buckets[key].append(x)
    for key, members in buckets.items():
        cls = key[-1]
        by_rank = defaultdict(float)
        for m in members:
            by_rank[m["pid"]] += m["dur"]
        if len(by_rank) < 2:
            continue
        mean = sum(by_rank.values()) / len(by_rank)
        if mean <= 0:
            continue
        for rank, dur in by_rank.items():
            if cls == "wait" and dur < fast_threshold * mean:
                # it waited least => the others waited on something; the
                # fastest waiter is on the slow critical path
                suspects[rank] += 1
            elif cls == "compute" and dur > slow_threshold * mean:
                suspects[rank] += 1
    return dict(suspects)


def detect_in_data_parallelism_group(trace, suspects, ratio=0.4):
    """Stage 2: for suspect ranks, confirm within collective sync groups:
    a rank slower than its peers in >ratio of its '_reduce'/allreduce sync
    groups is flagged Abnormal (reference :366-396)."""
    confirmed = []
    for rank, hits in suspects.items():
        if hits <= 5:
            continue
        slow_cnt, all_cnt = 0, 0
        for x in trace:
            if x.get("ph") != "X" or x["pid"] != rank:
                continue
            if x["name"] not in ("allreduce", "_reduce", "grad-sync-dp"):
                continue
            rel = x.get("args", {}).get("related_sync_op")
            if not rel:
                continue
            ids = [int(i) for i in rel.split()]
            peers = [t for t in trace if t.get("id") in ids and t["pid"] != rank]
            if not peers:
                continue
            all_cnt += 1
            mean_peer = sum(p["dur"] for p in peers) / len(peers)
            if x["dur"] < 0.9 * mean_peer:
                slow_cnt += 1
        if all_cnt > 0 and slow_cnt > ratio * all_cnt:
            confirmed.append(rank)
    return confirmed


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trace-dir", default="trace_output")
    ap.add_argument("--output", default=None)
    ap.add_argument("--detect", action="store_true")
    args = ap.parse_args()

    files = collect_benchmark_files(args.trace_dir)
    if not files:
        print(f"no benchmark-data files in {args.trace_dir}")
        return 1
    per_rank = {}
    for path, key in files:
        per_rank[key] = read_benchmark_file(path)
    events = aggregate_benchmark_data(per_rank)
    trace = benchmark_to_chrome_trace(events)
    trace = dependency(trace)
    trace = amendP2P(trace)

    out = args.output or os.path.join(args.trace_dir, "benchmark.json")
    with open(out, "w") as f:
        json.dump({"traceEvents": trace, "displayTimeUnit": "ms"}, f)
    print(f"wrote {out} ({len(trace)} events from {len(files)} ranks)")

    if args.detect:
        suspects = try_detect(trace)
        confirmed = detect_in_data_parallelism_group(trace, suspects)
        report = os.path.join(args.trace_dir, "abnormal.txt")
        with open(report, "w") as f:
            if confirmed:
                for r in confirmed:
                    f.write(f"Abnormal GPU: global rank {r} "
                            f"(suspect hits: {suspects.get(r)})\n")
            else:
                f.write("No abnormal GPU detected.\n")
            f.write(f"suspect counts: {json.dumps(suspects)}\n")
        print(f"wrote {report}; suspects={suspects} confirmed={confirmed}")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
