#!/usr/bin/env python3
"""MegaScan trace post-processing (reference profiling/process_*.py suite
collapsed into subcommands).

  python profiling/process_trace.py computing-ratio --trace-dir trace_out
  python profiling/process_trace.py sending-ratio   --trace-dir trace_out
  python profiling/process_trace.py iter-times      --trace-dir trace_out
  python profiling/process_trace.py windows         --trace-dir trace_out

computing-ratio: per rank, fraction of each iteration spent inside compute
scopes (forward/backward) vs communication/wait.
sending-ratio:   fraction spent in p2p send/recv scopes.
iter-times:      per-iteration wall time per rank.
windows:         busy windows (start, duration) per rank per iteration for
                 plotting pipeline occupancy.
"""

import argparse
import json
import os
import re
import sys
from collections import defaultdict

COMPUTE = re.compile(r"forward|backward|optimizer|transformer_layer")
SEND = re.compile(r"send|recv|p2p|pipeline")


def load(trace_dir):
    ranks = {}
    for f in sorted(os.listdir(trace_dir)):
        m = re.match(r"benchmark-data-(\d+)-pipeline-(\d+)-tensor-(\d+)\.json",
                     f)
        if not m:
            continue
        with open(os.path.join(trace_dir, f)) as fh:
            iters = []
            for line in fh:
                if not line.strip():
                    continue
                # a line holds every traced iteration's events; split on
                # the top-level "iteration" B records
                cur = None
                for ev in json.loads(line):
                    if ev.get("name") == "iteration" and ev.get("ph") == "B":
                        cur = [ev]
                        iters.append(cur)
                    elif cur is not None:
                        cur.append(ev)
        ranks[(int(m.group(1)), int(m.group(2)), int(m.group(3)))] = iters
    if not ranks:
        sys.exit(f"no benchmark-data files in {trace_dir}")
    return ranks


def _events(iteration):
    # a MegaScan line is a flat list of B/E records for one iteration
    if isinstance(iteration, list):
        return iteration
    return iteration.get("traceEvents", iteration.get("events", []))


def _iter_no(iteration):
    for ev in _events(iteration):
        if "iteration" in ev:
            return ev["iteration"]
    return "?"


def spans(iteration, pattern):
    """Top-level [start, end] spans of events matching pattern."""
    out = []
    stack = 0
    for ev in _events(iteration):
        if not pattern.search(ev.get("name", "")):
            continue
        if ev.get("ph") == "B":
            if stack == 0:
                start = ev["rel_ts"]
            stack += 1
        elif ev.get("ph") == "E":
            stack -= 1
            if stack == 0:
                out.append((start, ev["rel_ts"]))
    return out


def iter_span(iteration):
    ts = [e["rel_ts"] for e in _events(iteration) if "rel_ts" in e]
    return (min(ts), max(ts)) if ts else (0, 0)


def ratio_report(ranks, pattern, label):
    print(f"{'rank (dp,pp,tp)':>16s} {'iter':>5s} {label + ' ms':>12s} "
          f"{'iter ms':>9s} {'ratio':>7s}")
    for key, iters in sorted(ranks.items()):
        for it in iters:
            n = _iter_no(it)
            lo, hi = iter_span(it)
            total = sum(b - a for a, b in spans(it, pattern))
            span = max(hi - lo, 1e-9)
            print(f"{str(key):>16s} {n!s:>5s} {total / 1e6:12.2f} "
                  f"{span / 1e6:9.2f} {total / span:7.2%}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("mode", choices=["computing-ratio", "sending-ratio",
                                     "iter-times", "windows"])
    ap.add_argument("--trace-dir", required=True)
    args = ap.parse_args()
    ranks = load(args.trace_dir)

    if args.mode == "computing-ratio":
        ratio_report(ranks, COMPUTE, "compute")
    elif args.mode == "sending-ratio":
        ratio_report(ranks, SEND, "send/recv")
    elif args.mode == "iter-times":
        for key, iters in sorted(ranks.items()):
            for it in iters:
                lo, hi = iter_span(it)
                print(f"{key} iter {_iter_no(it)}: {(hi - lo) / 1e6:.2f} ms")
    else:  # windows
        out = defaultdict(list)
        for key, iters in sorted(ranks.items()):
            for it in iters:
                for a, b in spans(it, COMPUTE):
                    out[str(key)].append(
                        {"iter": _iter_no(it), "start": a,
                         "duration": b - a})
        print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
