#!/usr/bin/env python3
"""Merge several aggregated Chrome traces into one timeline (reference
profiling/merge_trace.py): each input gets its own process lane, offset so
runs are side by side.

  python profiling/merge_trace.py a.json b.json -o merged_trace.json
"""

import argparse
import json


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("traces", nargs="+")
    ap.add_argument("-o", "--output", default="merged_trace.json")
    args = ap.parse_args()
    merged = []
    for i, path in enumerate(args.traces):
        with open(path) as f:
            data = json.load(f)
        events = data["traceEvents"] if isinstance(data, dict) else data
        for ev in events:
            ev = dict(ev)
            ev["pid"] = f"run{i}:{ev.get('pid', 0)}"
            merged.append(ev)
    with open(args.output, "w") as f:
        json.dump({"traceEvents": merged}, f)
    print(f"wrote {args.output} ({len(merged)} events from "
          f"{len(args.traces)} traces)")


if __name__ == "__main__":
    main()
