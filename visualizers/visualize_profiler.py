#!/usr/bin/env python3
"""Render a tnn_amd Profiler dump (reference visualizers/
visualize_profiler.py Gantt analog).

Input: a chrome-trace JSON written by Profiler.export_chrome_trace (open
it in chrome://tracing / Perfetto for the interactive view); this script
prints a terminal Gantt + per-name totals for quick looks.

    python visualizers/visualize_profiler.py trace.json [--width 100]
"""

import argparse
import collections
import json


def main():
    p = argparse.ArgumentParser()
    p.add_argument("trace")
    p.add_argument("--width", type=int, default=100)
    p.add_argument("--rows", type=int, default=40)
    args = p.parse_args()

    with open(args.trace) as f:
        events = json.load(f)["traceEvents"]
    if not events:
        print("no events")
        return
    t0 = min(e["ts"] for e in events)
    t1 = max(e["ts"] + e["dur"] for e in events)
    span = max(t1 - t0, 1e-9)
    scale = args.width / span

    print(f"span {span / 1e3:.2f} ms, {len(events)} events")
    lanes = collections.defaultdict(list)
    for e in events:
        lanes[(str(e.get('pid', '')), str(e.get('tid', '')))].append(e)
    for (pid, tid), evs in sorted(lanes.items()):
        print(f"-- {pid}/{tid} --")
        for e in sorted(evs, key=lambda x: x["ts"])[:args.rows]:
            start = int((e["ts"] - t0) * scale)
            width = max(1, int(e["dur"] * scale))
            bar = " " * start + "#" * min(width, args.width - start)
            print(f"{bar:<{args.width}} {e['name'][:40]} "
                  f"({e['dur'] / 1e3:.2f} ms)")

    totals = collections.defaultdict(float)
    for e in events:
        totals[e["name"]] += e["dur"]
    print("\nper-name totals:")
    for name, dur in sorted(totals.items(), key=lambda kv: -kv[1])[:20]:
        print(f"  {dur / 1e3:10.2f} ms  {name}")


if __name__ == "__main__":
    main()
