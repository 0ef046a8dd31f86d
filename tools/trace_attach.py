#!/usr/bin/env python3
"""Trace one attach/detach lifecycle: every store event with a relative
timestamp, plus the attach-phase averages — the where-does-the-time-go
artifact behind profiles/attach_timeline.md.

Usage: python tools/trace_attach.py [--out FILE]
Runs on the real node path when a GPU is present, mock otherwise.
"""

from __future__ import annotations

import argparse
import json
import queue
import sys
import time

sys.path.insert(0, ".")

from cro_amd.bench_harness import attach_detach_cycle, build_local_stack  # noqa: E402


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="")
    args = p.parse_args()

    stack = build_local_stack(node_name="trace-node")
    events = stack.mgr.store.watch()
    stamped = []

    def pump():
        while True:
            ev = events.get()
            stamped.append((time.monotonic(), ev))

    import threading

    threading.Thread(target=pump, daemon=True).start()
    stack.mgr.start()
    attach_detach_cycle(stack, "warm")  # warm HIP ctx etc.
    time.sleep(0.2)
    stamped.clear()
    t0 = time.monotonic()
    timing = attach_detach_cycle(stack, "trace")
    time.sleep(0.3)
    stack.mgr.stop()

    lines = [
        f"# Attach/detach event timeline ({'real node path' if stack.gpu else 'mock node path'})",
        "",
        f"attach {timing['attach_ms']:.2f} ms, detach {timing['detach_ms']:.2f} ms",
        "",
        "| t (ms) | event | kind | object | state |",
        "|---|---|---|---|---|",
    ]
    for ts, ev in stamped:
        st = getattr(ev.object, "status", None)
        lines.append(
            f"| {(ts - t0) * 1e3:8.2f} | {ev.type} | {ev.object.kind} | "
            f"{ev.object.metadata.name[:24]} | {getattr(st, 'state', '')} |"
        )

    phases = {}
    for metric in stack.mgr.metrics.attach_phase_seconds.collect():
        for s in metric.samples:
            if s.name.endswith("_sum"):
                phases.setdefault(s.labels["phase"], {})["sum"] = s.value
            elif s.name.endswith("_count"):
                phases.setdefault(s.labels["phase"], {})["count"] = s.value
    lines += ["", "## Attach-phase averages (all cycles this process)", ""]
    for name, v in sorted(phases.items()):
        if v.get("count"):
            lines.append(f"- {name}: {v['sum'] * 1e3 / v['count']:.3f} ms avg over {int(v['count'])}")

    text = "\n".join(lines) + "\n"
    if args.out:
        with open(args.out, "w") as f:
            f.write(text)
    print(text)
    return 0


if __name__ == "__main__":
    sys.exit(main())
