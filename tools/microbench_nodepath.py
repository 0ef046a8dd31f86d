#!/usr/bin/env python3
"""Micro-benchmark the attach-path node operations on a real MI355X.

Times each operation the refresh/probe phases are built from, so
optimization targets come from measurement, not guesses:

    python tools/microbench_nodepath.py [--iters 50]
"""

import argparse
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def bench(fn, iters):
    xs = []
    for _ in range(iters):
        t0 = time.perf_counter()
        fn()
        xs.append((time.perf_counter() - t0) * 1e3)
    return {
        "p50_ms": round(statistics.median(xs), 4),
        "mean_ms": round(statistics.fmean(xs), 4),
        "max_ms": round(max(xs), 4),
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=50)
    args = p.parse_args()

    from cro_amd.controllers import build_manager
    from cro_amd.fabric.adapter import Adapter
    from cro_amd.fabric.mock import MockFabric
    from cro_amd.nodeops.amdgpu import AmdNodeOps
    from cro_amd.nodeops.execs import LocalNodeExec
    from cro_amd.nodeops.kfd import enumerate_gpus
    from cro_amd.nodeops.probe import probe_fn_for_nodeops, run_probe

    node = "ubench-node"
    execer = LocalNodeExec()
    mgr = build_manager(Adapter("DRA", MockFabric()), None)
    ops = AmdNodeOps(execer, client=mgr.client, cdi_dir="/tmp/ubench-cdi",
                     destructive=False, probe_fn=probe_fn_for_nodeops)
    gpus = ops.enumerate(node)
    assert gpus, "no GPUs"
    gpu = gpus[0]

    out = {}
    out["kfd_enumerate_cold"] = bench(
        lambda: (ops._invalidate_enum(node), ops.enumerate(node)), args.iters)
    out["kfd_enumerate_cached"] = bench(lambda: ops.enumerate(node), args.iters)
    out["kfd_raw_scan"] = bench(lambda: enumerate_gpus(execer, node), args.iters)
    out["publish_slice"] = bench(lambda: ops._publish_slice(node), args.iters)
    out["write_cdi"] = bench(lambda: ops.write_cdi(node, gpu.device_id), args.iters)
    out["is_visible"] = bench(lambda: ops.is_visible(node, gpu.device_id), args.iters)
    run_probe(0)  # first-touch (context + code object)
    out["probe_steady"] = bench(lambda: run_probe(0), max(args.iters // 5, 5))
    out["refresh_after_attach"] = bench(
        lambda: ops.refresh_after_attach(node), args.iters)
    out["driver_gate"] = bench(lambda: ops.ensure_driver(node), args.iters)

    print(json.dumps(out, indent=1))
    return 0


if __name__ == "__main__":
    sys.exit(main())
