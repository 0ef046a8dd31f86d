#!/usr/bin/env python3
"""Production-shape service soak: the standalone operator entrypoint
(real KFD node path when a GPU is present, MOCK fabric bound to the real
inventory, fail-closed bearer auth) driven through the REST API for
``--minutes`` of continuous attach→Running→delete cycles.

    python tools/service_soak.py --minutes 20 [--port P]

Prints one JSON summary line; exit 0 iff zero failed cycles and zero
operator error lines.
"""

import argparse
import json
import os
import signal
import socket
import statistics
import subprocess
import sys
import time

import httpx

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BASE = "/apis/cro.hpsys.ibm.ie.com/v1alpha1"
TOKEN = "soak-token"


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--minutes", type=float, default=20.0)
    p.add_argument("--port", type=int, default=0)
    p.add_argument("--size", type=int, default=1)
    args = p.parse_args()

    port = args.port or free_port()
    node = socket.gethostname()
    env = dict(os.environ)
    env.update({
        "DEVICE_RESOURCE_TYPE": "DRA",
        "CDI_PROVIDER_TYPE": "MOCK",
        "CRO_API_TOKEN": TOKEN,
        "CRO_AGENT_TOKEN": TOKEN,
        "CRO_METRICS_TOKEN": TOKEN,
    })
    argv = [sys.executable, "-m", "cro_amd.cmd.main",
            "--api-bind-address", f":{port}",
            "--metrics-bind-address", f":{free_port()}",
            "--health-probe-bind-address", f":{free_port()}",
            "--node", node,
            "--cdi-dir", os.path.join(os.environ.get("TMPDIR", "/tmp"), "cro-soak-cdi")]
    if not os.path.exists("/dev/kfd"):
        argv.append("--simulate-node-path")  # GPU-less dry runs
    import tempfile

    # file-backed stdout: a PIPE nobody drains blocks the child once the
    # 64 KB buffer fills under steady logging (found by split_soak.py)
    logf = tempfile.NamedTemporaryFile(
        mode="w+", prefix="service-soak-", suffix=".log", delete=False)
    proc = subprocess.Popen(
        argv,
        cwd=REPO, env=env,
        stdout=logf, stderr=subprocess.STDOUT, text=True,
    )
    http = httpx.Client(
        base_url=f"http://127.0.0.1:{port}", timeout=10,
        headers={"Authorization": f"Bearer {TOKEN}"},
    )
    deadline = time.monotonic() + 45
    while time.monotonic() < deadline:
        try:
            if http.get("/healthz").status_code == 200:
                break
        except Exception:
            time.sleep(0.3)
    else:
        print(json.dumps({"ok": False, "error": "operator never came up"}))
        proc.kill()
        return 1

    body = {
        "apiVersion": "cro.hpsys.ibm.ie.com/v1alpha1",
        "kind": "ComposabilityRequest",
        "metadata": {"name": "soak"},
        "spec": {"resource": {"type": "gpu", "model": "mi355x",
                              "size": args.size, "target_node": node,
                              "force_detach": True}},
    }

    cycles = failed = 0
    attach_ms = []
    t_end = time.monotonic() + args.minutes * 60
    try:
        while time.monotonic() < t_end:
            t0 = time.monotonic()
            r = http.post(f"{BASE}/composabilityrequests", json=body)
            if r.status_code != 201:
                failed += 1
                time.sleep(0.2)
                continue
            ok = False
            cyc_deadline = time.monotonic() + 60
            while time.monotonic() < cyc_deadline:
                g = http.get(f"{BASE}/composabilityrequests/soak")
                if g.status_code == 200 and g.json()["status"]["state"] == "Running":
                    ok = True
                    break
                time.sleep(0.002)
            t1 = time.monotonic()
            http.delete(f"{BASE}/composabilityrequests/soak")
            gone = False
            cyc_deadline = time.monotonic() + 60
            while time.monotonic() < cyc_deadline:
                if http.get(f"{BASE}/composabilityrequests/soak").status_code == 404:
                    gone = True
                    break
                time.sleep(0.002)
            cycles += 1
            if ok and gone:
                attach_ms.append((t1 - t0) * 1e3)
            else:
                failed += 1
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=20)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait()
        logf.flush()
        logf.seek(0)
        out = logf.read()

    err_lines = [l for l in out.splitlines()
                 if "ERROR" in l or "Traceback" in l]
    result = {
        "ok": failed == 0 and not err_lines,
        "minutes": args.minutes,
        "cycles": cycles,
        "failed": failed,
        "attach_p50_ms": round(statistics.median(attach_ms), 3) if attach_ms else None,
        "attach_mean_ms": round(statistics.fmean(attach_ms), 3) if attach_ms else None,
        "attach_p99_ms": round(sorted(attach_ms)[max(int(len(attach_ms) * 0.99) - 1, 0)], 3) if attach_ms else None,
        "cycle_rate_per_sec": round(cycles / (args.minutes * 60), 2),
        "operator_error_lines": err_lines[:5],
        "node_path": "real KFD/CDI/probe" if os.path.exists("/dev/kfd") else "mock",
    }
    print(json.dumps(result))
    return 0 if result["ok"] else 1


if __name__ == "__main__":
    sys.exit(main())
