#!/usr/bin/env python3
"""Split-topology soak: serve-only API server process + a SEPARATE
operator process connected over HTTP (RemoteClient informer protocol,
bearer auth), real node path when a GPU is present; continuous lifecycle
cycles driven against the apiserver for ``--minutes``.

    python tools/split_soak.py --minutes 10

Prints one JSON summary; exit 0 iff zero failed cycles and zero error
lines in either process.
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
TOKEN = "split-soak-token"


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--minutes", type=float, default=10.0)
    args = p.parse_args()

    node = socket.gethostname()
    env = dict(os.environ)
    env.update({
        "DEVICE_RESOURCE_TYPE": "DRA",
        "CDI_PROVIDER_TYPE": "MOCK",
        "CRO_API_TOKEN": TOKEN,
        "CRO_AGENT_TOKEN": TOKEN,
        "CRO_METRICS_TOKEN": TOKEN,
    })
    api_port = free_port()
    procs = []

    logs = []

    def spawn(*extra):
        # file-backed stdout: a PIPE nobody drains blocks the child once
        # the 64 KB buffer fills (the operator logs steadily) — this very
        # tool found that failure mode
        import tempfile

        logf = tempfile.NamedTemporaryFile(
            mode="w+", prefix="split-soak-", suffix=".log", delete=False)
        logs.append(logf)
        proc = subprocess.Popen(
            [sys.executable, "-m", "cro_amd.cmd.main",
             "--metrics-bind-address", f":{free_port()}",
             "--health-probe-bind-address", f":{free_port()}",
             *extra],
            cwd=REPO, env=env,
            stdout=logf, stderr=subprocess.STDOUT, text=True,
        )
        procs.append(proc)
        return proc

    apiserver = spawn("--api-bind-address", f":{api_port}", "--serve-only")
    http = httpx.Client(
        base_url=f"http://127.0.0.1:{api_port}", timeout=10,
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
        print(json.dumps({"ok": False, "error": "apiserver never came up"}))
        apiserver.kill()
        return 1

    gpu = os.path.exists("/dev/kfd")
    operator_args = ["--api-server", f"http://127.0.0.1:{api_port}",
                     "--api-bind-address", f":{free_port()}",
                     "--node", node,
                     "--cdi-dir", os.path.join(
                         os.environ.get("TMPDIR", "/tmp"), "cro-split-cdi")]
    if not gpu:
        operator_args.append("--simulate-node-path")
    operator = spawn(*operator_args)

    # remote mode: the operator does not register its node — do it here
    r = http.post(f"{BASE}/nodes", json={
        "apiVersion": "v1", "kind": "Node", "metadata": {"name": node},
        "status": {"capacity": {"milli_cpu": 128000, "memory": 2 << 40,
                                "allowed_pod_number": 256}},
    })
    assert r.status_code == 201, r.text

    body = {
        "apiVersion": "cro.hpsys.ibm.ie.com/v1alpha1",
        "kind": "ComposabilityRequest",
        "metadata": {"name": "split-soak"},
        "spec": {"resource": {"type": "gpu", "model": "mi355x", "size": 1,
                              "target_node": node, "force_detach": True}},
    }

    cycles = failed = 0
    attach_ms = []
    t_end = time.monotonic() + args.minutes * 60
    try:
        while time.monotonic() < t_end:
            t0 = time.monotonic()
            if http.post(f"{BASE}/composabilityrequests", json=body).status_code != 201:
                failed += 1
                time.sleep(0.2)
                continue
            ok = gone = False
            cyc_deadline = time.monotonic() + 90
            while time.monotonic() < cyc_deadline:
                g = http.get(f"{BASE}/composabilityrequests/split-soak")
                if g.status_code == 200 and g.json()["status"]["state"] == "Running":
                    ok = True
                    break
                time.sleep(0.005)
            t1 = time.monotonic()
            http.delete(f"{BASE}/composabilityrequests/split-soak")
            cyc_deadline = time.monotonic() + 90
            while time.monotonic() < cyc_deadline:
                if http.get(f"{BASE}/composabilityrequests/split-soak").status_code == 404:
                    gone = True
                    break
                time.sleep(0.005)
            cycles += 1
            if ok and gone:
                attach_ms.append((t1 - t0) * 1e3)
            else:
                failed += 1
    finally:
        outs = []
        # operator first, apiserver second: the operator's recorder/
        # controllers drain against a live API instead of logging
        # connection-refused noise
        for proc in reversed(procs):
            proc.send_signal(signal.SIGTERM)
            try:
                proc.wait(timeout=20)
            except subprocess.TimeoutExpired:
                proc.kill()
                proc.wait()
        for logf in logs:
            logf.flush()
            logf.seek(0)
            outs.append(logf.read())

    if os.environ.get("CRO_SOAK_DEBUG"):
        for i, out in enumerate(outs):
            with open(f"/tmp/split_soak_proc{i}.log", "w") as f:
                f.write(out)
    err_lines = [l for out in outs for l in out.splitlines()
                 if "ERROR" in l or "Traceback" in l]
    result = {
        "ok": failed == 0 and not err_lines,
        "minutes": args.minutes,
        "cycles": cycles,
        "failed": failed,
        "attach_p50_ms": round(statistics.median(attach_ms), 3) if attach_ms else None,
        "attach_p99_ms": round(sorted(attach_ms)[max(int(len(attach_ms) * 0.99) - 1, 0)], 3) if attach_ms else None,
        "cycle_rate_per_sec": round(cycles / (args.minutes * 60), 2),
        "error_lines": err_lines[:5],
        "node_path": "real KFD/CDI/probe" if gpu else "mock",
        "topology": "serve-only apiserver + remote operator (informer protocol, bearer auth)",
    }
    print(json.dumps(result))
    return 0 if result["ok"] else 1


if __name__ == "__main__":
    sys.exit(main())
