#!/usr/bin/env python3
"""Failover-churn soak: one apiserver + TWO leader-elected operator
replicas; lifecycle cycles run continuously while the current Lease
holder is SIGKILLed (crash, no voluntary release) and restarted every
``--kill-every`` seconds.  Proves the fleet keeps converging through
repeated failovers with no lost or stuck requests.

    python tools/failover_soak.py --minutes 8 --kill-every 30
"""

import argparse
import json
import os
import signal
import socket
import statistics
import subprocess
import sys
import tempfile
import time

import httpx

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BASE = "/apis/cro.hpsys.ibm.ie.com/v1alpha1"
TOKEN = "failover-soak-token"
LEASE = "/apis/coordination.k8s.io/v1/leases/c5744f42.hpsys.ibm.ie.com"


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--minutes", type=float, default=8.0)
    p.add_argument("--kill-every", type=float, default=30.0)
    args = p.parse_args()

    node = socket.gethostname()
    env = dict(os.environ)
    env.update({
        "DEVICE_RESOURCE_TYPE": "DRA", "CDI_PROVIDER_TYPE": "MOCK",
        "CRO_API_TOKEN": TOKEN, "CRO_AGENT_TOKEN": TOKEN,
        "CRO_METRICS_TOKEN": TOKEN,
    })
    gpu = os.path.exists("/dev/kfd")
    logs = []
    procs = {}

    def spawn(tag, *extra):
        logf = tempfile.NamedTemporaryFile(
            mode="w+", prefix=f"failover-{tag}-", suffix=".log", delete=False)
        logs.append(logf)
        proc = subprocess.Popen(
            [sys.executable, "-m", "cro_amd.cmd.main",
             "--metrics-bind-address", f":{free_port()}",
             "--health-probe-bind-address", f":{free_port()}",
             *extra],
            cwd=REPO, env=env, stdout=logf, stderr=subprocess.STDOUT, text=True,
        )
        procs[tag] = proc
        return proc

    api_port = free_port()
    spawn("apiserver", "--api-bind-address", f":{api_port}", "--serve-only")
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
        return 1

    def operator_args():
        a = ["--api-server", f"http://127.0.0.1:{api_port}",
             "--api-bind-address", f":{free_port()}",
             "--leader-elect",
             "--leader-lease-duration", "3",
             "--leader-renew-deadline", "2",
             "--leader-retry-period", "0.3",
             "--node", node,
             "--cdi-dir", os.path.join(
                 os.environ.get("TMPDIR", "/tmp"), "cro-failover-cdi")]
        if not gpu:
            a.append("--simulate-node-path")
        return a

    spawn("op-a", *operator_args())
    spawn("op-b", *operator_args())

    r = http.post(f"{BASE}/nodes", json={
        "apiVersion": "v1", "kind": "Node", "metadata": {"name": node},
        "status": {"capacity": {"milli_cpu": 128000, "memory": 2 << 40,
                                "allowed_pod_number": 256}},
    })
    assert r.status_code == 201, r.text

    def holder() -> str:
        try:
            resp = http.get(LEASE)
            return resp.json()["spec"]["holderIdentity"] if resp.status_code == 200 else ""
        except Exception:
            return ""

    body = {
        "apiVersion": "cro.hpsys.ibm.ie.com/v1alpha1",
        "kind": "ComposabilityRequest",
        "metadata": {"name": "fsoak"},
        "spec": {"resource": {"type": "gpu", "model": "mi355x", "size": 1,
                              "target_node": node, "force_detach": True}},
    }

    cycles = failed = kills = 0
    leaked = -1
    attach_ms = []
    t_end = time.monotonic() + args.minutes * 60
    next_kill = time.monotonic() + args.kill_every
    try:
        while time.monotonic() < t_end:
            if time.monotonic() >= next_kill:
                h = holder()
                victim = None
                for tag in ("op-a", "op-b"):
                    if f"_{procs[tag].pid}" in h:
                        victim = tag
                        break
                if victim:
                    procs[victim].kill()
                    procs[victim].wait(timeout=10)
                    kills += 1
                    spawn(victim, *operator_args())  # replace the replica
                next_kill = time.monotonic() + args.kill_every

            t0 = time.monotonic()
            resp = http.post(f"{BASE}/composabilityrequests", json=body)
            if resp.status_code != 201:
                time.sleep(0.1)
                continue  # AlreadyExists while a teardown finishes — not a failure
            ok = gone = False
            while time.monotonic() < t0 + 60:
                g = http.get(f"{BASE}/composabilityrequests/fsoak")
                if g.status_code == 200 and g.json()["status"]["state"] == "Running":
                    ok = True
                    break
                time.sleep(0.005)
            t1 = time.monotonic()
            http.delete(f"{BASE}/composabilityrequests/fsoak")
            while time.monotonic() < t1 + 60:
                if http.get(f"{BASE}/composabilityrequests/fsoak").status_code == 404:
                    gone = True
                    break
                time.sleep(0.005)
            cycles += 1
            if ok and gone:
                attach_ms.append((t1 - t0) * 1e3)
            else:
                failed += 1
        time.sleep(2)  # drain any in-flight teardown before the leak check
        leaked = http_safe_list(http, f"{BASE}/composableresources")
    finally:
        for tag in ("op-a", "op-b", "apiserver"):
            proc = procs.get(tag)
            if proc is None or proc.poll() is not None:
                continue
            proc.send_signal(signal.SIGTERM)
            try:
                proc.wait(timeout=20)
            except subprocess.TimeoutExpired:
                proc.kill()
                proc.wait()

    result = {
        "ok": failed == 0 and kills >= 2 and leaked == 0,
        "minutes": args.minutes,
        "cycles": cycles,
        "failed": failed,
        "leader_kills": kills,
        "leaked_resources": leaked,
        "attach_p50_ms": round(statistics.median(attach_ms), 3) if attach_ms else None,
        "attach_p99_ms": round(sorted(attach_ms)[max(int(len(attach_ms) * 0.99) - 1, 0)], 3) if attach_ms else None,
        "node_path": "real KFD/CDI/probe" if gpu else "mock",
    }
    print(json.dumps(result))
    return 0 if result["ok"] else 1


def http_safe_list(http, url) -> int:
    try:
        resp = http.get(url)
        return len(resp.json().get("items", [])) if resp.status_code == 200 else -1
    except Exception:
        return -1


if __name__ == "__main__":
    sys.exit(main())
