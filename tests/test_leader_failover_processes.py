"""Leader-election failover across REAL operator processes (the cluster
deployment shape the reference's lease election protects,
cmd/main.go:137-155).

One serve-only apiserver process owns the store; two full operator
processes connect over HTTP with --leader-elect.  Exactly one wins the
Lease and reconciles; SIGKILL-ing the leader (a crash — no voluntary
release) lets the standby take over within the lease duration and drive
new work.  Every request carries the fail-closed bearer token.
"""

import json
import os
import signal
import socket
import subprocess
import sys
import time

import httpx
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BASE = "/apis/cro.hpsys.ibm.ie.com/v1alpha1"
TOKEN = "failover-tok"
AUTH = {"Authorization": f"Bearer {TOKEN}"}


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def wait_http(url, timeout=30, proc=None):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        try:
            if httpx.get(url, timeout=1).status_code == 200:
                return
        except Exception:
            time.sleep(0.2)
        if proc is not None and proc.poll() is not None:
            raise AssertionError(f"process died: {proc.stdout.read()}")
    raise AssertionError(f"{url} never came up")


def wait_for(pred, timeout=30, interval=0.2):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(interval)
    return False


@pytest.mark.timeout(180)
def test_leader_failover_across_processes():
    env = dict(os.environ)
    env.update({
        "DEVICE_RESOURCE_TYPE": "DRA", "CDI_PROVIDER_TYPE": "MOCK",
        "CRO_API_TOKEN": TOKEN,
    })
    api_port = free_port()
    procs = []

    def spawn(*args):
        p = subprocess.Popen(
            [sys.executable, "-m", "cro_amd.cmd.main",
             "--metrics-bind-address", f":{free_port()}",
             "--health-probe-bind-address", f":{free_port()}",
             *args],
            cwd=REPO, env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        )
        procs.append(p)
        return p

    try:
        apiserver = spawn("--api-bind-address", f":{api_port}", "--serve-only")
        base = f"http://127.0.0.1:{api_port}"
        wait_http(base + "/healthz", proc=apiserver)

        # register a node with capacity (serve-only has no controllers)
        resp = httpx.post(f"{base}{BASE}/nodes", headers=AUTH, json={
            "apiVersion": "v1", "kind": "Node", "metadata": {"name": "node0"},
            "status": {"capacity": {"milli_cpu": 64000, "memory": 1 << 40,
                                    "allowed_pod_number": 128}},
        }, timeout=5)
        assert resp.status_code == 201, resp.text

        lease_args = [
            "--api-server", base, "--leader-elect", "--simulate-node-path",
            "--leader-lease-duration", "2", "--leader-renew-deadline", "1.5",
            "--leader-retry-period", "0.2",
            "--api-bind-address",  # each operator's own (unused) API port
        ]
        op1 = spawn(*lease_args, f":{free_port()}")
        op2 = spawn(*lease_args, f":{free_port()}")

        # exactly one captured the Lease
        def holder():
            r = httpx.get(
                f"{base}/apis/coordination.k8s.io/v1/leases/"
                "c5744f42.hpsys.ibm.ie.com", headers=AUTH, timeout=5)
            if r.status_code != 200:
                return ""
            return r.json()["spec"]["holderIdentity"]

        assert wait_for(lambda: holder() != "", timeout=30)
        first_holder = holder()
        assert f"_{op1.pid}" in first_holder or f"_{op2.pid}" in first_holder

        # the leader reconciles: a request reaches Running
        resp = httpx.post(f"{base}{BASE}/composabilityrequests", headers=AUTH, json={
            "apiVersion": "cro.hpsys.ibm.ie.com/v1alpha1",
            "kind": "ComposabilityRequest",
            "metadata": {"name": "before-failover"},
            "spec": {"resource": {"type": "gpu", "model": "mi355x", "size": 1,
                                  "target_node": "node0"}},
        }, timeout=5)
        assert resp.status_code == 201, resp.text

        def state(name):
            r = httpx.get(f"{base}{BASE}/composabilityrequests/{name}",
                          headers=AUTH, timeout=5)
            return r.json().get("status", {}).get("state", "") if r.status_code == 200 else ""

        assert wait_for(lambda: state("before-failover") == "Running", timeout=40), \
            state("before-failover")

        # CRASH the leader (SIGKILL: no voluntary lease release)
        leader = op1 if f"_{op1.pid}" in first_holder else op2
        leader.kill()
        leader.wait(timeout=10)

        # the standby takes over after lease expiry and drives NEW work
        # (same type+model on another node — webhook rule 3 allows it)
        resp = httpx.post(f"{base}{BASE}/nodes", headers=AUTH, json={
            "apiVersion": "v1", "kind": "Node", "metadata": {"name": "node1"},
            "status": {"capacity": {"milli_cpu": 64000, "memory": 1 << 40,
                                    "allowed_pod_number": 128}},
        }, timeout=5)
        assert resp.status_code == 201
        resp = httpx.post(f"{base}{BASE}/composabilityrequests", headers=AUTH, json={
            "apiVersion": "cro.hpsys.ibm.ie.com/v1alpha1",
            "kind": "ComposabilityRequest",
            "metadata": {"name": "after-failover"},
            "spec": {"resource": {"type": "gpu", "model": "mi355x", "size": 1,
                                  "target_node": "node1"}},
        }, timeout=5)
        assert resp.status_code == 201, resp.text

        assert wait_for(
            lambda: holder() not in ("", first_holder), timeout=30
        ), f"standby never took over (holder={holder()!r})"
        assert wait_for(lambda: state("after-failover") == "Running", timeout=60), \
            state("after-failover")
        # lease bookkeeping recorded the takeover
        r = httpx.get(
            f"{base}/apis/coordination.k8s.io/v1/leases/"
            "c5744f42.hpsys.ibm.ie.com", headers=AUTH, timeout=5)
        assert r.json()["spec"]["leaseTransitions"] >= 1
    finally:
        for p in procs:
            if p.poll() is None:
                p.send_signal(signal.SIGTERM)
        for p in procs:
            try:
                p.wait(timeout=15)
            except subprocess.TimeoutExpired:
                p.kill()
