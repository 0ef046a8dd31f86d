"""Operator process tests: entrypoint serving + checkpoint/resume semantics
(the reference's restart story: all progress lives in object status, a new
manager resumes mid-state-machine — SURVEY.md §5.4)."""

import os
import signal
import socket
import subprocess
import sys
import time

import httpx
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.mark.timeout(120)
def test_main_serves_and_shuts_down(tmp_path):
    port = free_port()
    env = dict(os.environ)
    env.update({
        "DEVICE_RESOURCE_TYPE": "DRA", "CDI_PROVIDER_TYPE": "MOCK",
        "CRO_API_TOKEN": "test-api-token", "CRO_AGENT_TOKEN": "test-agent-token",
        "CRO_METRICS_TOKEN": "test-metrics-token",
    })
    proc = subprocess.Popen(
        [
            sys.executable, "-m", "cro_amd.cmd.main",
            "--api-bind-address", f":{port}",
            "--metrics-bind-address", f":{free_port()}",
            "--node", "test-node",
            "--cdi-dir", str(tmp_path / "cdi"),
            "--leader-elect",
            "--data-dir", str(tmp_path / "data"),
        ],
        cwd=REPO,
        env=env,
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        text=True,
    )
    try:
        base = f"http://127.0.0.1:{port}"
        deadline = time.monotonic() + 30
        up = False
        while time.monotonic() < deadline:
            try:
                if httpx.get(base + "/healthz", timeout=1).status_code == 200:
                    up = True
                    break
            except Exception:
                time.sleep(0.2)
        assert up, proc.stdout.read() if proc.poll() is not None else "no healthz"

        # API surface is fail-closed: no token → 401 k8s Status
        resp = httpx.get(
            base + "/apis/cro.hpsys.ibm.ie.com/v1alpha1/composabilityrequests",
            timeout=5,
        )
        assert resp.status_code == 401
        assert resp.json()["reason"] == "Unauthorized"
        # ... and live with the bearer token
        resp = httpx.get(
            base + "/apis/cro.hpsys.ibm.ie.com/v1alpha1/composabilityrequests",
            headers={"Authorization": "Bearer test-api-token"},
            timeout=5,
        )
        assert resp.status_code == 200
        assert resp.json()["items"] == []
        # metrics exposed through the API process (bearer-gated)
        assert httpx.get(base + "/metrics", timeout=5).status_code == 401
        assert b"cro_reconcile_total" in httpx.get(
            base + "/metrics", timeout=5,
            headers={"Authorization": "Bearer test-metrics-token"},
        ).content
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait()
    assert proc.returncode == 0


def test_operator_restart_resumes_state_machine():
    """A new manager over the same store finishes what the old one started:
    the CR status IS the checkpoint."""
    from cro_amd.api.v1alpha1.types import ComposabilityRequest
    from cro_amd.bench_harness import build_local_stack
    from cro_amd.controllers import build_manager
    from tests.conftest import make_request

    stack = build_local_stack(node_name="node0", use_gpu=False)
    # do NOT start the manager: create the request and hand-advance only the
    # request controller so children exist but are not yet Online
    stack.mgr.client.create(make_request("r1", size=2, target_node="node0"))
    rec = stack.mgr.request_reconciler
    for _ in range(3):
        rec.reconcile("r1")
    mid = stack.mgr.client.get(ComposabilityRequest, "r1")
    assert mid.status.state == "Updating"  # checkpoint: mid-flight

    # "restart": fresh manager instance over the same store
    mgr2 = build_manager(stack.mgr.resource_reconciler.adapter, None, store=stack.mgr.store)
    mgr2.resource_reconciler.node_ops = stack.ops
    mgr2.start()  # informer-cache replay re-queues every stored object
    try:
        assert mgr2.wait_for(
            lambda: mgr2.client.get(ComposabilityRequest, "r1").status.state == "Running",
            timeout=10,
        )
    finally:
        mgr2.stop()


def test_token_autogeneration(tmp_path):
    """No CRO_API_TOKEN in the env → the entrypoint generates one, persists
    it 0600 under --data-dir, and the API refuses unauthenticated access
    while accepting the generated token (fail-closed default)."""
    port = free_port()
    data = tmp_path / "data"
    env = {k: v for k, v in os.environ.items()
           if k not in ("CRO_API_TOKEN", "CRO_AGENT_TOKEN")}
    env.update({"DEVICE_RESOURCE_TYPE": "DRA", "CDI_PROVIDER_TYPE": "MOCK"})
    proc = subprocess.Popen(
        [
            sys.executable, "-m", "cro_amd.cmd.main",
            "--api-bind-address", f":{port}",
            "--metrics-bind-address", f":{free_port()}",
            "--serve-only",
            "--data-dir", str(data),
        ],
        cwd=REPO, env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    try:
        base = f"http://127.0.0.1:{port}"
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            try:
                if httpx.get(base + "/healthz", timeout=1).status_code == 200:
                    break
            except Exception:
                time.sleep(0.2)
        else:
            raise AssertionError(proc.stdout.read() if proc.poll() is not None else "no healthz")
        url = base + "/apis/cro.hpsys.ibm.ie.com/v1alpha1/nodes"
        assert httpx.get(url, timeout=5).status_code == 401
        token_path = data / "api.token"
        assert token_path.exists()
        assert (token_path.stat().st_mode & 0o777) == 0o600
        token = token_path.read_text().strip()
        resp = httpx.get(url, headers={"Authorization": f"Bearer {token}"}, timeout=5)
        assert resp.status_code == 200
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


@pytest.mark.timeout(120)
def test_main_serves_tls(tmp_path):
    """--tls-cert-file/--tls-key-file serve the API over TLS (the
    reference's webhook/metrics are TLS-only behind cert-manager certs,
    cmd/main.go:109-127 + config/certmanager)."""
    cert = tmp_path / "tls.crt"
    key = tmp_path / "tls.key"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(cert), "-days", "1",
         "-subj", "/CN=127.0.0.1",
         "-addext", "subjectAltName=IP:127.0.0.1"],
        check=True, capture_output=True,
    )
    port = free_port()
    env = dict(os.environ)
    env.update({"DEVICE_RESOURCE_TYPE": "DRA", "CDI_PROVIDER_TYPE": "MOCK",
                "CRO_API_TOKEN": "tls-tok"})
    proc = subprocess.Popen(
        [
            sys.executable, "-m", "cro_amd.cmd.main",
            "--api-bind-address", f":{port}",
            "--metrics-bind-address", f":{free_port()}",
            "--serve-only",
            "--tls-cert-file", str(cert), "--tls-key-file", str(key),
        ],
        cwd=REPO, env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    try:
        base = f"https://127.0.0.1:{port}"
        deadline = time.monotonic() + 30
        up = False
        while time.monotonic() < deadline:
            try:
                if httpx.get(base + "/healthz", timeout=1, verify=False).status_code == 200:
                    up = True
                    break
            except Exception:
                time.sleep(0.2)
        assert up, proc.stdout.read() if proc.poll() is not None else "no TLS healthz"
        # plain HTTP against the TLS port must fail
        with pytest.raises(Exception):
            httpx.get(f"http://127.0.0.1:{port}/healthz", timeout=2)
        # croctl reaches it with --insecure-skip-tls-verify or the CA
        from cro_amd.cmd.croctl import main as croctl

        assert croctl(["--server", base, "--insecure-skip-tls-verify",
                       "--token", "tls-tok", "get", "nodes"]) == 0
        assert croctl(["--server", base, "--certificate-authority", str(cert),
                       "--token", "tls-tok", "get", "nodes"]) == 0
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


def test_tls_flags_must_pair():
    from cro_amd.cmd.main import main as entry

    with pytest.raises(SystemExit):
        entry(["--tls-cert-file", "/tmp/only-cert.pem"])


@pytest.mark.timeout(120)
def test_data_dir_survives_restart(tmp_path):
    """--data-dir: SIGTERM the operator, restart it, the fleet state is
    still there (standalone etcd analog, end to end)."""
    data = tmp_path / "data"
    env = dict(os.environ)
    env.update({"DEVICE_RESOURCE_TYPE": "DRA", "CDI_PROVIDER_TYPE": "MOCK",
                "CRO_API_TOKEN": "persist-tok"})
    auth = {"Authorization": "Bearer persist-tok"}

    def start(port):
        return subprocess.Popen(
            [
                sys.executable, "-m", "cro_amd.cmd.main",
                "--api-bind-address", f":{port}",
                "--metrics-bind-address", f":{free_port()}",
                "--serve-only",
                "--data-dir", str(data),
            ],
            cwd=REPO, env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        )

    def wait_up(port, proc):
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            try:
                if httpx.get(f"http://127.0.0.1:{port}/healthz", timeout=1).status_code == 200:
                    return True
            except Exception:
                time.sleep(0.2)
        raise AssertionError(proc.stdout.read() if proc.poll() is not None else "no healthz")

    port = free_port()
    proc = start(port)
    try:
        wait_up(port, proc)
        base = f"http://127.0.0.1:{port}/apis/cro.hpsys.ibm.ie.com/v1alpha1"
        resp = httpx.post(f"{base}/composabilityrequests", headers=auth, json={
            "apiVersion": "cro.hpsys.ibm.ie.com/v1alpha1",
            "kind": "ComposabilityRequest",
            "metadata": {"name": "persist-me"},
            "spec": {"resource": {"type": "gpu", "model": "mi355x", "size": 2,
                                  "target_node": "nodeX"}},
        })
        assert resp.status_code == 201, resp.text
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()

    port2 = free_port()
    proc2 = start(port2)
    try:
        wait_up(port2, proc2)
        base2 = f"http://127.0.0.1:{port2}/apis/cro.hpsys.ibm.ie.com/v1alpha1"
        resp = httpx.get(f"{base2}/composabilityrequests/persist-me",
                         headers=auth, timeout=5)
        assert resp.status_code == 200, resp.text
        assert resp.json()["spec"]["resource"]["size"] == 2
    finally:
        proc2.send_signal(signal.SIGTERM)
        try:
            proc2.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc2.kill()


@pytest.mark.timeout(120)
def test_webhook_served_from_entrypoint(tmp_path):
    """The production entrypoint serves the AdmissionReview endpoint on
    :9443 with the cert-dir TLS material (cmd/main.go:196-201 parity) —
    a CREATE violating webhook rule 1 is rejected THROUGH the HTTPS
    endpoint that config/webhook/manifests.yaml registers."""
    certdir = tmp_path / "serving-certs"
    certdir.mkdir()
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(certdir / "tls.key"), "-out", str(certdir / "tls.crt"),
         "-days", "1", "-subj", "/CN=127.0.0.1",
         "-addext", "subjectAltName=IP:127.0.0.1"],
        check=True, capture_output=True,
    )
    api_port, wh_port = free_port(), free_port()
    env = dict(os.environ)
    env.update({"DEVICE_RESOURCE_TYPE": "DRA", "CDI_PROVIDER_TYPE": "MOCK",
                "CRO_API_TOKEN": "wh-tok", "ENABLE_WEBHOOKS": "true"})
    proc = subprocess.Popen(
        [
            sys.executable, "-m", "cro_amd.cmd.main",
            "--api-bind-address", f":{api_port}",
            "--metrics-bind-address", f":{free_port()}",
            "--webhook-bind-address", f":{wh_port}",
            "--webhook-cert-dir", str(certdir),
            "--serve-only",
        ],
        cwd=REPO, env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    try:
        base = f"https://127.0.0.1:{wh_port}"
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            try:
                if httpx.get(base + "/healthz", timeout=1,
                             verify=str(certdir / "tls.crt")).status_code == 200:
                    break
            except Exception:
                time.sleep(0.2)
        else:
            raise AssertionError(
                proc.stdout.read() if proc.poll() is not None else "no webhook healthz")

        review = {
            "apiVersion": "admission.k8s.io/v1",
            "kind": "AdmissionReview",
            "request": {
                "uid": "test-uid-1",
                "operation": "CREATE",
                "object": {
                    "apiVersion": "cro.hpsys.ibm.ie.com/v1alpha1",
                    "kind": "ComposabilityRequest",
                    "metadata": {"name": "bad"},
                    "spec": {"resource": {
                        "type": "gpu", "model": "mi355x", "size": 1,
                        "allocation_policy": "differentnode",
                        "target_node": "nodeX",  # rule 1 violation
                    }},
                },
            },
        }
        resp = httpx.post(
            base + "/validate-cro-hpsys-ibm-ie-com-v1alpha1-composabilityrequest",
            json=review, timeout=5, verify=str(certdir / "tls.crt"),
        )
        assert resp.status_code == 200
        body = resp.json()
        assert body["response"]["uid"] == "test-uid-1"
        assert body["response"]["allowed"] is False
        assert "differentnode" in body["response"]["status"]["message"]

        # a valid CREATE passes through the same endpoint
        review["request"]["uid"] = "test-uid-2"
        review["request"]["object"]["spec"]["resource"].pop("target_node")
        resp = httpx.post(
            base + "/validate-cro-hpsys-ibm-ie-com-v1alpha1-composabilityrequest",
            json=review, timeout=5, verify=str(certdir / "tls.crt"),
        )
        assert resp.json()["response"]["allowed"] is True
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
    assert proc.returncode == 0


@pytest.mark.timeout(120)
def test_metrics_served_as_servicemonitor_expects(tmp_path):
    """config/prometheus/monitor.yaml declares scheme=https + bearer token:
    the dedicated metrics listener must serve exactly that (VERDICT r1 #6:
    TLS from the cert-dir mount, CRO_METRICS_TOKEN enforced), and the
    kubelet probe port (:8081) must answer /healthz /readyz."""
    certdir = tmp_path / "metrics-certs"
    certdir.mkdir()
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(certdir / "tls.key"), "-out", str(certdir / "tls.crt"),
         "-days", "1", "-subj", "/CN=127.0.0.1",
         "-addext", "subjectAltName=IP:127.0.0.1"],
        check=True, capture_output=True,
    )
    api_port, m_port, h_port = free_port(), free_port(), free_port()
    env = dict(os.environ)
    env.update({
        "DEVICE_RESOURCE_TYPE": "DRA", "CDI_PROVIDER_TYPE": "MOCK",
        "CRO_API_TOKEN": "t", "CRO_METRICS_TOKEN": "scrape-tok",
    })
    proc = subprocess.Popen(
        [
            sys.executable, "-m", "cro_amd.cmd.main",
            "--api-bind-address", f":{api_port}",
            "--metrics-bind-address", f":{m_port}",
            "--metrics-cert-dir", str(certdir),
            "--health-probe-bind-address", f":{h_port}",
            "--serve-only",
        ],
        cwd=REPO, env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    try:
        base = f"https://127.0.0.1:{m_port}"
        ca = str(certdir / "tls.crt")
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            try:
                if httpx.get(base + "/healthz", timeout=1, verify=ca).status_code == 200:
                    break
            except Exception:
                time.sleep(0.2)
        else:
            raise AssertionError(
                proc.stdout.read() if proc.poll() is not None else "no metrics healthz")

        # the exact scrape the ServiceMonitor performs: HTTPS + bearer
        resp = httpx.get(base + "/metrics", verify=ca, timeout=5,
                         headers={"Authorization": "Bearer scrape-tok"})
        assert resp.status_code == 200
        assert b"cro_reconcile_total" in resp.content
        # no/wrong token → 401
        assert httpx.get(base + "/metrics", verify=ca, timeout=5).status_code == 401
        # plain HTTP against the TLS port fails
        with pytest.raises(Exception):
            httpx.get(f"http://127.0.0.1:{m_port}/metrics", timeout=2)
        # kubelet probe port answers (plain HTTP, no auth — probe semantics)
        assert httpx.get(f"http://127.0.0.1:{h_port}/healthz", timeout=5).status_code == 200
        assert httpx.get(f"http://127.0.0.1:{h_port}/readyz", timeout=5).status_code == 200
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
    assert proc.returncode == 0


@pytest.mark.timeout(120)
def test_boot_with_fti_fm_backend(tmp_path):
    """The operator boots with a real fabric backend selected (FTI FM):
    provider construction + client wiring must not require a live fabric
    (contacted lazily), and the process serves and shuts down cleanly."""
    port = free_port()
    env = dict(os.environ)
    env.update({
        "DEVICE_RESOURCE_TYPE": "DRA",
        "CDI_PROVIDER_TYPE": "FTI_CDI",
        "FTI_CDI_API_TYPE": "FM",
        "FTI_CDI_ENDPOINT": "fabric.invalid",
        "FTI_CDI_TENANT_ID": "t-1",
    })
    proc = subprocess.Popen(
        [
            sys.executable, "-m", "cro_amd.cmd.main",
            "--api-bind-address", f":{port}",
            "--metrics-bind-address", f":{free_port()}",
            "--syncer-period", "3600",
            "--cdi-dir", str(tmp_path / "cdi"),
        ],
        cwd=REPO, env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    try:
        deadline = time.monotonic() + 30
        up = False
        while time.monotonic() < deadline:
            try:
                if httpx.get(f"http://127.0.0.1:{port}/healthz", timeout=1).status_code == 200:
                    up = True
                    break
            except Exception:
                time.sleep(0.2)
        assert up, proc.stdout.read() if proc.poll() is not None else "no healthz"
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
    assert proc.returncode == 0


@pytest.mark.timeout(120)
def test_split_mode_token_mismatch_fails_fast(tmp_path):
    """--api-server with a wrong bearer token must exit with a clear error
    immediately, not loop on silent 401 watch reconnects."""
    port = free_port()
    env = dict(os.environ)
    env.update({"DEVICE_RESOURCE_TYPE": "DRA", "CDI_PROVIDER_TYPE": "MOCK",
                "CRO_API_TOKEN": "right-token"})
    server = subprocess.Popen(
        [sys.executable, "-m", "cro_amd.cmd.main",
         "--api-bind-address", f":{port}",
         "--metrics-bind-address", f":{free_port()}",
         "--health-probe-bind-address", f":{free_port()}",
         "--serve-only"],
        cwd=REPO, env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    try:
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            try:
                if httpx.get(f"http://127.0.0.1:{port}/healthz", timeout=1).status_code == 200:
                    break
            except Exception:
                time.sleep(0.2)
        env2 = dict(env)
        env2["CRO_API_TOKEN"] = "wrong-token"
        op = subprocess.run(
            [sys.executable, "-m", "cro_amd.cmd.main",
             "--api-server", f"http://127.0.0.1:{port}",
             "--api-bind-address", f":{free_port()}",
             "--metrics-bind-address", f":{free_port()}",
             "--health-probe-bind-address", f":{free_port()}",
             "--simulate-node-path"],
            cwd=REPO, env=env2,
            capture_output=True, text=True, timeout=60,
        )
        assert op.returncode == 1
        assert "rejected our bearer token" in op.stdout + op.stderr
    finally:
        server.send_signal(signal.SIGTERM)
        try:
            server.wait(timeout=15)
        except subprocess.TimeoutExpired:
            server.kill()
