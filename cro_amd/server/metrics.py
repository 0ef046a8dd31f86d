"""Dedicated authenticated metrics listener (cmd/main.go:109-127 parity).

The reference serves /metrics on :8443 over HTTPS behind an authn/authz
filter; this is the standalone analog: TLS from the cert-manager mount and
a bearer token (CRO_METRICS_TOKEN — the ServiceMonitor sends it from the
cro-amd-tokens Secret, config/prometheus/monitor.yaml).  Plain-HTTP and
tokenless modes exist only behind explicit opt-outs; /healthz stays open
for probes.
"""

from __future__ import annotations

import hmac
import logging
import os
import threading
from typing import Optional

from fastapi import FastAPI, HTTPException, Request, Response

log = logging.getLogger(__name__)


def build_metrics_app(token: str = "") -> FastAPI:
    app = FastAPI(title="cro-amd metrics")

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    @app.get("/metrics")
    def metrics(request: Request):
        import prometheus_client

        if token:
            auth = request.headers.get("authorization", "")
            if not hmac.compare_digest(auth, f"Bearer {token}"):
                raise HTTPException(401, "metrics require a valid bearer token")
        return Response(
            prometheus_client.generate_latest(),
            media_type=prometheus_client.CONTENT_TYPE_LATEST,
        )

    return app


class HealthServer:
    """Plain-HTTP /healthz + /readyz on the --health-probe-bind-address
    port (cmd/main.go:205-212 parity) — kubelet probes in manager.yaml
    point here.  A leader-election standby still reports ready, as
    controller-runtime replicas do (the Deployment rollout must complete
    with a passive standby)."""

    def __init__(self, port: int, host: str = "0.0.0.0"):
        import uvicorn

        app = FastAPI(title="cro-amd health")

        @app.get("/healthz")
        def healthz():
            return {"status": "ok"}

        @app.get("/readyz")
        def readyz():
            return {"status": "ok"}

        self.server = uvicorn.Server(
            uvicorn.Config(app, host=host, port=port, log_level="warning",
                           timeout_graceful_shutdown=5)
        )
        self._thread: Optional[threading.Thread] = None

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self.server.run, name="health-server", daemon=True
        )
        self._thread.start()

    def stop(self, timeout: float = 5.0) -> None:
        self.server.should_exit = True
        if self._thread is not None:
            self._thread.join(timeout)


class MetricsServer:
    """uvicorn on its own thread; TLS when cert_dir has tls.crt/tls.key."""

    def __init__(
        self,
        port: int,
        token: str = "",
        cert_dir: str = "",
        certfile: Optional[str] = None,
        keyfile: Optional[str] = None,
        host: str = "0.0.0.0",
    ):
        import uvicorn

        if not certfile and cert_dir:
            c = os.path.join(cert_dir, "tls.crt")
            k = os.path.join(cert_dir, "tls.key")
            if os.path.exists(c) and os.path.exists(k):
                certfile, keyfile = c, k
        self.tls = bool(certfile)
        if not self.tls:
            log.warning(
                "metrics listener on :%d has no TLS material (cert dir %r); "
                "serving plain HTTP — config/prometheus/monitor.yaml expects "
                "HTTPS, mount the serving cert in production", port, cert_dir,
            )
        self.server = uvicorn.Server(
            uvicorn.Config(
                build_metrics_app(token),
                host=host,
                port=port,
                log_level="warning",
                ssl_certfile=certfile,
                ssl_keyfile=keyfile,
                timeout_graceful_shutdown=5,
            )
        )
        self._thread: Optional[threading.Thread] = None

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self.server.run, name="metrics-server", daemon=True
        )
        self._thread.start()

    def stop(self, timeout: float = 5.0) -> None:
        self.server.should_exit = True
        if self._thread is not None:
            self._thread.join(timeout)
