"""Operator process entrypoint (cmd/main.go parity).

Flags mirror the reference (cmd/main.go:68-81) where they still apply, plus
the standalone-mode additions:

    python -m cro_amd.cmd.main \
        --metrics-bind-address :8443 --health-probe-bind-address :8081 \
        --api-bind-address :8080 --leader-elect \
        --node <name> [--cdi-dir /etc/cdi] [--destructive]

Env config (composableresource_adapter.go:40-76 and backends):
  DEVICE_RESOURCE_TYPE, CDI_PROVIDER_TYPE, FTI_CDI_API_TYPE,
  FTI_CDI_ENDPOINT, FTI_CDI_TENANT_ID, FTI_CDI_CLUSTER_ID,
  NEC_CDIM_IP, LAYOUT_APPLY_PORT, CONFIGURATION_MANAGER_PORT,
  NEC_PROVISIONAL_GPU_UUID, SUNFISH_ENDPOINT, ENABLE_WEBHOOKS,
  CRO_FTI_USERNAME/PASSWORD/CLIENT_ID/CLIENT_SECRET/REALM.

Leader election (--leader-elect) uses a coordination.k8s.io Lease held
through the client surface (runtime/lease.py) — the same protocol as the
reference's lease-based election (cmd/main.go:137-155, LeaderElectionID
c5744f42.hpsys.ibm.ie.com). Against a remote API server this excludes
replicas on different nodes; against the embedded store it is a no-op
guard (the store is process-local). Lost leadership is fatal: the process
exits so the standby's controllers never overlap with ours.

Security posture (fail closed): the API (/apis) and node-agent (/agent)
surfaces require bearer tokens. When CRO_API_TOKEN / CRO_AGENT_TOKEN are
unset, tokens are auto-generated at startup and written 0600 under
--data-dir (or /var/run/cro-amd) — a default deployment never exposes an
unauthenticated write surface.
"""

from __future__ import annotations

import argparse
import logging
import os
import signal
import sys
import threading


def parse_port(addr: str, default: int) -> int:
    if not addr:
        return default
    return int(addr.rsplit(":", 1)[-1])


def main(argv=None) -> int:
    from .. import __version__

    p = argparse.ArgumentParser(prog="cro-amd-operator")
    p.add_argument("--version", action="version", version=f"cro-amd-operator {__version__}")
    p.add_argument("--metrics-bind-address", default=":8443")
    p.add_argument("--health-probe-bind-address", default=":8081")
    p.add_argument("--api-bind-address", default=":8080")
    p.add_argument("--leader-elect", action="store_true",
                   help="gate controllers behind a coordination Lease "
                   "(holder renews every --leader-retry-period; a standby "
                   "takes over when renewal stops)")
    p.add_argument("--leader-lease-duration", type=float, default=15.0)
    p.add_argument("--leader-renew-deadline", type=float, default=10.0)
    p.add_argument("--leader-retry-period", type=float, default=2.0)
    p.add_argument("--webhook-bind-address", default=":9443",
                   help="AdmissionReview endpoint (ValidatingWebhook"
                   "Configuration target); serves when ENABLE_WEBHOOKS "
                   "is not 'false' and TLS material exists")
    p.add_argument("--webhook-cert-dir",
                   default="/tmp/k8s-webhook-server/serving-certs",
                   help="directory with tls.crt/tls.key (cert-manager "
                   "mount; kubebuilder default path)")
    p.add_argument("--webhook-insecure", action="store_true",
                   help="serve the webhook over plain HTTP (tests only; "
                   "a real apiserver requires TLS)")
    p.add_argument("--metrics-cert-dir",
                   default="/tmp/k8s-metrics-server/serving-certs",
                   help="directory with tls.crt/tls.key for the metrics "
                   "listener (cert-manager mount); without certs the "
                   "metrics port serves plain HTTP with a warning")
    p.add_argument("--metrics-secure", action="store_true", default=True,
                   help="require the bearer token on the dedicated metrics "
                   "port (reference --metrics-secure parity; default on)")
    p.add_argument("--node", default=os.environ.get("NODE_NAME", ""),
                   help="local node name this operator instance manages")
    p.add_argument("--api-server", default="",
                   help="run controllers against a remote cro-amd API server "
                   "(URL) instead of the embedded store")
    p.add_argument("--serve-only", action="store_true",
                   help="serve the API/store without running controllers "
                   "(the apiserver half of a split deployment)")
    p.add_argument("--cdi-dir", default="/etc/cdi")
    p.add_argument("--data-dir", default="",
                   help="standalone durability: persist the object store "
                   "here (atomic snapshots; reload on restart — the etcd "
                   "analog). Ignored with --api-server.")
    p.add_argument("--destructive", action="store_true",
                   help="perform real PCI remove/rescan and module unload")
    p.add_argument("--simulate-node-path", action="store_true",
                   help="use the in-memory node-path double instead of real "
                   "KFD/sysfs (CI and control-plane demos on GPU-less "
                   "machines; BASELINE config #1 shape)")
    p.add_argument("--max-concurrent-reconciles", type=int, default=8)
    p.add_argument("--syncer-period", type=float, default=60.0)
    p.add_argument("--zap-log-level", default="info")
    p.add_argument("--log-format", choices=["text", "json"], default="text")
    p.add_argument("--tls-cert-file", default="",
                   help="serve the API over TLS (the reference serves its "
                   "webhook and metrics over TLS from cert-manager certs)")
    p.add_argument("--tls-key-file", default="")
    args = p.parse_args(argv)
    if bool(args.tls_cert_file) != bool(args.tls_key_file):
        p.error("--tls-cert-file and --tls-key-file must be given together")

    if args.log_format == "json":
        import json as _json

        class JsonFormatter(logging.Formatter):
            def format(self, record):
                return _json.dumps({
                    "ts": self.formatTime(record),
                    "level": record.levelname,
                    "logger": record.name,
                    "msg": record.getMessage(),
                })

        handler = logging.StreamHandler()
        handler.setFormatter(JsonFormatter())
        logging.basicConfig(
            level=getattr(logging, args.zap_log_level.upper(), logging.INFO),
            handlers=[handler],
        )
    else:
        logging.basicConfig(
            level=getattr(logging, args.zap_log_level.upper(), logging.INFO),
            format="%(asctime)s %(levelname)s %(name)s %(message)s",
        )
    # httpx/httpcore log every request at INFO — in remote mode that is
    # 30+ lines per reconcile cycle of pure noise (and enough volume to
    # block the process if a supervisor attaches a pipe it never drains).
    # Production norm: transport logs only at WARNING unless debugging.
    if args.zap_log_level.lower() != "debug":
        for noisy in ("httpx", "httpcore"):
            logging.getLogger(noisy).setLevel(logging.WARNING)

    log = logging.getLogger("cro_amd.main")

    # -- fail-closed bearer tokens ------------------------------------------
    # Deployments set these from the cro-amd-tokens Secret; standalone runs
    # get fresh random tokens written 0600 so nothing serves unauthenticated.
    token_dir = args.data_dir or "/var/run/cro-amd"
    api_token = os.environ.get("CRO_API_TOKEN", "")
    agent_token = os.environ.get("CRO_AGENT_TOKEN", "")
    metrics_token = os.environ.get("CRO_METRICS_TOKEN", "")
    generated = {}
    if not api_token:
        import secrets

        api_token = secrets.token_hex(24)
        generated["api.token"] = ("CRO_API_TOKEN", api_token)
    if not agent_token and args.node:
        import secrets

        agent_token = secrets.token_hex(24)
        generated["agent.token"] = ("CRO_AGENT_TOKEN", agent_token)
    if not metrics_token and args.metrics_secure:
        import secrets

        metrics_token = secrets.token_hex(24)
        generated["metrics.token"] = ("CRO_METRICS_TOKEN", metrics_token)
    if generated:
        try:
            os.makedirs(token_dir, exist_ok=True)
            for fname, (envname, value) in generated.items():
                path = os.path.join(token_dir, fname)
                fd = os.open(path, os.O_WRONLY | os.O_CREAT | os.O_TRUNC, 0o600)
                with os.fdopen(fd, "w") as f:
                    f.write(value)
                log.warning(
                    "no %s configured; generated one at %s (set the env var "
                    "from a Secret for multi-process deployments)",
                    envname, path,
                )
        except OSError as exc:
            log.warning("could not persist generated tokens (%s); clients "
                        "of this process must read them from the log", exc)
    # the shared API app's /metrics route reads CRO_METRICS_TOKEN directly
    if metrics_token:
        os.environ["CRO_METRICS_TOKEN"] = metrics_token

    from ..controllers import build_manager
    from ..fabric.adapter import new_adapter
    from ..nodeops.amdgpu import AmdNodeOps
    from ..nodeops.execs import LocalNodeExec

    adapter = new_adapter()

    remote = None
    if args.api_server:
        from ..runtime.remote import RemoteClient

        # SharedInformer read cache (client-go shape): reconciles read from
        # the watch-fed local cache instead of paying a GET RTT per read.
        # CRO_CLIENT_CACHE=off falls back to direct reads.
        remote = RemoteClient(
            args.api_server, token=api_token,
            cache=os.environ.get("CRO_CLIENT_CACHE", "on") != "off",
        )
        # fail fast on auth/connectivity: a token mismatch (each process
        # auto-generates its own when the Secret is not set) would
        # otherwise surface only as silent 401 watch-reconnect loops
        from ..api.v1alpha1.types import Node
        from ..runtime.errors import ApiError

        deadline = 30.0
        import time as _time

        t0 = _time.monotonic()
        while True:
            try:
                remote.list(Node)
                break
            except ApiError as exc:
                if "401" in str(exc):
                    log.error(
                        "API server %s rejected our bearer token (401). Set "
                        "CRO_API_TOKEN identically on both processes (the "
                        "cro-amd-tokens Secret); auto-generated tokens are "
                        "per-process.", args.api_server)
                    return 1
                if _time.monotonic() - t0 > deadline:
                    log.error("API server %s unreachable: %s", args.api_server, exc)
                    return 1
                _time.sleep(0.5)
            except Exception as exc:
                if _time.monotonic() - t0 > deadline:
                    log.error("API server %s unreachable: %s", args.api_server, exc)
                    return 1
                _time.sleep(0.5)
    store = None
    if remote is None and args.data_dir:
        from ..runtime.store import InMemoryStore

        store = InMemoryStore(
            persist_path=os.path.join(args.data_dir, "state.json")
        )
        log.info("durable store at %s", args.data_dir)
    mgr = build_manager(
        adapter,
        None,  # node_ops installed below
        store=store,
        client=remote,
        max_concurrent_reconciles=args.max_concurrent_reconciles,
        enable_webhook=os.environ.get("ENABLE_WEBHOOKS", "") != "false",
        syncer_period=args.syncer_period,
    )
    # FTI/NEC providers resolve Node→machine through the cluster client,
    # which does not exist until the manager is built — wire it now (a
    # None client here would crash the first fabric call in production)
    if getattr(adapter.provider, "client", "n/a") is None:
        adapter.provider.client = mgr.client
    probe_fn = None
    try:
        from ..nodeops.probe import load_library, probe_fn_for_nodeops

        if load_library(required=False) is not None:
            probe_fn = probe_fn_for_nodeops
    except Exception:
        log.warning("gfx950 probe library unavailable; health probe disabled")
    execer = LocalNodeExec()
    if args.simulate_node_path:
        from ..nodeops.amdgpu import MockNodeOps

        gpu_ops = MockNodeOps(client=mgr.client)
        node_ops = gpu_ops
        # bridge: fabric composition makes the device node-visible after
        # the rescan analog (the bench-harness binding)
        provider = adapter.provider
        orig_add = getattr(provider, "add_resource", None)
        if orig_add is not None:
            def _sim_add(resource, _orig=orig_add, _ops=gpu_ops):
                did, cdi = _orig(resource)
                _ops.fabric_composed(resource.spec.target_node, did)
                return did, cdi

            provider.add_resource = _sim_add
        log.info("node path SIMULATED (--simulate-node-path)")
    else:
        gpu_ops = AmdNodeOps(
            execer,
            client=mgr.client,
            cdi_dir=args.cdi_dir,
            destructive=args.destructive,
            probe_fn=probe_fn,
        )
        from ..nodeops.composite import CompositeNodeOps
        from ..nodeops.cxl import CxlNodeOps

        node_ops = CompositeNodeOps(
            {
                "gpu": gpu_ops,
                "cxlmemory": CxlNodeOps(
                    execer, cdi_dir=args.cdi_dir, destructive=args.destructive
                ),
            }
        )
    mgr.resource_reconciler.node_ops = node_ops
    if hasattr(mgr, "syncer"):
        mgr.syncer.node_ops = node_ops

    # standalone mode owns the store, so the local node registers itself
    # (cluster mode gets Nodes from the apiserver)
    if remote is None and args.node:
        from ..api.v1alpha1.types import Node
        from ..runtime.errors import AlreadyExistsError

        node_obj = Node()
        node_obj.metadata.name = args.node
        node_obj.status.capacity.milli_cpu = (os.cpu_count() or 1) * 1000
        try:
            with open("/proc/meminfo") as f:
                for line in f:
                    if line.startswith("MemTotal:"):
                        node_obj.status.capacity.memory = int(line.split()[1]) * 1024
                        break
        except OSError:
            pass
        node_obj.status.capacity.allowed_pod_number = 110
        node_obj.status.capacity.ephemeral_storage = 1 << 40
        try:
            mgr.client.create(node_obj)
            log.info("registered local node %s", args.node)
        except AlreadyExistsError:
            pass

    # MOCK fabric + a local node: bind the pool to the real inventory so
    # attaches hand out devices that actually exist (the bench-harness
    # binding, made available to the production entrypoint for demos and
    # single-node operation without a physical fabric)
    if os.environ.get("CDI_PROVIDER_TYPE", "") == "MOCK" and args.node and not args.destructive:
        try:
            gpus = gpu_ops.enumerate(args.node)
        except Exception:
            gpus = []
        if gpus:
            from ..fabric.mock import MockFabric

            fabric = MockFabric(
                bind_inventory=[
                    {
                        "device_id": g.device_id,
                        "cdi_device_id": f"amd.com/gpu={g.device_id}",
                        "model": "mi355x",
                    }
                    for g in gpus
                ]
            )
            gpu_ops._sim_detached = {g.device_id for g in gpus}
            gpu_ops._invalidate_enum(args.node)

            orig_add = fabric.add_resource

            def _bound_add(resource):
                did, cdi = orig_add(resource)
                gpu_ops.simulate_compose(resource.spec.target_node, did)
                return did, cdi

            fabric.add_resource = _bound_add
            adapter.provider = fabric
            log.info("MOCK fabric bound to %d local device(s)", len(gpus))

    from ..server.api import build_app
    from ..server.agent_api import build_agent_app

    import uvicorn

    app = build_app(mgr.client, token=api_token)
    if args.node and not args.simulate_node_path:
        # node-agent surface for off-node controllers (the simulated node
        # path has no NodeExec to expose)
        build_agent_app(
            node_ops.execer, node_name=args.node, app=app, token=agent_token
        )
    server = uvicorn.Server(
        uvicorn.Config(
            app,
            host="0.0.0.0",
            port=parse_port(args.api_bind_address, 8080),
            log_level="warning",
            ssl_certfile=args.tls_cert_file or None,
            ssl_keyfile=args.tls_key_file or None,
            # bound SIGTERM: open watch STREAMS otherwise keep graceful
            # shutdown waiting until the watch timeout (minutes)
            timeout_graceful_shutdown=5,
        )
    )

    stop = threading.Event()

    def handle_signal(signum, frame):
        log.info("signal %s; shutting down", signum)
        server.should_exit = True
        stop.set()

    signal.signal(signal.SIGTERM, handle_signal)
    signal.signal(signal.SIGINT, handle_signal)

    # -- admission webhook listener (:9443) ---------------------------------
    # The reference serves its ValidatingWebhook from the manager process
    # (cmd/main.go:196-201); config/webhook/manifests.yaml registers this
    # endpoint with failurePolicy=Fail, so the listener MUST exist whenever
    # the manifests are applied.
    webhook_server = None
    webhook_thread = None
    enable_webhooks = os.environ.get("ENABLE_WEBHOOKS", "") != "false"
    if enable_webhooks:
        cert = os.path.join(args.webhook_cert_dir, "tls.crt")
        key = os.path.join(args.webhook_cert_dir, "tls.key")
        have_tls = os.path.exists(cert) and os.path.exists(key)
        if have_tls or args.webhook_insecure:
            from ..api.v1alpha1.types import ComposabilityRequest
            from ..webhook.server import build_app as build_webhook_app

            webhook_app = build_webhook_app(
                lambda: mgr.client.list(ComposabilityRequest)
            )
            webhook_server = uvicorn.Server(
                uvicorn.Config(
                    webhook_app,
                    host="0.0.0.0",
                    port=parse_port(args.webhook_bind_address, 9443),
                    log_level="warning",
                    ssl_certfile=cert if have_tls else None,
                    ssl_keyfile=key if have_tls else None,
                    timeout_graceful_shutdown=5,
                )
            )
            webhook_thread = threading.Thread(
                target=webhook_server.run, name="webhook-server", daemon=True
            )
            webhook_thread.start()
            log.info(
                "admission webhook serving on %s (%s)",
                args.webhook_bind_address,
                "TLS" if have_tls else "INSECURE plain HTTP",
            )
        else:
            log.warning(
                "ENABLE_WEBHOOKS is on but %s has no tls.crt/tls.key; "
                "NOT serving the AdmissionReview endpoint — do not apply "
                "config/webhook/manifests.yaml without certificates",
                args.webhook_cert_dir,
            )

    # -- dedicated authenticated metrics listener ---------------------------
    # The reference serves :8443 metrics over HTTPS behind authn/authz
    # (cmd/main.go:109-127); here: TLS from --metrics-cert-dir + the bearer
    # token the ServiceMonitor sends from the cro-amd-tokens Secret.
    metrics_server = None
    if args.metrics_bind_address:
        from ..server.metrics import MetricsServer

        metrics_server = MetricsServer(
            parse_port(args.metrics_bind_address, 8443),
            token=metrics_token if args.metrics_secure else "",
            cert_dir=args.metrics_cert_dir,
            certfile=args.tls_cert_file or None,
            keyfile=args.tls_key_file or None,
        )
        metrics_server.start()
        log.info(
            "metrics serving on %s (%s, %s)",
            args.metrics_bind_address,
            "TLS" if metrics_server.tls else "plain HTTP",
            "bearer-token" if (metrics_token and args.metrics_secure) else "open",
        )

    # -- kubelet health-probe listener (manager.yaml probes on :8081) -------
    health_server = None
    if args.health_probe_bind_address:
        from ..server.metrics import HealthServer

        health_server = HealthServer(
            parse_port(args.health_probe_bind_address, 8081))
        health_server.start()

    # -- leader election + controller start ---------------------------------
    elector = None
    if not args.serve_only:
        def start_controllers():
            mgr.start()
            log.info("manager started (%d reconcile workers per controller)",
                     args.max_concurrent_reconciles)

        if args.leader_elect:
            from ..runtime.lease import LeaderElector

            import socket as _socket

            def lost_leadership():
                # fatal, as in controller-runtime: never run controllers
                # concurrently with the new leader
                log.error("leadership lost; shutting down")
                server.should_exit = True
                stop.set()

            elector = LeaderElector(
                mgr.client,
                identity=f"{_socket.gethostname()}_{os.getpid()}",
                lease_duration=args.leader_lease_duration,
                renew_deadline=args.leader_renew_deadline,
                retry_period=args.leader_retry_period,
                on_started_leading=start_controllers,
                on_stopped_leading=lost_leadership,
            )
            elector.start()
            log.info("waiting for leader lease as %s", elector.identity)
        else:
            start_controllers()
    else:
        log.info("serve-only mode: API/store up, controllers disabled")

    server.run()  # serves API + healthz/readyz/metrics until signal
    if elector is not None:
        elector.stop()  # releases the lease for fast standby takeover
    if webhook_server is not None:
        webhook_server.should_exit = True
        if webhook_thread is not None:
            webhook_thread.join(timeout=5)
    if metrics_server is not None:
        metrics_server.stop()
    if health_server is not None:
        health_server.stop()
    mgr.stop()
    if store is not None:
        store.close()  # final flush of the durable snapshot
    return 0


if __name__ == "__main__":
    sys.exit(main())
