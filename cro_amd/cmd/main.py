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

Leader election uses an exclusive flock on a lock file — the single-node
analog of the reference's lease-based election (cmd/main.go:137-155); in a
multi-replica cluster deployment the lease path belongs to the kube client
integration.
"""

from __future__ import annotations

import argparse
import fcntl
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
    p.add_argument("--leader-elect", action="store_true")
    p.add_argument("--leader-elect-lock", default="/var/run/cro-amd/leader.lock")
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
    log = logging.getLogger("cro_amd.main")

    lock_file = None
    if args.leader_elect:
        os.makedirs(os.path.dirname(args.leader_elect_lock), exist_ok=True)
        lock_file = open(args.leader_elect_lock, "w")
        log.info("waiting for leader lock %s", args.leader_elect_lock)
        fcntl.flock(lock_file, fcntl.LOCK_EX)
        log.info("acquired leadership")

    from ..controllers import build_manager
    from ..fabric.adapter import new_adapter
    from ..nodeops.amdgpu import AmdNodeOps
    from ..nodeops.execs import LocalNodeExec

    adapter = new_adapter()

    remote = None
    if args.api_server:
        from ..runtime.remote import RemoteClient

        remote = RemoteClient(args.api_server)
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
        metrics_port=parse_port(args.metrics_bind_address, 8443),
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

    if not args.serve_only:
        mgr.start()
        log.info("manager started (%d reconcile workers per controller)",
                 args.max_concurrent_reconciles)
    else:
        log.info("serve-only mode: API/store up, controllers disabled")

    from ..server.api import build_app
    from ..server.agent_api import build_agent_app

    import uvicorn

    app = build_app(mgr.client)
    if args.node:  # node-agent surface for off-node controllers
        build_agent_app(node_ops.execer, node_name=args.node, app=app)
    server = uvicorn.Server(
        uvicorn.Config(
            app,
            host="0.0.0.0",
            port=parse_port(args.api_bind_address, 8080),
            log_level="warning",
            ssl_certfile=args.tls_cert_file or None,
            ssl_keyfile=args.tls_key_file or None,
        )
    )

    stop = threading.Event()

    def handle_signal(signum, frame):
        log.info("signal %s; shutting down", signum)
        server.should_exit = True
        stop.set()

    signal.signal(signal.SIGTERM, handle_signal)
    signal.signal(signal.SIGINT, handle_signal)

    server.run()  # serves API + healthz/readyz/metrics until signal
    mgr.stop()
    if store is not None:
        store.close()  # final flush of the durable snapshot
    if lock_file is not None:
        lock_file.close()
    return 0


if __name__ == "__main__":
    sys.exit(main())
