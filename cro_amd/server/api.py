"""REST API surface for the standalone operator.

In cluster mode the kube-apiserver owns the CRD objects; in standalone mode
(no Kubernetes) this server IS the API: apiserver-shaped routes over the
in-process store, so tooling can drive the operator the same way in both
modes:

    GET/POST      /apis/cro.hpsys.ibm.ie.com/v1alpha1/{plural}
    GET/PUT/DELETE /apis/cro.hpsys.ibm.ie.com/v1alpha1/{plural}/{name}
    PUT           /apis/cro.hpsys.ibm.ie.com/v1alpha1/{plural}/{name}/status
    GET           /apis/cro.hpsys.ibm.ie.com/v1alpha1/nodes        (+ PUT)
    GET           /metrics, /healthz, /readyz

Admission (webhook rules + schema validation) runs on every write through
the store's admission chain, exactly as in-process reconciles see it.
"""

from __future__ import annotations

from fastapi import FastAPI, HTTPException, Request, Response

from .. import API_VERSION
from ..api.v1alpha1.types import (
    BareMetalHost,
    ComposabilityRequest,
    ComposableResource,
    DaemonSet,
    DeviceConfig,
    DeviceTaintRule,
    Event,
    Lease,
    Machine,
    Node,
    ResourceSlice,
)
from ..runtime.client import Client
from ..runtime.errors import (
    AdmissionDenied,
    AlreadyExistsError,
    ConflictError,
    NotFoundError,
)

PLURALS = {
    "composabilityrequests": ComposabilityRequest,
    "composableresources": ComposableResource,
    "resourceslices": ResourceSlice,
    "devicetaintrules": DeviceTaintRule,
    "nodes": Node,
    "events": Event,
    "leases": Lease,
    "machines": Machine,
    "baremetalhosts": BareMetalHost,
    "deviceconfigs": DeviceConfig,
    "daemonsets": DaemonSet,
}

BASE = "/apis/cro.hpsys.ibm.ie.com/v1alpha1"


def _cls(plural: str):
    cls = PLURALS.get(plural)
    if cls is None:
        raise HTTPException(404, f"unknown resource {plural!r}")
    return cls


def _dump(obj) -> dict:
    return obj.model_dump(by_alias=True)


def _http_error(exc: Exception) -> HTTPException:
    """Error taxonomy over HTTP: the reason field lets a remote client map
    back to the exact error class (409 covers both AlreadyExists and
    optimistic-concurrency Conflict, as in the k8s API)."""
    if isinstance(exc, NotFoundError):
        return HTTPException(404, {"reason": "NotFound", "message": str(exc)})
    if isinstance(exc, AlreadyExistsError):
        return HTTPException(409, {"reason": "AlreadyExists", "message": str(exc)})
    if isinstance(exc, ConflictError):
        return HTTPException(409, {"reason": "Conflict", "message": str(exc)})
    if isinstance(exc, AdmissionDenied):
        return HTTPException(403, {"reason": "Forbidden", "message": str(exc)})
    if isinstance(exc, ValueError):
        return HTTPException(422, {"reason": "Invalid", "message": str(exc)})
    return HTTPException(500, {"reason": "InternalError", "message": str(exc)})


def build_app(client: Client, token: str = None) -> FastAPI:
    """apiserver-shaped app over ``client``.

    ``token`` (default: env ``CRO_API_TOKEN``) gates every ``/apis`` route
    with ``Authorization: Bearer`` — the standalone analog of the
    kube-apiserver's authn in front of the CRD API. The production
    entrypoint auto-generates one when unset (cmd/main.py), so a default
    split deployment is never an open write surface to PCI remove/rescan
    (--destructive). Health and readiness stay unauthenticated, matching
    kubelet probe semantics; /metrics has its own token (below).
    """
    import hmac as _hmac
    import os as _os

    if token is None:
        token = _os.environ.get("CRO_API_TOKEN", "")

    app = FastAPI(title="cro-amd API", version=API_VERSION)

    if token:
        @app.middleware("http")
        async def _api_auth(request: Request, call_next):
            if request.url.path.startswith("/apis"):
                auth = request.headers.get("authorization", "")
                if not _hmac.compare_digest(auth, f"Bearer {token}"):
                    from fastapi.responses import JSONResponse

                    return JSONResponse(
                        status_code=401,
                        content={
                            "kind": "Status",
                            "apiVersion": "v1",
                            "status": "Failure",
                            "reason": "Unauthorized",
                            "message": "API requires a valid bearer token",
                            "code": 401,
                        },
                    )
            return await call_next(request)

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    @app.get("/readyz")
    def readyz():
        return {"status": "ok"}

    @app.get("/metrics")
    def metrics(request: Request):
        """Prometheus endpoint; CRO_METRICS_TOKEN enables bearer-token
        authentication (the reference serves metrics behind an
        authn/authz filter, cmd/main.go:109-127)."""
        import os

        import prometheus_client

        token = os.environ.get("CRO_METRICS_TOKEN", "")
        if token:
            auth = request.headers.get("authorization", "")
            if auth != f"Bearer {token}":
                raise HTTPException(401, "metrics require a valid bearer token")
        return Response(
            prometheus_client.generate_latest(),
            media_type=prometheus_client.CONTENT_TYPE_LATEST,
        )

    @app.get(BASE + "/{plural}")
    async def list_objects(
        plural: str,
        labelSelector: str = "",
        watch: bool = False,
        resourceVersion: int = -1,
    ):
        cls = _cls(plural)
        if watch:
            return await _watch_stream(cls, resourceVersion)
        labels = None
        if labelSelector:
            labels = dict(part.split("=", 1) for part in labelSelector.split(","))
        items = client.list(cls, labels)
        return {
            "apiVersion": API_VERSION,
            "kind": cls.KIND + "List",
            "items": [_dump(o) for o in items],
        }

    async def _watch_stream(cls, resource_version: int = -1):
        """k8s-style list+watch, newline-delimited JSON. Every event line
        carries ``rv`` (the store's event sequence) as a resume token:
        reconnecting with ``?resourceVersion=<rv>`` replays only missed
        events from the bounded watch-cache; an aged-out token gets one
        ``{"type": "ERROR", "reason": "Expired"}`` line followed by the
        full list replay (the apiserver's 410-Gone contract, in-stream)."""
        import asyncio
        import json as _json
        import queue as _queue

        from fastapi.responses import StreamingResponse

        async def gen():
            import os
            import time as _time

            # k8s watch-timeout contract: every stream is closed server-side
            # after a bounded lifetime and the client reconnects (cheap with
            # rv resume tokens). Bounds the damage of half-open connections
            # — a client reading keepalives from a stale server otherwise
            # never notices it should reconnect.
            lifetime = float(os.environ.get("CRO_WATCH_TIMEOUT", "300"))
            stream_deadline = _time.monotonic() + lifetime
            events = client.watch([cls.KIND])
            store = getattr(client, "store", None)
            last_seq = 0
            try:
                buffered = None
                if resource_version >= 0 and store is not None:
                    buffered = store.events_since(resource_version, [cls.KIND])
                if buffered is not None:
                    for ev in buffered:
                        last_seq = ev.seq
                        yield _json.dumps(
                            {"type": ev.type, "object": _dump(ev.object), "rv": ev.seq}
                        ) + "\n"
                else:
                    if resource_version >= 0:
                        yield _json.dumps({"type": "ERROR", "reason": "Expired"}) + "\n"
                    snapshot_rv = store.current_seq() if store is not None else 0
                    for obj in client.list(cls):
                        yield _json.dumps(
                            {"type": "ADDED", "object": _dump(obj), "rv": snapshot_rv}
                        ) + "\n"
                loop = asyncio.get_running_loop()
                while True:
                    if _time.monotonic() >= stream_deadline:
                        return  # watch timeout; client resumes by rv token
                    try:
                        ev = await loop.run_in_executor(None, events.get, True, 1.0)
                    except _queue.Empty:
                        yield "\n"  # keepalive; also surfaces disconnects
                        continue
                    except RuntimeError:
                        return  # event loop / executor shutting down
                    if ev.seq and ev.seq <= last_seq:
                        continue  # already served from the resume buffer
                    last_seq = ev.seq or last_seq
                    yield _json.dumps(
                        {"type": ev.type, "object": _dump(ev.object), "rv": ev.seq}
                    ) + "\n"
            finally:
                # disconnects must release the watcher or every later event
                # fans out to dead queues forever
                store = getattr(client, "store", None)
                if store is not None and hasattr(store, "stop_watch"):
                    store.stop_watch(events)

        return StreamingResponse(gen(), media_type="application/x-ndjson")

    @app.get(BASE + "/{plural}/{name:path}")
    def get_object(plural: str, name: str):
        cls = _cls(plural)
        try:
            return _dump(client.get(cls, name))
        except Exception as exc:
            raise _http_error(exc)

    @app.post(BASE + "/{plural}", status_code=201)
    async def create_object(plural: str, request: Request):
        cls = _cls(plural)
        try:
            obj = cls.model_validate(await request.json())
            return _dump(client.create(obj))
        except Exception as exc:
            raise _http_error(exc)

    @app.put(BASE + "/{plural}/{name:path}")
    async def update_object(plural: str, name: str, request: Request):
        cls = _cls(plural)
        if name.endswith("/status"):
            name = name[: -len("/status")]
            try:
                obj = cls.model_validate(await request.json())
                obj.metadata.name = name
                return _dump(client.update_status(obj))
            except Exception as exc:
                raise _http_error(exc)
        try:
            obj = cls.model_validate(await request.json())
            obj.metadata.name = name
            return _dump(client.update(obj))
        except Exception as exc:
            raise _http_error(exc)

    @app.delete(BASE + "/{plural}/{name:path}", status_code=202)
    def delete_object(plural: str, name: str):
        cls = _cls(plural)
        try:
            client.delete(cls, name)
        except Exception as exc:
            raise _http_error(exc)
        return {"status": "deleted"}

    return app
