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


def _merge_patch(target, patch):
    """RFC 7386 JSON Merge Patch: dicts merge recursively, ``null`` deletes
    a key, everything else replaces."""
    if not isinstance(patch, dict):
        return patch
    if not isinstance(target, dict):
        target = {}
    out = dict(target)
    for key, value in patch.items():
        if value is None:
            out.pop(key, None)
        else:
            out[key] = _merge_patch(out.get(key), value)
    return out


def _status_body(code: int, reason: str, message: str) -> dict:
    """metav1.Status — the exact error body a kube-apiserver returns, so
    any k8s client library can consume this server's errors."""
    return {
        "kind": "Status",
        "apiVersion": "v1",
        "metadata": {},
        "status": "Failure",
        "message": message,
        "reason": reason,
        "code": code,
    }


def _error_response(exc: Exception):
    """Error taxonomy over HTTP as top-level metav1.Status objects (the
    kube-apiserver wire shape; 409 covers both AlreadyExists and
    optimistic-concurrency Conflict, exactly as in the k8s API)."""
    from fastapi.responses import JSONResponse

    if isinstance(exc, NotFoundError):
        body = _status_body(404, "NotFound", str(exc))
    elif isinstance(exc, AlreadyExistsError):
        body = _status_body(409, "AlreadyExists", str(exc))
    elif isinstance(exc, ConflictError):
        body = _status_body(409, "Conflict", str(exc))
    elif isinstance(exc, AdmissionDenied):
        body = _status_body(403, "Forbidden", str(exc))
    elif isinstance(exc, ValueError):
        body = _status_body(422, "Invalid", str(exc))
    else:
        body = _status_body(500, "InternalError", str(exc))
    return JSONResponse(status_code=body["code"], content=body)


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

    def _current_rv() -> int:
        store = getattr(client, "store", None)
        return store.current_seq() if store is not None else 0

    async def list_objects(
        plural: str,
        labelSelector: str = "",
        fieldSelector: str = "",
        watch: bool = False,
        resourceVersion: str = "",
        allowWatchBookmarks: bool = False,
        sendInitialEvents: bool = False,
        timeoutSeconds: float = 0,
    ):
        cls = _cls(plural)
        if watch:
            return await _watch_stream(
                cls, resourceVersion, allowWatchBookmarks,
                sendInitialEvents, timeoutSeconds,
            )
        labels = None
        if labelSelector:
            labels = dict(part.split("=", 1) for part in labelSelector.split(","))
        items = client.list(cls, labels)
        if fieldSelector:
            # the metadata.name form kubectl/client-go use (equality only,
            # matching the k8s supported fields for custom resources)
            for part in fieldSelector.split(","):
                key, _, value = part.partition("=")
                if key == "metadata.name":
                    items = [o for o in items if o.metadata.name == value]
                else:
                    return _error_response(ValueError(
                        f"unsupported fieldSelector {key!r} (metadata.name only)"))
        # ListMeta.resourceVersion is the watch-resume token for the
        # list-then-watch informer protocol (apiserver parity)
        return {
            "apiVersion": cls.model_fields["apiVersion"].default,
            "kind": cls.KIND + "List",
            "metadata": {"resourceVersion": str(_current_rv())},
            "items": [_dump(o) for o in items],
        }

    async def _watch_stream(
        cls,
        resource_version: str = "",
        bookmarks: bool = False,
        send_initial: bool = False,
        timeout_seconds: float = 0,
    ):
        """kube-apiserver watch semantics, one JSON WatchEvent per line:

        * ``?resourceVersion=N``        → stream events after N; an aged-out
          N gets ONE ``{"type":"ERROR","object":metav1.Status(410 Expired)}``
          line and the stream ends — the client re-lists (the real 410-Gone
          contract; no in-stream re-list).
        * no rv / ``resourceVersion=0`` → stream from "now" (clients that
          want state do list-then-watch, as informers do).
        * ``sendInitialEvents=true``    → the 1.27+ WatchList protocol:
          synthetic ADDED per current object, then a BOOKMARK annotated
          ``k8s.io/initial-events-end`` at the snapshot rv, then live events.
        * ``allowWatchBookmarks=true``  → periodic BOOKMARK events carry the
          current rv so idle clients keep a fresh resume token.
        * every event object's ``metadata.resourceVersion`` IS the resume
          token (object RVs and watch tokens share one counting space).
        """
        import asyncio
        import json as _json
        import queue as _queue

        from fastapi.responses import StreamingResponse

        # absent rv → watch from "now" (k8s watch default). A PRESENT rv —
        # including "0", which a fresh store's list legitimately returns —
        # resumes from the event log, so a client that lists an empty store
        # and watches from its rv never loses events created in the gap
        # (k8s "0" means "any version"; exact replay satisfies it).
        try:
            rv = int(resource_version) if resource_version != "" else None
        except ValueError:
            rv = None

        def bookmark_line(at_rv: int, initial_end: bool = False) -> str:
            obj = {
                "kind": cls.KIND,
                "apiVersion": cls.model_fields["apiVersion"].default,
                "metadata": {"resourceVersion": str(at_rv)},
            }
            if initial_end:
                obj["metadata"]["annotations"] = {"k8s.io/initial-events-end": "true"}
            return _json.dumps({"type": "BOOKMARK", "object": obj}) + "\n"

        async def gen():
            import os
            import time as _time

            # k8s watch-timeout contract: every stream is closed server-side
            # after a bounded lifetime and the client reconnects (cheap with
            # rv resume tokens). Bounds the damage of half-open connections.
            lifetime = timeout_seconds or float(os.environ.get("CRO_WATCH_TIMEOUT", "300"))
            stream_deadline = _time.monotonic() + lifetime
            events = client.watch([cls.KIND])
            store = getattr(client, "store", None)
            last_seq = 0
            try:
                if send_initial:
                    snapshot_rv = _current_rv()
                    for obj in client.list(cls):
                        yield _json.dumps(
                            {"type": "ADDED", "object": _dump(obj)}
                        ) + "\n"
                    yield bookmark_line(snapshot_rv, initial_end=True)
                    last_seq = snapshot_rv
                elif rv is not None and store is not None:
                    buffered = store.events_since(rv, [cls.KIND])
                    if buffered is None:
                        # aged-out token → 410 Expired, stream ends
                        yield _json.dumps({
                            "type": "ERROR",
                            "object": _status_body(
                                410, "Expired",
                                f"too old resource version: {rv}"),
                        }) + "\n"
                        return
                    for ev in buffered:
                        last_seq = ev.seq
                        yield _json.dumps(
                            {"type": ev.type, "object": _dump(ev.object)}
                        ) + "\n"
                    last_seq = max(last_seq, rv)
                else:
                    last_seq = _current_rv()  # watch from "now"
                loop = asyncio.get_running_loop()
                while True:
                    if _time.monotonic() >= stream_deadline:
                        return  # watch timeout; client resumes by rv token
                    try:
                        ev = await loop.run_in_executor(None, events.get, True, 1.0)
                    except _queue.Empty:
                        if bookmarks:
                            yield bookmark_line(max(last_seq, _current_rv()))
                        else:
                            yield "\n"  # legacy keepalive
                        continue
                    except RuntimeError:
                        return  # event loop / executor shutting down
                    if ev.seq and ev.seq <= last_seq:
                        continue  # already served from the resume buffer
                    last_seq = ev.seq or last_seq
                    yield _json.dumps(
                        {"type": ev.type, "object": _dump(ev.object)}
                    ) + "\n"
            finally:
                # disconnects must release the watcher or every later event
                # fans out to dead queues forever
                store = getattr(client, "store", None)
                if store is not None and hasattr(store, "stop_watch"):
                    store.stop_watch(events)

        return StreamingResponse(gen(), media_type="application/json")

    def get_object(plural: str, name: str):
        cls = _cls(plural)
        try:
            return _dump(client.get(cls, name))
        except Exception as exc:
            return _error_response(exc)

    async def create_object(plural: str, request: Request):
        cls = _cls(plural)
        try:
            obj = cls.model_validate(await request.json())
            return _dump(client.create(obj))
        except Exception as exc:
            return _error_response(exc)

    async def update_object(plural: str, name: str, request: Request):
        cls = _cls(plural)
        if name.endswith("/status"):
            name = name[: -len("/status")]
            try:
                obj = cls.model_validate(await request.json())
                obj.metadata.name = name
                return _dump(client.update_status(obj))
            except Exception as exc:
                return _error_response(exc)
        try:
            obj = cls.model_validate(await request.json())
            obj.metadata.name = name
            return _dump(client.update(obj))
        except Exception as exc:
            return _error_response(exc)

    async def patch_object(plural: str, name: str, request: Request):
        """RFC 7386 JSON Merge Patch (``Content-Type:
        application/merge-patch+json`` — what ``kubectl patch`` sends by
        default for custom resources): fetch, merge (null deletes), write
        back through the normal admission/validation chain.  The
        read-merge-write carries the stored resourceVersion and retries
        briefly on concurrent-writer conflicts, as the apiserver does."""
        cls = _cls(plural)
        ctype = request.headers.get("content-type", "")
        if "merge-patch" not in ctype and "application/json" not in ctype:
            return _error_response(ValueError(
                f"unsupported patch content-type {ctype!r} "
                "(application/merge-patch+json)"))
        try:
            patch = await request.json()
        except Exception as exc:
            return _error_response(ValueError(f"unreadable patch body: {exc}"))
        status_sub = name.endswith("/status")
        if status_sub:
            name = name[: -len("/status")]
        last: Exception = None
        for _ in range(8):
            try:
                current = client.get(cls, name)
                merged = _merge_patch(current.model_dump(by_alias=True), patch)
                obj = cls.model_validate(merged)
                obj.metadata.name = name
                # carry the rv we read so a concurrent writer conflicts
                obj.metadata.resourceVersion = current.metadata.resourceVersion
                if status_sub:
                    return _dump(client.update_status(obj))
                return _dump(client.update(obj))
            except ConflictError as exc:
                last = exc
            except Exception as exc:
                return _error_response(exc)
        return _error_response(last)

    def delete_object(plural: str, name: str):
        cls = _cls(plural)
        try:
            client.delete(cls, name)
        except Exception as exc:
            return _error_response(exc)
        # metav1.Status success body, as the apiserver returns for deletes
        return {
            "kind": "Status", "apiVersion": "v1", "metadata": {},
            "status": "Success",
            "details": {"name": name, "kind": plural},
        }

    # Register the handlers under every API group the kinds belong to:
    # the home CRD group plus the REAL k8s groups for the native kinds,
    # so k8s tooling addresses ResourceSlices/DeviceTaintRules/Leases at
    # their canonical paths (resource.k8s.io is what the reference writes,
    # internal/utils/gpus.go:894-989).
    for base in (
        BASE,
        "/apis/resource.k8s.io/v1alpha3",
        "/apis/coordination.k8s.io/v1",
    ):
        app.get(base + "/{plural}")(list_objects)
        app.get(base + "/{plural}/{name:path}")(get_object)
        app.post(base + "/{plural}", status_code=201)(create_object)
        app.put(base + "/{plural}/{name:path}")(update_object)
        app.patch(base + "/{plural}/{name:path}")(patch_object)
        app.delete(base + "/{plural}/{name:path}")(delete_object)

    return app
