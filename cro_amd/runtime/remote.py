"""RemoteClient: the Client surface over an apiserver-shaped HTTP API.

Lets the operator process run against a remote API server (the cluster
deployment shape — controllers on the control plane, node agents on nodes)
with the same Client/watch surface the in-memory store provides, so every
controller runs unchanged in either mode.

Watch uses k8s-style list+watch streaming (ndjson); each watched kind gets
a reader thread feeding the subscriber queue with the same WatchEvent shape
controllers consume.  Admission and schema validation run server-side.
"""

from __future__ import annotations

import json
import logging
import queue
import threading
from typing import Dict, List, Optional, Type, TypeVar, Union

import httpx

from ..api.v1alpha1.types import ALL_KINDS, K8sObject
from .errors import (
    AdmissionDenied,
    AlreadyExistsError,
    ApiError,
    ConflictError,
    NotFoundError,
)
from .store import WatchEvent

log = logging.getLogger(__name__)

BASE = "/apis/cro.hpsys.ibm.ie.com/v1alpha1"
T = TypeVar("T", bound=K8sObject)

_PLURALS = {cls.KIND: plural for plural, cls in {
    "composabilityrequests": ALL_KINDS["ComposabilityRequest"],
    "composableresources": ALL_KINDS["ComposableResource"],
    "resourceslices": ALL_KINDS["ResourceSlice"],
    "devicetaintrules": ALL_KINDS["DeviceTaintRule"],
    "nodes": ALL_KINDS["Node"],
    "events": ALL_KINDS["Event"],
    "leases": ALL_KINDS["Lease"],
    "machines": ALL_KINDS["Machine"],
    "baremetalhosts": ALL_KINDS["BareMetalHost"],
    "deviceconfigs": ALL_KINDS["DeviceConfig"],
    "daemonsets": ALL_KINDS["DaemonSet"],
}.items()}


def _raise_for(resp: httpx.Response) -> None:
    if 200 <= resp.status_code < 300:
        return
    # error bodies are top-level metav1.Status objects (kube-apiserver wire
    # shape); tolerate FastAPI's {"detail": ...} wrapper for foreign servers
    reason, message = "", resp.text
    try:
        body = resp.json()
        if isinstance(body, dict):
            if body.get("kind") == "Status" or "reason" in body:
                reason = body.get("reason", "")
                message = body.get("message", resp.text)
            else:
                detail = body.get("detail", {})
                if isinstance(detail, dict):
                    reason = detail.get("reason", "")
                    message = detail.get("message", str(detail))
                else:
                    message = str(detail)
    except ValueError:
        pass
    if resp.status_code == 404 or reason == "NotFound":
        raise NotFoundError(message)
    if reason == "AlreadyExists":
        raise AlreadyExistsError(message)
    if reason == "Conflict" or resp.status_code == 409:
        raise ConflictError(message)
    if resp.status_code == 403:
        raise AdmissionDenied(message)
    if resp.status_code == 422:
        raise ValueError(message)
    raise ApiError(f"{resp.status_code}: {message}")


class RemoteClient:
    """HTTP Client over the apiserver-shaped REST API.

    The URL scheme is the kube-apiserver scheme for cluster-scoped CRDs
    (``/apis/cro.hpsys.ibm.ie.com/v1alpha1/<plural>[/<name>[/status]]``),
    so the same client reaches a real kube-apiserver given credentials:
    ``token`` (or ``CRO_API_TOKEN``) sends ``Authorization: Bearer`` —
    the in-cluster ServiceAccount token — and ``ca`` (or ``CRO_API_CA``)
    verifies the serving cert (``verify=False`` to skip)."""

    def __init__(
        self,
        base_url: str,
        transport: Optional[httpx.BaseTransport] = None,
        token: Optional[str] = None,
        ca: Optional[str] = None,
        verify: Union[bool, str, None] = None,
        cache: bool = False,
    ):
        """``cache=True`` enables the client-go SharedInformer read path:
        one list+watch informer per kind maintains a local object cache;
        ``get``/``list`` are served from it once synced (a reconcile then
        costs zero read RTTs, exactly as controller-runtime reads from its
        cache), writes go to the server and update the cache read-your-
        writes-style. Reads may be marginally stale — level-triggered
        reconciles + optimistic concurrency absorb that, as in k8s."""
        import os

        self.base_url = base_url.rstrip("/")
        if token is None:
            token = os.environ.get("CRO_API_TOKEN", "")
        if ca is None:
            ca = os.environ.get("CRO_API_CA", "")
        if verify is None:
            verify = ca if ca else True
        headers = {"Authorization": f"Bearer {token}"} if token else None
        self._http = httpx.Client(
            base_url=self.base_url, transport=transport, timeout=30,
            headers=headers, verify=verify,
        )
        self._stop = threading.Event()
        self._watch_threads: List[threading.Thread] = []
        self._cache_enabled = cache
        self._informers: Dict[str, "_Informer"] = {}
        self._informer_lock = threading.Lock()

    # -- informer cache ----------------------------------------------------

    def _informer(self, kind: str) -> "_Informer":
        with self._informer_lock:
            inf = self._informers.get(kind)
            if inf is None:
                inf = _Informer(self, kind)
                self._informers[kind] = inf
                inf.start()
            return inf

    # -- kind plumbing -----------------------------------------------------

    @staticmethod
    def _resolve(cls_or_kind: Union[Type[T], str]):
        if isinstance(cls_or_kind, str):
            cls = ALL_KINDS[cls_or_kind]
        else:
            cls = cls_or_kind
        return cls, _PLURALS[cls.KIND]

    # -- Client surface ----------------------------------------------------

    def create(self, obj: T) -> T:
        cls, plural = self._resolve(type(obj))
        resp = self._http.post(f"{BASE}/{plural}", json=obj.model_dump(by_alias=True))
        _raise_for(resp)
        created = cls.model_validate(resp.json())
        self._offer_cache(created)
        return created

    def _offer_cache(self, obj) -> None:
        """Read-your-writes: fold a write response into the informer cache
        (kept only if newer than the cached rv)."""
        if not self._cache_enabled:
            return
        inf = self._informers.get(obj.kind)
        if inf is not None:
            inf.offer(obj)

    def get(self, cls_or_kind, name: str):
        cls, plural = self._resolve(cls_or_kind)
        if self._cache_enabled:
            inf = self._informer(cls.KIND)
            if inf.synced.is_set():
                cached = inf.get_cached(name)
                if cached is None:
                    raise NotFoundError(f"{cls.KIND}/{name} not found")
                return cached
        resp = self._http.get(f"{BASE}/{plural}/{name}")
        _raise_for(resp)
        return cls.model_validate(resp.json())

    def try_get(self, cls_or_kind, name: str):
        try:
            return self.get(cls_or_kind, name)
        except NotFoundError:
            return None

    def list(self, cls_or_kind, labels: Optional[Dict[str, str]] = None, copy: bool = True):
        cls, plural = self._resolve(cls_or_kind)
        if self._cache_enabled:
            inf = self._informer(cls.KIND)
            if inf.synced.is_set():
                return inf.list_cached(labels)
        params = {}
        if labels:
            params["labelSelector"] = ",".join(f"{k}={v}" for k, v in labels.items())
        resp = self._http.get(f"{BASE}/{plural}", params=params)
        _raise_for(resp)
        return [cls.model_validate(item) for item in resp.json()["items"]]

    def update(self, obj: T) -> T:
        cls, plural = self._resolve(type(obj))
        resp = self._http.put(
            f"{BASE}/{plural}/{obj.metadata.name}", json=obj.model_dump(by_alias=True)
        )
        _raise_for(resp)
        updated = cls.model_validate(resp.json())
        self._offer_cache(updated)
        return updated

    def update_status(self, obj: T) -> T:
        cls, plural = self._resolve(type(obj))
        resp = self._http.put(
            f"{BASE}/{plural}/{obj.metadata.name}/status",
            json=obj.model_dump(by_alias=True),
        )
        _raise_for(resp)
        updated = cls.model_validate(resp.json())
        self._offer_cache(updated)
        return updated

    def delete(self, obj_or_cls, name: Optional[str] = None) -> None:
        if name is None:
            cls, plural = self._resolve(type(obj_or_cls))
            name = obj_or_cls.metadata.name
        else:
            cls, plural = self._resolve(obj_or_cls)
        resp = self._http.delete(f"{BASE}/{plural}/{name}")
        _raise_for(resp)

    # -- watch (Controller.start_watch consumes this) ----------------------

    def watch(self, kinds: Optional[List[str]] = None) -> "queue.Queue[WatchEvent]":
        q: "queue.Queue[WatchEvent]" = queue.Queue()
        for kind in kinds or list(_PLURALS):
            if self._cache_enabled:
                # SharedInformer shape: one stream per kind, subscribers
                # get a cache replay + the shared live feed
                self._informer(kind).subscribe(q)
            else:
                t = threading.Thread(
                    target=self._watch_kind, args=(kind, q),
                    name=f"remote-watch-{kind}", daemon=True,
                )
                t.start()
                self._watch_threads.append(t)
        return q

    def _watch_kind(self, kind: str, q, on_event=None, on_synced=None) -> None:
        """The informer protocol, exactly as client-go runs it against a
        kube-apiserver: LIST (take ListMeta.resourceVersion) → synthesize
        ADDED for the current objects (cache replay) → WATCH from that rv
        with bookmarks → on 410 Expired (ERROR event) or an aged-out
        reconnect, re-list.  Resume tokens are object resourceVersions —
        there is no out-of-band framing field.

        ``on_event(WatchEvent)`` (default: ``q.put``) receives every event;
        ``on_synced(names)`` fires after each initial list completes with
        the set of listed names — a cache re-syncing on reconnect prunes
        entries deleted during the disconnect (client-go re-list sync)."""
        deliver = on_event if on_event is not None else q.put
        cls, plural = self._resolve(kind)
        last_rv = -1
        while not self._stop.is_set():
            try:
                if last_rv < 0:
                    resp = self._http.get(f"{BASE}/{plural}")
                    _raise_for(resp)
                    body = resp.json()
                    last_rv = int(body.get("metadata", {}).get("resourceVersion", 0))
                    listed = set()
                    for item in body["items"]:
                        obj = cls.model_validate(item)
                        listed.add(obj.metadata.name)
                        deliver(WatchEvent("ADDED", obj))
                    if on_synced is not None:
                        on_synced(listed)
                params = {
                    "watch": "true",
                    "resourceVersion": str(last_rv),
                    "allowWatchBookmarks": "true",
                }
                with self._http.stream(
                    "GET", f"{BASE}/{plural}", params=params, timeout=None
                ) as resp:
                    for line in resp.iter_lines():
                        if self._stop.is_set():
                            return
                        if not line.strip():
                            continue  # keepalive
                        ev = json.loads(line)
                        if ev["type"] == "ERROR":
                            # 410 Expired → full re-list on next loop turn
                            last_rv = -1
                            break
                        obj_rv = (
                            ev.get("object", {})
                            .get("metadata", {})
                            .get("resourceVersion", "")
                        )
                        if obj_rv:
                            last_rv = int(obj_rv)
                        if ev["type"] == "BOOKMARK":
                            continue  # resume-token refresh only
                        deliver(WatchEvent(ev["type"], cls.model_validate(ev["object"])))
            except Exception as exc:
                if self._stop.is_set():
                    return
                log.debug("watch %s disconnected (%s); reconnecting", kind, exc)
                self._stop.wait(0.2)

    def stop_watch(self, q) -> None:
        """Unsubscribe a queue produced by watch() (cache mode): stopped
        controllers must not keep receiving fan-out into dead queues.
        Legacy (uncached) watch threads exit with close() instead."""
        for inf in self._informers.values():
            inf.unsubscribe(q)

    def close(self) -> None:
        self._stop.set()
        self._http.close()


class _Informer:
    """client-go SharedInformer analog: ONE list+watch stream per kind
    feeding a local object cache plus any number of subscriber queues.
    Reads from the cache cost zero RTTs; subscribers joining later get a
    replay of the current cache before live events."""

    def __init__(self, client: "RemoteClient", kind: str):
        self.client = client
        self.kind = kind
        self.synced = threading.Event()
        self._lock = threading.Lock()
        self._store: Dict[str, object] = {}
        # deletion tombstones: name -> rv of the DELETED event. A write
        # RESPONSE being folded in (offer) can race the watch thread's
        # DELETED and resurrect a dead object without this (observed as a
        # Cleaning loop forever deleting a phantom child).
        self._tombstones: Dict[str, int] = {}
        self._subs: List["queue.Queue[WatchEvent]"] = []
        self._thread: Optional[threading.Thread] = None

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self.client._watch_kind,
            args=(self.kind, None),
            kwargs={"on_event": self._apply, "on_synced": self._sync_complete},
            name=f"informer-{self.kind}",
            daemon=True,
        )
        self._thread.start()
        self.client._watch_threads.append(self._thread)

    def _sync_complete(self, listed_names) -> None:
        """End of an initial list (first sync OR a 410 re-list): any cached
        object absent from the fresh list was deleted while disconnected —
        prune it and deliver the synthetic DELETED client-go would."""
        stale_events = []
        with self._lock:
            for name in list(self._store):
                if name not in listed_names:
                    obj = self._store.pop(name)
                    self._tombstones[name] = self._rv(obj)
                    stale_events.append(WatchEvent("DELETED", obj))
            subs = list(self._subs)
        for ev in stale_events:
            for q in subs:
                q.put(ev)
        self.synced.set()

    # -- cache maintenance ---------------------------------------------------

    @staticmethod
    def _rv(obj) -> int:
        try:
            return int(obj.metadata.resourceVersion)
        except (TypeError, ValueError):
            return 0

    def _apply(self, ev: WatchEvent) -> None:
        name = ev.object.metadata.name
        with self._lock:
            if ev.type == "DELETED":
                self._store.pop(name, None)
                self._tombstones[name] = self._rv(ev.object)
                if len(self._tombstones) > 8192:  # bound (names are uuids)
                    for old in sorted(self._tombstones, key=self._tombstones.get)[:4096]:
                        del self._tombstones[old]
            else:
                rv = self._rv(ev.object)
                if rv > self._tombstones.get(name, -1):  # re-created object
                    self._tombstones.pop(name, None)
                    cur = self._store.get(name)
                    if cur is None or rv >= self._rv(cur):
                        self._store[name] = ev.object
            subs = list(self._subs)
        for q in subs:
            q.put(ev)

    def offer(self, obj) -> None:
        """Fold a WRITE RESPONSE into the cache (read-your-writes): kept
        only if at least as new as the cached entry and not superseded by
        a DELETED tombstone (one rv space makes the comparison exact).
        No fan-out — the watch stream delivers the canonical event."""
        name = obj.metadata.name
        with self._lock:
            if self._rv(obj) <= self._tombstones.get(name, -1):
                return  # deleted after this write; do not resurrect
            cur = self._store.get(name)
            if cur is None or self._rv(obj) >= self._rv(cur):
                self._store[name] = obj.clone()

    # -- consumers -----------------------------------------------------------

    def subscribe(self, q: "queue.Queue[WatchEvent]") -> None:
        with self._lock:
            for obj in self._store.values():
                q.put(WatchEvent("ADDED", obj))
            self._subs.append(q)

    def unsubscribe(self, q) -> None:
        with self._lock:
            if q in self._subs:
                self._subs.remove(q)

    def get_cached(self, name: str):
        with self._lock:
            obj = self._store.get(name)
        return obj.clone() if obj is not None else None

    def list_cached(self, labels: Optional[Dict[str, str]] = None):
        with self._lock:
            items = list(self._store.values())
        if labels:
            items = [
                o for o in items
                if all(o.metadata.labels.get(k) == v for k, v in labels.items())
            ]
        return [o.clone() for o in items]
