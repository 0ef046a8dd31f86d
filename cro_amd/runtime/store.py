"""In-process API server: typed object store with k8s-faithful semantics.

This is the coordination bus of the operator (the analog of the kube-apiserver
the reference talks to, and of the envtest apiserver its tests run against —
suite_test.go:318-409).  It implements the exact subset of apiserver behavior
the reconcile semantics depend on:

* optimistic concurrency — update/status-update require the caller's
  ``resourceVersion`` to match, else :class:`ConflictError` (the reference's
  state machines rely on conflict-retry for idempotence);
* a separate **status subresource** — ``update`` never changes status,
  ``update_status`` never changes spec/metadata (kubebuilder
  ``+kubebuilder:subresource:status`` on both CRDs);
* **finalizer** semantics — delete on an object with finalizers sets
  ``deletionTimestamp`` and fires MODIFIED; the object is removed only when a
  later update clears the finalizer list (both controllers' Deleting states);
* **admission** hooks on create/update of the main resource (not status),
  matching the validating webhook registration
  (composabilityrequest_webhook.go:49: verbs=create;update, no subresource);
* **watch** — per-subscriber event queues carrying deep copies, the event
  source for controllers (watch-driven reconciles are how this build beats
  the reference's 30 s poll quantum).

In cluster deployments persistence belongs to etcd behind the real
apiserver (runtime/client.py keeps the Client surface identical).  In
standalone mode, ``persist_path`` gives this store its own durability:
debounced atomic snapshots reloaded on restart with RV/watch-sequence
continuity (the ``--data-dir`` flag).
"""

from __future__ import annotations

import itertools
import queue
import threading
import time
import uuid as uuidlib
from collections import deque
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional

from ..api import _schema_validation
from ..api.v1alpha1.types import K8sObject
from .errors import AlreadyExistsError, ConflictError, NotFoundError

ADDED = "ADDED"
MODIFIED = "MODIFIED"
DELETED = "DELETED"


@dataclass
class WatchEvent:
    """READ-ONLY for consumers: ``object``/``old_object`` may be shared
    across subscriber queues (single fan-out copy). Level-triggered
    reconcilers re-get from the store before mutating — never write
    through an event object."""

    type: str  # ADDED | MODIFIED | DELETED
    object: K8sObject
    old_object: Optional[K8sObject] = None
    seq: int = 0  # global event sequence — the watch resume token


# admission validator: fn(operation: "CREATE"|"UPDATE", old: obj|None, new: obj)
AdmissionFn = Callable[[str, Optional[K8sObject], K8sObject], None]


def _now_rfc3339() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


class _Watcher:
    __slots__ = ("kinds", "queue", "closed")

    def __init__(self, kinds: Optional[List[str]]):
        self.kinds = set(kinds) if kinds else None
        self.queue: "queue.Queue[WatchEvent]" = queue.Queue()
        self.closed = False

    def wants(self, kind: str) -> bool:
        return self.kinds is None or kind in self.kinds


class InMemoryStore:
    def __init__(
        self,
        persist_path: Optional[str] = None,
        persist_debounce: float = 0.05,
    ) -> None:
        """``persist_path`` enables write-through durability: every
        mutation schedules a debounced atomic snapshot (tmp + fsync +
        rename) and a restarted store reloads objects, the RV counter and
        the watch sequence from it — the standalone analog of etcd (the
        reference's CRD status IS its checkpoint; without
        this a standalone restart would forget every request and rely on
        the syncer to repair orphaned fabric attachments)."""
        self._lock = threading.RLock()
        self._objects: Dict[str, Dict[str, K8sObject]] = {}
        self._rv = itertools.count(1)
        self._watchers: List[_Watcher] = []
        self._admission: Dict[str, List[AdmissionFn]] = {}
        # bounded event history for resourceVersion-resuming watches
        # (apiserver watch-cache analog): a reconnecting client replays
        # only what it missed instead of a full re-list
        self._event_seq = 0
        self._event_log: "deque[WatchEvent]" = deque(maxlen=4096)
        self._persist_path = persist_path
        self._persist_debounce = persist_debounce
        self._dirty = threading.Event()
        self._persist_stop = False
        self._persist_thread: Optional[threading.Thread] = None
        if persist_path:
            self._load()
            self._persist_thread = threading.Thread(
                target=self._persist_loop, name="store-persist", daemon=True
            )
            self._persist_thread.start()

    # -- durability --------------------------------------------------------

    def _load(self) -> None:
        import json
        import os

        from ..api.v1alpha1.types import ALL_KINDS

        if not os.path.exists(self._persist_path):
            return
        try:
            with open(self._persist_path) as f:
                snap = json.load(f)
        except (OSError, ValueError):
            import logging

            logging.getLogger(__name__).exception(
                "store snapshot %s unreadable; starting empty", self._persist_path
            )
            return
        self._rv = itertools.count(int(snap.get("rv", 0)) + 1)
        # seq continuity keeps pre-restart watch tokens comparable: a
        # token < seq with an empty log → Expired → clean re-list.
        # seq and rv share one counting space (apiserver parity); old
        # snapshots carried them separately, so take the max.
        self._event_seq = max(int(snap.get("seq", 0)), int(snap.get("rv", 0)))
        for kind, objs in snap.get("objects", {}).items():
            cls = ALL_KINDS.get(kind)
            if cls is None:
                continue
            for name, dump in objs.items():
                try:
                    self._objects.setdefault(kind, {})[name] = cls.model_validate(dump)
                except Exception:
                    import logging

                    logging.getLogger(__name__).exception(
                        "skipping unreadable %s/%s in snapshot", kind, name
                    )

    def _snapshot(self) -> dict:
        with self._lock:
            return {
                "rv": self._peek_rv(),
                "seq": self._event_seq,
                "objects": {
                    kind: {
                        name: obj.model_dump(by_alias=True)
                        for name, obj in objs.items()
                    }
                    for kind, objs in self._objects.items()
                },
            }

    def _peek_rv(self) -> int:
        # itertools.count cannot be peeked; track via a probe without
        # consuming: replace the counter with one continuing from n+1
        n = next(self._rv)
        self._rv = itertools.count(n + 1)
        return n

    def persist_now(self) -> None:
        """Synchronous atomic snapshot (graceful shutdown / tests)."""
        if not self._persist_path:
            return
        import json
        import os
        import tempfile

        snap = self._snapshot()
        directory = os.path.dirname(self._persist_path) or "."
        os.makedirs(directory, exist_ok=True)
        fd, tmp = tempfile.mkstemp(dir=directory, prefix=".state-")
        try:
            with os.fdopen(fd, "w") as f:
                json.dump(snap, f)
                f.flush()
                os.fsync(f.fileno())
            os.replace(tmp, self._persist_path)
        except BaseException:
            try:
                os.unlink(tmp)
            except OSError:
                pass
            raise

    def _persist_loop(self) -> None:
        while not self._persist_stop:
            self._dirty.wait()
            if self._persist_stop:
                return
            time.sleep(self._persist_debounce)  # coalesce write bursts
            self._dirty.clear()
            try:
                self.persist_now()
            except Exception:  # pragma: no cover - disk trouble
                import logging

                logging.getLogger(__name__).exception("store snapshot failed")

    def close(self) -> None:
        """Flush and stop the persistence thread."""
        if self._persist_path:
            self._persist_stop = True
            self._dirty.set()
            if self._persist_thread is not None:
                self._persist_thread.join(timeout=2)
            self.persist_now()

    # -- admission ---------------------------------------------------------

    def register_admission(self, kind: str, fn: AdmissionFn) -> None:
        self._admission.setdefault(kind, []).append(fn)

    def _admit(self, op: str, old: Optional[K8sObject], new: K8sObject) -> None:
        for fn in self._admission.get(new.kind, ()):  # failurePolicy=fail
            fn(op, old, new)

    # -- watch -------------------------------------------------------------

    def watch(self, kinds: Optional[List[str]] = None) -> "queue.Queue[WatchEvent]":
        w = _Watcher(kinds)
        with self._lock:
            self._watchers.append(w)
        return w.queue

    def stop_watch(self, q: "queue.Queue[WatchEvent]") -> None:
        """Unsubscribe a watcher (long-lived processes creating transient
        watches must release them or every event fans out to dead queues)."""
        with self._lock:
            for w in self._watchers:
                if w.queue is q:
                    w.closed = True
            self._watchers = [w for w in self._watchers if not w.closed]

    def current_seq(self) -> int:
        with self._lock:
            return self._event_seq

    def events_since(
        self, seq: int, kinds: Optional[List[str]] = None
    ) -> Optional[List[WatchEvent]]:
        """Events after resume token ``seq``, or None when the token has
        aged out of the bounded log (client must fall back to re-list)."""
        wanted = set(kinds) if kinds else None
        with self._lock:
            if seq == self._event_seq:
                return []  # caught up
            if seq > self._event_seq:
                # token from a previous store incarnation (server restarted
                # with fresh state) — not comparable; force a full re-list
                return None
            if not self._event_log or self._event_log[0].seq > seq + 1:
                return None  # compacted past the token
            return [
                WatchEvent(e.type, e.object.clone(), None, e.seq)
                for e in self._event_log
                if e.seq > seq and (wanted is None or e.object.kind in wanted)
            ]

    def _notify(self, ev: WatchEvent) -> None:
        # ONE logical resourceVersion space (apiserver/etcd-revision parity):
        # every write bumps the object RV exactly once and the event's resume
        # token IS that RV, so clients resume a watch from the last event
        # object's metadata.resourceVersion exactly as k8s informers do.
        # callers hold self._lock (all mutators notify inside their
        # critical section), so the seq assignment and log append are
        # atomic with the mutation itself
        self._event_seq = int(ev.object.metadata.resourceVersion)
        ev.seq = self._event_seq
        self._event_log.append(
            WatchEvent(ev.type, ev.object.clone(), None, ev.seq)
        )
        if self._persist_path:
            self._dirty.set()
        # ONE copy shared by every subscriber queue (read-only contract on
        # WatchEvent — consumers re-get from the store before mutating, as
        # level-triggered reconcilers do; the per-subscriber deep copies
        # this replaces were the hot path's largest allocation source).
        # The stored object itself is never handed out, and the event-log
        # copy above stays pristine for replays.
        shared = None
        shared_old = None
        for w in self._watchers:
            if not w.closed and w.wants(ev.object.kind):
                if shared is None:
                    shared = ev.object.clone()
                    shared_old = ev.old_object.clone() if ev.old_object else None
                w.queue.put(WatchEvent(ev.type, shared, shared_old, ev.seq))

    # -- CRUD --------------------------------------------------------------

    def create(self, obj: K8sObject) -> K8sObject:
        obj = obj.clone()
        _schema_validation.validate_spec(obj)
        with self._lock:
            bucket = self._objects.setdefault(obj.kind, {})
            if not obj.metadata.name:
                if obj.metadata.generateName:
                    obj.metadata.name = obj.metadata.generateName + uuidlib.uuid4().hex[:6]
                else:
                    raise ValueError("object has neither name nor generateName")
            if obj.metadata.name in bucket:
                raise AlreadyExistsError(f"{obj.kind}/{obj.metadata.name} already exists")
            self._admit("CREATE", None, obj)
            obj.metadata.uid = str(uuidlib.uuid4())
            obj.metadata.resourceVersion = str(next(self._rv))
            obj.metadata.generation = 1
            obj.metadata.creationTimestamp = _now_rfc3339()
            obj.metadata.deletionTimestamp = None
            bucket[obj.metadata.name] = obj
            self._notify(WatchEvent(ADDED, obj))
            return obj.clone()

    def get(self, kind: str, name: str) -> K8sObject:
        # stored objects are immutable-after-insert (every mutator swaps in
        # a fresh object), so the snapshot ref is taken under the lock and
        # the copy happens OUTSIDE it — clone cost never extends the
        # store's critical section (the contended-bench queueing term)
        with self._lock:
            try:
                stored = self._objects[kind][name]
            except KeyError:
                raise NotFoundError(f"{kind}/{name} not found") from None
        return stored.clone()

    def list(
        self,
        kind: str,
        label_selector: Optional[Dict[str, str]] = None,
        copy: bool = True,
    ) -> List[K8sObject]:
        """``copy=False`` returns the stored objects themselves — READ-ONLY
        snapshots for hot validation/allocation paths (admission lists every
        request per CREATE; deep-copying a 400-object fleet per call is the
        O(n²) term). Callers must not mutate them."""
        with self._lock:
            items = list(self._objects.get(kind, {}).values())
        if label_selector:
            items = [
                o
                for o in items
                if all(o.metadata.labels.get(k) == v for k, v in label_selector.items())
            ]
        if not copy:
            return items
        return [o.clone() for o in items]  # outside the lock (see get)

    def update(self, obj: K8sObject) -> K8sObject:
        """Update metadata+spec; status is preserved from the stored object."""
        obj = obj.clone()
        _schema_validation.validate_spec(obj)
        with self._lock:
            stored = self._require(obj.kind, obj.metadata.name)
            self._check_rv(stored, obj)
            # no-op update: no RV bump, no watch event (apiserver parity)
            incoming_meta = obj.metadata.model_copy(
                update={
                    "resourceVersion": stored.metadata.resourceVersion,
                    "uid": stored.metadata.uid,
                    "generation": stored.metadata.generation,
                    "creationTimestamp": stored.metadata.creationTimestamp,
                    "deletionTimestamp": stored.metadata.deletionTimestamp,
                }
            )
            # compare the WHOLE object minus status (not just spec): kinds
            # with top-level fields (Event count/last_seen) must not be
            # misread as no-ops
            incoming_dump = obj.model_copy(
                update={"metadata": incoming_meta}
            ).model_dump(exclude={"status"})
            if incoming_dump == stored.model_dump(exclude={"status"}):
                return stored.clone()
            self._admit("UPDATE", stored, obj)
            new = obj
            if hasattr(stored, "status"):
                new.status = stored.status.clone()
            spec_changed = getattr(stored, "spec", None) != getattr(new, "spec", None)
            new.metadata.uid = stored.metadata.uid
            new.metadata.creationTimestamp = stored.metadata.creationTimestamp
            new.metadata.deletionTimestamp = stored.metadata.deletionTimestamp
            new.metadata.generation = stored.metadata.generation + (1 if spec_changed else 0)
            new.metadata.resourceVersion = str(next(self._rv))
            self._objects[obj.kind][obj.metadata.name] = new
            if new.metadata.deletionTimestamp and not new.metadata.finalizers:
                return self._finalize_delete(new)
            self._notify(WatchEvent(MODIFIED, new, stored))
            return new.clone()

    def update_status(self, obj: K8sObject) -> K8sObject:
        """Status-subresource update: only .status is applied.

        A no-op write (identical status) returns the stored object without
        bumping resourceVersion or firing a watch event — apiserver parity;
        without this, a reconciler that re-writes the same error on every
        retry generates its own watch events and hot-loops.
        """
        with self._lock:
            stored = self._require(obj.kind, obj.metadata.name)
            self._check_rv(stored, obj)
            if stored.status == obj.status:
                return stored.clone()
            new = stored.clone()
            new.status = obj.status.clone()
            new.metadata.resourceVersion = str(next(self._rv))
            self._objects[obj.kind][obj.metadata.name] = new
            self._notify(WatchEvent(MODIFIED, new, stored))
            return new.clone()

    def delete(self, kind: str, name: str) -> None:
        with self._lock:
            stored = self._require(kind, name)
            if stored.metadata.finalizers:
                if stored.metadata.deletionTimestamp is None:
                    # swap, never mutate in place: get/list hand out refs
                    # and clone outside the lock (immutable-after-insert)
                    new = stored.clone()
                    new.metadata.deletionTimestamp = _now_rfc3339()
                    new.metadata.resourceVersion = str(next(self._rv))
                    self._objects[kind][name] = new
                    self._notify(WatchEvent(MODIFIED, new, stored))
                return
            self._finalize_delete(stored)

    # -- internals ---------------------------------------------------------

    def _require(self, kind: str, name: str) -> K8sObject:
        try:
            return self._objects[kind][name]
        except KeyError:
            raise NotFoundError(f"{kind}/{name} not found") from None

    @staticmethod
    def _check_rv(stored: K8sObject, incoming: K8sObject) -> None:
        if (
            incoming.metadata.resourceVersion
            and incoming.metadata.resourceVersion != stored.metadata.resourceVersion
        ):
            raise ConflictError(
                f"{stored.kind}/{stored.metadata.name}: stale resourceVersion "
                f"{incoming.metadata.resourceVersion} (stored {stored.metadata.resourceVersion})"
            )

    def _finalize_delete(self, stored: K8sObject) -> K8sObject:
        del self._objects[stored.kind][stored.metadata.name]
        # a delete is a write: it gets its own resourceVersion, and the
        # DELETED event's object carries it (etcd-revision semantics — the
        # resume-token space stays in lockstep with object RVs). Copy
        # before bumping: earlier get/list snapshots may still alias
        # ``stored`` (immutable-after-insert contract).
        final = stored.clone()
        final.metadata.resourceVersion = str(next(self._rv))
        self._notify(WatchEvent(DELETED, final))
        return final
