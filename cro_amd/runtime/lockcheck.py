"""Lock-order race/deadlock detector — the threaded runtime's sanitizer.

The reference is single-binary Go and could run its suite under the Go
race detector (it does not — SURVEY.md §5.2); CPython has no TSan, so this
module provides the systematic analog this heavily threaded runtime needs:

* ``instrument()`` patches ``threading.Lock`` so every mutex CREATED
  inside the context is tracked (locks are keyed by their construction
  site, so all instances of e.g. the store lock form one graph node).
* While instrumented, each thread's held-lock set is maintained; acquiring
  B while holding A records the order edge A→B with a witness traceback.
* ``Report.assert_clean()`` fails on:
  - **order inversion**: a cycle in the lock-order graph (two code paths
    taking the same pair of locks in opposite orders — the classic
    deadlock precondition, caught even when the interleaving never
    actually deadlocked during the run);
  - **self-deadlock**: a blocking re-acquire of a held ``Lock`` by its own
    holder — detected and raised at acquire time instead of hanging the
    test run.  Non-blocking probe acquires (``acquire(False)`` — used by
    ``threading.Condition._is_owned``) are exempt: they cannot hang and
    legitimately probe held locks.

``RLock`` is tracked too (the store's central lock is one): reentrant
re-acquires are legal (no self-deadlock check) and the ``Condition``
private hooks (``_release_save``/``_acquire_restore``/``_is_owned``) are
delegated with held-set bookkeeping so ``Condition.wait`` does not
desynchronize the graph.

Used by ``tests/test_race_discipline.py``, which drives the full operator
stack (manager, both controllers, workqueue, store, node ops) through a
concurrent churn burst under instrumentation — the ``go test -race``
stand-in for every release.
"""

from __future__ import annotations

import threading
import traceback
from contextlib import contextmanager
from typing import Dict, List, Set, Tuple

_REAL_LOCK = threading.Lock
_REAL_RLOCK = threading.RLock


class LockOrderError(AssertionError):
    pass


class Report:
    """Shared state for one instrumentation session."""

    def __init__(self):
        self._mu = _REAL_LOCK()
        # per-thread stack of (site, lock id) currently held
        self._held: Dict[int, List[Tuple[str, int]]] = {}
        # order edges: (site_a, site_b) -> witness traceback string
        self.edges: Dict[Tuple[str, str], str] = {}
        self.self_deadlocks: List[str] = []

    # -- called by tracked locks -------------------------------------------

    def check_blocking_acquire(self, site: str, lock_id: int) -> None:
        """Raises if this thread already holds the lock (would hang)."""
        tid = threading.get_ident()
        with self._mu:
            held = self._held.get(tid, [])
            if any(lid == lock_id for _, lid in held):
                witness = "".join(traceback.format_stack(limit=12))
                self.self_deadlocks.append(
                    f"thread re-acquired lock created at {site}\n{witness}"
                )
                raise LockOrderError(
                    f"self-deadlock: lock created at {site} re-acquired by "
                    f"its holder (would hang outside the detector)"
                )

    def record_acquired(self, site: str, lock_id: int) -> None:
        tid = threading.get_ident()
        with self._mu:
            held = self._held.setdefault(tid, [])
            for prev_site, _ in held:
                if prev_site != site and (prev_site, site) not in self.edges:
                    self.edges[(prev_site, site)] = "".join(
                        traceback.format_stack(limit=10)
                    )
            held.append((site, lock_id))

    def record_released(self, site: str, lock_id: int) -> None:
        tid = threading.get_ident()
        with self._mu:
            held = self._held.get(tid, [])
            for i in range(len(held) - 1, -1, -1):
                if held[i][1] == lock_id:
                    del held[i]
                    break

    def record_released_all(self, site: str, lock_id: int) -> None:
        """Drop every held entry for a lock (full recursion-count release,
        the Condition.wait _release_save path)."""
        tid = threading.get_ident()
        with self._mu:
            held = self._held.get(tid, [])
            held[:] = [(s, lid) for s, lid in held if lid != lock_id]

    # -- analysis -----------------------------------------------------------

    def cycles(self) -> List[List[str]]:
        graph: Dict[str, Set[str]] = {}
        for a, b in self.edges:
            graph.setdefault(a, set()).add(b)
            graph.setdefault(b, set())
        found: List[List[str]] = []
        WHITE, GRAY, BLACK = 0, 1, 2
        color = {n: WHITE for n in graph}
        stack: List[str] = []

        def dfs(n: str) -> None:
            color[n] = GRAY
            stack.append(n)
            for m in graph[n]:
                if color[m] == GRAY:
                    found.append(stack[stack.index(m):] + [m])
                elif color[m] == WHITE:
                    dfs(m)
            stack.pop()
            color[n] = BLACK

        for n in graph:
            if color[n] == WHITE:
                dfs(n)
        return found

    def assert_clean(self) -> None:
        if self.self_deadlocks:
            raise LockOrderError(
                f"{len(self.self_deadlocks)} self-deadlock(s):\n"
                + "\n".join(self.self_deadlocks[:3])
            )
        cyc = self.cycles()
        if cyc:
            detail = []
            for path in cyc[:3]:
                detail.append(" -> ".join(path))
                for a, b in zip(path, path[1:]):
                    w = self.edges.get((a, b))
                    if w:
                        detail.append(f"  edge {a} -> {b} witnessed at:\n{w}")
            raise LockOrderError(
                f"lock-order inversion ({len(cyc)} cycle(s)):\n" + "\n".join(detail)
            )


class _TrackedLock:
    __slots__ = ("_lock", "_site", "_report")

    def __init__(self, site: str, report: Report):
        self._lock = _REAL_LOCK()
        self._site = site
        self._report = report

    def acquire(self, blocking: bool = True, timeout: float = -1):
        if blocking:
            self._report.check_blocking_acquire(self._site, id(self))
        got = self._lock.acquire(blocking, timeout)
        if got:
            self._report.record_acquired(self._site, id(self))
        return got

    acquire_lock = acquire  # legacy alias some stdlib paths use

    def release(self):
        self._lock.release()
        self._report.record_released(self._site, id(self))

    release_lock = release

    def __enter__(self):
        self.acquire()
        return self

    def __exit__(self, *exc):
        self.release()
        return False

    def locked(self):
        return self._lock.locked()


class _TrackedRLock:
    """Tracked reentrant lock.  Re-acquire by the holder is legal; order
    edges are still recorded against OTHER held locks.  The Condition
    integration hooks delegate to the real RLock with held-set fixup."""

    __slots__ = ("_lock", "_site", "_report")

    def __init__(self, site: str, report: Report):
        self._lock = _REAL_RLOCK()
        self._site = site
        self._report = report

    def acquire(self, blocking: bool = True, timeout: float = -1):
        got = self._lock.acquire(blocking, timeout)
        if got:
            self._report.record_acquired(self._site, id(self))
        return got

    def release(self):
        self._lock.release()
        self._report.record_released(self._site, id(self))

    def __enter__(self):
        self.acquire()
        return self

    def __exit__(self, *exc):
        self.release()
        return False

    # -- Condition hooks ----------------------------------------------------

    def _is_owned(self):
        return self._lock._is_owned()

    def _release_save(self):
        state = self._lock._release_save()  # drops the FULL recursion count
        self._report.record_released_all(self._site, id(self))
        return state

    def _acquire_restore(self, state):
        self._lock._acquire_restore(state)
        self._report.record_acquired(self._site, id(self))


def _creation_site() -> str:
    # first frame outside this module and outside threading/queue internals
    for frame in reversed(traceback.extract_stack(limit=12)[:-2]):
        fn = frame.filename
        if "lockcheck" not in fn and "threading" not in fn:
            return f"{fn}:{frame.lineno}"
    return "unknown"


@contextmanager
def instrument():
    """Patch Lock construction; yields the :class:`Report`.

    Only locks CREATED inside the context are tracked — build the world
    under test inside.  Construction patching (not acquisition patching)
    keeps pre-existing locks, foreign threads and pytest internals
    untouched; tracked locks keep working after the context exits.
    """
    report = Report()

    def make_lock():
        return _TrackedLock(_creation_site(), report)

    def make_rlock():
        return _TrackedRLock(_creation_site(), report)

    threading.Lock = make_lock
    threading.RLock = make_rlock
    try:
        yield report
    finally:
        threading.Lock = _REAL_LOCK
        threading.RLock = _REAL_RLOCK
