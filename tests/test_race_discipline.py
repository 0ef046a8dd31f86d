"""Systematic thread-race discipline (SURVEY.md §5.2 — the go-test -race
analog for the threaded runtime).

Part 1 proves the detector itself catches the two bug classes; Part 2 runs
the FULL operator stack (manager threads, 8-way workqueue workers, both
controllers, syncer, event recorder, store, mock node ops) through a
concurrent churn burst with every runtime lock instrumented and asserts a
clean lock-order graph — no inversion cycles, no self-deadlocks — across
all lock pairs actually exercised.
"""

import threading

import pytest

from cro_amd.runtime import lockcheck


# -- Part 1: the detector catches real bugs ---------------------------------


def test_detector_flags_order_inversion():
    with lockcheck.instrument() as report:
        a = threading.Lock()
        b = threading.Lock()

        def ab():
            with a:
                with b:
                    pass

        def ba():
            with b:
                with a:
                    pass

        t1 = threading.Thread(target=ab)
        t2 = threading.Thread(target=ba)
        t1.start(); t1.join()
        t2.start(); t2.join()
    with pytest.raises(lockcheck.LockOrderError, match="inversion"):
        report.assert_clean()


def test_detector_flags_self_deadlock():
    with lockcheck.instrument() as report:
        a = threading.Lock()
        with pytest.raises(lockcheck.LockOrderError, match="self-deadlock"):
            with a:
                a.acquire()  # would hang forever without the detector
    assert report.self_deadlocks


def test_detector_accepts_consistent_order():
    with lockcheck.instrument() as report:
        a = threading.Lock()
        b = threading.Lock()
        for _ in range(3):
            with a:
                with b:
                    pass
    report.assert_clean()
    assert len(report.edges) == 1


def test_detector_nonblocking_probe_is_exempt():
    """Condition._is_owned probes a held lock with acquire(False); that is
    not a deadlock and must not be flagged (queue.Queue relies on it)."""
    import queue

    with lockcheck.instrument() as report:
        q = queue.Queue()
        q.put(1)
        assert q.get() == 1
        # explicit probe: held lock + non-blocking acquire → False, no raise
        a = threading.Lock()
        with a:
            assert a.acquire(False) is False
    report.assert_clean()


# -- Part 2: the operator runtime is inversion-free -------------------------


@pytest.mark.timeout(120)
def test_operator_runtime_lock_order_clean():
    """Concurrent churn across the whole threaded stack under
    instrumentation: every lock pair the runtime actually takes nested
    must form an acyclic order."""
    with lockcheck.instrument() as report:
        import copy

        from cro_amd.api.v1alpha1.types import ComposabilityRequest, Node
        from cro_amd.bench_harness import attach_detach_cycle, build_local_stack

        stack = build_local_stack(node_name="race-node", use_gpu=False,
                                  syncer_period=0.05)
        # one node per worker (same-model CRs on one node are inadmissible
        # under webhook rule 3 — the bench contention shape)
        for w in range(4):
            n = Node()
            n.metadata.name = f"race-node-{w}"
            n.status.capacity.milli_cpu = 64000
            n.status.capacity.memory = 1 << 40
            n.status.capacity.allowed_pod_number = 128
            stack.mgr.client.create(n)
            stack.ops.set_driver(f"race-node-{w}", True)
        stack.mgr.start()
        errs = []

        def worker(wid):
            proxy = copy.copy(stack)
            proxy.node_name = f"race-node-{wid}"
            events = stack.mgr.store.watch(["ComposabilityRequest"])
            try:
                for i in range(6):
                    attach_detach_cycle(
                        proxy, f"race-{wid}-{i}", size=1, events=events,
                        timeout=30,
                    )
            except Exception as exc:
                errs.append(f"{wid}: {exc}")
            finally:
                stack.mgr.store.stop_watch(events)

        threads = [threading.Thread(target=worker, args=(w,)) for w in range(4)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        stack.mgr.stop()
        assert not errs, errs
        assert not stack.mgr.client.list(ComposabilityRequest)

    # the run must have exercised nested locking at all for this to mean
    # anything — then the graph must be clean
    assert report.edges, "no nested lock acquisitions observed (instrumentation broken?)"
    report.assert_clean()
