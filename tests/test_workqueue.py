import threading
import time

from cro_amd.runtime.workqueue import RateLimitedQueue


def test_dedup_while_queued():
    q = RateLimitedQueue()
    q.add("a")
    q.add("a")
    q.add("b")
    assert q.get(timeout=0.1) == "a"
    assert q.get(timeout=0.1) == "b"
    assert q.get(timeout=0.05) is None


def test_dirty_requeue_while_processing():
    q = RateLimitedQueue()
    q.add("a")
    key = q.get(timeout=0.1)
    q.add("a")  # re-added mid-processing
    assert q.get(timeout=0.05) is None  # not delivered while processing
    q.done(key)
    assert q.get(timeout=0.1) == "a"  # dirty → requeued


def test_add_after_ordering():
    q = RateLimitedQueue()
    q.add_after("late", 0.15)
    q.add_after("early", 0.02)
    t0 = time.monotonic()
    first = q.get(timeout=1)
    second = q.get(timeout=1)
    assert (first, second) == ("early", "late")
    assert time.monotonic() - t0 >= 0.14


def test_rate_limited_backoff_grows():
    q = RateLimitedQueue(base_delay=0.01, max_delay=0.2)
    t0 = time.monotonic()
    q.add_rate_limited("a")  # ~0.01
    assert q.get(timeout=1) == "a"
    q.done("a")
    first_delay = time.monotonic() - t0
    t1 = time.monotonic()
    q.add_rate_limited("a")  # ~0.02
    assert q.get(timeout=1) == "a"
    q.done("a")
    second_delay = time.monotonic() - t1
    assert second_delay > first_delay
    assert q.num_failures("a") == 2
    q.forget("a")
    assert q.num_failures("a") == 0


def test_shutdown_unblocks_consumers():
    q = RateLimitedQueue()
    results = []

    def consume():
        results.append(q.get(timeout=5))

    t = threading.Thread(target=consume)
    t.start()
    time.sleep(0.05)
    q.shutdown()
    t.join(timeout=1)
    assert results == [None]


def test_concurrent_producers_consumers():
    q = RateLimitedQueue()
    seen = []
    lock = threading.Lock()

    def worker():
        while True:
            k = q.get(timeout=0.3)
            if k is None:
                return
            with lock:
                seen.append(k)
            q.done(k)

    threads = [threading.Thread(target=worker) for _ in range(4)]
    for t in threads:
        t.start()
    for i in range(200):
        q.add(f"k{i}")
    for t in threads:
        t.join(timeout=2)
    assert len(set(seen)) == 200
