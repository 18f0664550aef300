"""Workqueue semantics tests (client-go contract the reference relies on)."""

import threading
import time

from agac.kube.workqueue import (
    BucketRateLimiter,
    ItemExponentialFailureRateLimiter,
    MaxOfRateLimiter,
    RateLimitingQueue,
)


def make_queue(name="test"):
    # fast limiter so tests don't sleep
    limiter = ItemExponentialFailureRateLimiter(base_delay=0.001, max_delay=0.05)
    return RateLimitingQueue(rate_limiter=limiter, name=name)


def test_fifo_and_done():
    q = make_queue()
    q.add("a")
    q.add("b")
    assert len(q) == 2
    item, shutdown = q.get()
    assert item == "a" and not shutdown
    q.done("a")
    item, _ = q.get()
    assert item == "b"


def test_dedup_while_queued():
    q = make_queue()
    q.add("a")
    q.add("a")
    assert len(q) == 1


def test_requeue_while_processing():
    # A key added while being processed must re-appear exactly once after done.
    q = make_queue()
    q.add("a")
    item, _ = q.get()
    q.add("a")  # processing → marked dirty, not queued yet
    assert len(q) == 0
    q.done("a")
    assert len(q) == 1
    item, _ = q.get()
    assert item == "a"
    q.done("a")
    assert len(q) == 0


def test_add_after_delivers():
    q = make_queue()
    q.add_after("x", 0.02)
    assert len(q) == 0
    item, shutdown = q.get(timeout=10.0)
    assert item == "x" and not shutdown


def test_add_after_ordering():
    q = make_queue()
    q.add_after("late", 0.05)
    q.add_after("early", 0.005)
    first, _ = q.get(timeout=10.0)
    assert first == "early"
    second, _ = q.get(timeout=10.0)
    assert second == "late"


def test_rate_limited_backoff_and_forget():
    lim = ItemExponentialFailureRateLimiter(base_delay=0.01, max_delay=1.0)
    assert lim.when("k") == 0.01
    assert lim.when("k") == 0.02
    assert lim.when("k") == 0.04
    assert lim.num_requeues("k") == 3
    lim.forget("k")
    assert lim.num_requeues("k") == 0
    assert lim.when("k") == 0.01


def test_exponential_cap():
    lim = ItemExponentialFailureRateLimiter(base_delay=0.01, max_delay=0.05)
    for _ in range(10):
        delay = lim.when("k")
    assert delay == 0.05


def test_bucket_rate_limiter_burst():
    lim = BucketRateLimiter(qps=10.0, burst=3)
    assert lim.when("a") == 0.0
    assert lim.when("a") == 0.0
    assert lim.when("a") == 0.0
    assert lim.when("a") > 0.0  # past burst → throttled


def test_max_of_rate_limiter():
    lim = MaxOfRateLimiter(
        ItemExponentialFailureRateLimiter(base_delay=0.5, max_delay=10.0),
        BucketRateLimiter(qps=1000.0, burst=1000),
    )
    assert lim.when("a") == 0.5


def test_shutdown_drains_then_signals():
    q = make_queue()
    q.add("a")
    q.shut_down()
    item, shutdown = q.get()
    assert item == "a" and not shutdown
    q.done("a")
    item, shutdown = q.get()
    assert item is None and shutdown


def test_shutdown_wakes_blocked_worker():
    q = make_queue()
    results = []

    def worker():
        item, shutdown = q.get()
        results.append((item, shutdown))

    t = threading.Thread(target=worker)
    t.start()
    time.sleep(0.05)
    q.shut_down()
    t.join(timeout=10.0)
    assert not t.is_alive()
    assert results == [(None, True)]


def test_concurrent_producers_consumers():
    q = make_queue()
    n = 200
    seen = []
    seen_lock = threading.Lock()

    def producer(start):
        for i in range(start, start + n // 2):
            q.add(i)

    def consumer():
        while True:
            item, shutdown = q.get()
            if shutdown:
                return
            with seen_lock:
                seen.append(item)
            q.done(item)

    consumers = [threading.Thread(target=consumer) for _ in range(4)]
    for c in consumers:
        c.start()
    producers = [threading.Thread(target=producer, args=(0,)), threading.Thread(target=producer, args=(n // 2,))]
    for p in producers:
        p.start()
    for p in producers:
        p.join()
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline:
        with seen_lock:
            if len(seen) == n:
                break
        time.sleep(0.01)
    q.shut_down()
    for c in consumers:
        c.join(timeout=2)
    assert sorted(seen) == list(range(n))
