"""Concurrency stress tests — the analogue of running the Go suite under
-race (SURVEY.md §5 notes the reference CI never enables it).  Hammer the
store/informer/queue stack from many threads and assert convergence and
invariants."""

import random
import threading
import time

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.kube.client import InMemoryKubeClient
from agac.kube.informer import SharedInformerFactory, wait_for_cache_sync
from agac.kube.store import ConflictError, NotFoundError
from agac.kube.workqueue import ItemExponentialFailureRateLimiter, RateLimitingQueue


def wait_until(pred, timeout=10.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(0.01)
    return pred()


def test_store_concurrent_writers_keep_rv_monotonic():
    client = InMemoryKubeClient()
    client.create(
        corev1.Service(metadata=ObjectMeta(name="s", namespace="default"))
    )
    conflicts = [0]
    lock = threading.Lock()

    def writer(tid):
        for i in range(50):
            try:
                obj = client.get("Service", "default", "s")
                obj.metadata.annotations[f"t{tid}"] = str(i)
                client.update(obj)
            except ConflictError:
                with lock:
                    conflicts[0] += 1

    threads = [threading.Thread(target=writer, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=30)
        assert not t.is_alive()
    # optimistic concurrency must have rejected at least some races
    final = client.get("Service", "default", "s")
    assert int(final.metadata.resource_version) >= 1


def test_informer_under_churn_converges_to_store_state():
    client = InMemoryKubeClient()
    factory = SharedInformerFactory(client, resync_period=0)
    informer = factory.services()
    stop = threading.Event()
    factory.start(stop)
    try:
        assert wait_for_cache_sync(stop, informer)

        rng = random.Random(42)
        live = set()
        lock = threading.Lock()

        def churn(tid):
            r = random.Random(tid)
            for i in range(60):
                name = f"svc-{tid}-{r.randint(0, 9)}"
                op = r.random()
                try:
                    if op < 0.5:
                        client.create(
                            corev1.Service(
                                metadata=ObjectMeta(name=name, namespace="default")
                            )
                        )
                        with lock:
                            live.add(name)
                    elif op < 0.8:
                        obj = client.get("Service", "default", name)
                        obj.metadata.annotations["i"] = str(i)
                        client.update(obj)
                    else:
                        client.delete("Service", "default", name)
                        with lock:
                            live.discard(name)
                except Exception:
                    pass

        threads = [threading.Thread(target=churn, args=(i,)) for i in range(6)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=30)
            assert not t.is_alive()

        store_names = {
            o.metadata.name for o in client.list("Service")[0]
        }

        def cache_matches():
            cache_names = {o.metadata.name for o in informer.lister().list()}
            return cache_names == store_names

        assert wait_until(cache_matches), "informer cache diverged from store"
    finally:
        stop.set()


def test_queue_no_lost_items_under_contention():
    q = RateLimitingQueue(
        rate_limiter=ItemExponentialFailureRateLimiter(0.0001, 0.001), name="stress"
    )
    n_items = 300
    processed = {}
    lock = threading.Lock()

    def consumer():
        while True:
            item, shutdown = q.get()
            if shutdown:
                return
            with lock:
                processed[item] = processed.get(item, 0) + 1
            q.done(item)

    consumers = [threading.Thread(target=consumer) for _ in range(6)]
    for c in consumers:
        c.start()

    def producer(base):
        for i in range(n_items // 3):
            q.add(f"k-{base}-{i}")
            q.add_rate_limited(f"k-{base}-{i}")  # duplicate adds coalesce

    producers = [threading.Thread(target=producer, args=(i,)) for i in range(3)]
    for p in producers:
        p.start()
    for p in producers:
        p.join(timeout=30)

    expected = {f"k-{b}-{i}" for b in range(3) for i in range(n_items // 3)}
    assert wait_until(lambda: set(processed) == expected, timeout=15)
    q.shut_down()
    for c in consumers:
        c.join(timeout=5)
        assert not c.is_alive()


def test_fake_aws_concurrent_ensures_single_lock_consistency():
    """Concurrent accelerator creates + listener attach from many threads:
    ownership maps stay consistent."""
    from agac.cloudprovider.aws import types as t
    from agac.cloudprovider.fake import FakeAWSBackend

    backend = FakeAWSBackend()

    def worker(tid):
        for i in range(20):
            acc = backend.ga.create_accelerator(f"a-{tid}-{i}")
            listener = backend.ga.create_listener(
                acc.accelerator_arn, [t.PortRange(80, 80)], "TCP"
            )
            backend.ga.create_endpoint_group(listener.listener_arn, "us-east-1")

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(5)]
    for t_ in threads:
        t_.start()
    for t_ in threads:
        t_.join(timeout=30)
        assert not t_.is_alive()

    accs, _ = backend.ga.list_accelerators()
    assert len(accs) == 100
    for acc in accs:
        listeners, _ = backend.ga.list_listeners(acc.accelerator_arn)
        assert len(listeners) == 1
        groups, _ = backend.ga.list_endpoint_groups(listeners[0].listener_arn)
        assert len(groups) == 1


def test_watch_sever_chaos_reconverges():
    """Random watch severing while churning: the resume-from-last-rv path
    must always reconverge the cache to store state (bounded: 2 trials)."""
    from agac.kube.informer import SharedInformerFactory, wait_for_cache_sync

    for trial in range(2):
        client = InMemoryKubeClient()
        factory = SharedInformerFactory(client, resync_period=0)
        informer = factory.services()
        stop = threading.Event()
        factory.start(stop)
        try:
            assert wait_for_cache_sync(stop, informer)
            stop_chaos = threading.Event()

            def churn(tid):
                r = random.Random(trial * 100 + tid)
                for i in range(120):
                    name = f"s-{r.randint(0, 10)}"
                    try:
                        op = r.random()
                        if op < 0.45:
                            client.create(
                                corev1.Service(
                                    metadata=ObjectMeta(name=name, namespace="d")
                                )
                            )
                        elif op < 0.8:
                            o = client.get("Service", "d", name)
                            o.metadata.annotations["i"] = str(i)
                            client.update(o)
                        else:
                            client.delete("Service", "d", name)
                    except Exception:
                        pass

            def severer():
                r = random.Random(trial + 999)
                while not stop_chaos.is_set():
                    w = informer._watch
                    if w is not None and r.random() < 0.6:
                        try:
                            w.stop()
                        except Exception:
                            pass
                    time.sleep(0.02)

            sever_thread = threading.Thread(target=severer, daemon=True)
            sever_thread.start()
            threads = [threading.Thread(target=churn, args=(t,)) for t in range(3)]
            for t in threads:
                t.start()
            for t in threads:
                t.join(30)
            stop_chaos.set()
            sever_thread.join(5)

            def converged():
                store_state = {
                    o.metadata.name: o.metadata.resource_version
                    for o in client.list("Service")[0]
                }
                cache_state = {
                    o.metadata.name: o.metadata.resource_version
                    for o in informer.lister().list()
                }
                return store_state == cache_state

            assert wait_until(converged), f"trial {trial} diverged"
        finally:
            stop.set()
