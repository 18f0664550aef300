"""Event recorder aggregation + Prometheus metrics hooks."""

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.kube.client import InMemoryKubeClient
from agac.kube.events import EventRecorder


def mk_obj():
    return corev1.Service(metadata=ObjectMeta(name="web", namespace="default", uid="u1"))


def list_events(client):
    items, _ = client.list("Event")
    return items


def test_event_creates_object_with_reference():
    client = InMemoryKubeClient()
    recorder = EventRecorder(client, "test-controller")
    recorder.event(mk_obj(), "Normal", "GlobalAcceleratorCreated", "created arn:x")
    assert recorder.flush()
    events = list_events(client)
    assert len(events) == 1
    ev = events[0]
    assert ev.reason == "GlobalAcceleratorCreated"
    assert ev.type == "Normal"
    assert ev.involved_object.kind == "Service"
    assert ev.involved_object.name == "web"
    assert ev.source.component == "test-controller"
    assert ev.count == 1


def test_repeated_event_aggregates_count():
    client = InMemoryKubeClient()
    recorder = EventRecorder(client, "c")
    for _ in range(3):
        recorder.event(mk_obj(), "Normal", "Reason", "same message")
    assert recorder.flush()
    events = list_events(client)
    assert len(events) == 1
    assert events[0].count == 3


def test_different_messages_make_distinct_events():
    client = InMemoryKubeClient()
    recorder = EventRecorder(client, "c")
    recorder.eventf(mk_obj(), "Normal", "Reason", "msg %d", 1)
    recorder.eventf(mk_obj(), "Normal", "Reason", "msg %d", 2)
    assert recorder.flush()
    assert len(list_events(client)) == 2


def test_recorder_never_raises():
    class BrokenClient(InMemoryKubeClient):
        def create(self, obj):
            raise RuntimeError("apiserver down")

    recorder = EventRecorder(BrokenClient(), "c")
    recorder.event(mk_obj(), "Normal", "Reason", "m")  # must not raise
    assert recorder.flush()  # writer thread survives the failure


def test_recording_is_async_and_flushable():
    """client-go EventBroadcaster behavior: event() never blocks on API
    I/O; flush() drains the buffer."""
    import threading as _threading
    import time as _time

    gate = _threading.Event()

    class SlowClient(InMemoryKubeClient):
        def create(self, obj):
            gate.wait(2.0)
            return super().create(obj)

    client = SlowClient()
    recorder = EventRecorder(client, "c")
    t0 = _time.monotonic()
    recorder.event(mk_obj(), "Normal", "Reason", "m")
    assert _time.monotonic() - t0 < 0.5  # did not block on the slow create
    gate.set()
    assert recorder.flush()
    assert len(list_events(client)) == 1


def test_metrics_counters_observable():
    from prometheus_client import REGISTRY

    from agac import metrics

    metrics.observe_reconcile("test-queue-xyz", "success", 0.01)
    metrics.observe_aws_call("elbv2-test", "DescribeLoadBalancers")
    value = REGISTRY.get_sample_value(
        "agac_reconcile_total", {"queue": "test-queue-xyz", "outcome": "success"}
    )
    assert value == 1.0
    value = REGISTRY.get_sample_value(
        "agac_aws_api_calls_total",
        {"service": "elbv2-test", "operation": "DescribeLoadBalancers"},
    )
    assert value == 1.0


def test_workqueue_depth_gauge():
    from prometheus_client import REGISTRY

    from agac.kube.workqueue import ItemExponentialFailureRateLimiter, RateLimitingQueue

    q = RateLimitingQueue(
        rate_limiter=ItemExponentialFailureRateLimiter(0.001, 0.01),
        name="depth-test-q",
    )
    q.add("a")
    q.add("b")
    assert REGISTRY.get_sample_value("agac_workqueue_depth", {"queue": "depth-test-q"}) == 2.0
    q.get()
    assert REGISTRY.get_sample_value("agac_workqueue_depth", {"queue": "depth-test-q"}) == 1.0


def test_workqueue_latency_metrics_observed():
    """client-go-style queue metrics: queue wait, work duration, retries."""
    prometheus_client = __import__("pytest").importorskip("prometheus_client")
    from agac import metrics
    from agac.kube.workqueue import RateLimitingQueue

    q = RateLimitingQueue(name="metrics-test-queue")

    def hist_count(metric):
        return metric.labels(queue="metrics-test-queue")._sum.get()

    q.add("a")
    item, shutdown = q.get(timeout=1.0)
    assert item == "a" and not shutdown
    q.done("a")
    q.add_rate_limited("b")
    item, _ = q.get(timeout=2.0)
    q.done(item)
    assert hist_count(metrics.WORKQUEUE_QUEUE_DURATION) >= 0.0
    assert hist_count(metrics.WORKQUEUE_WORK_DURATION) >= 0.0
    retries = metrics.WORKQUEUE_RETRIES.labels(queue="metrics-test-queue")
    assert retries._value.get() >= 1


def test_recorder_threads_do_not_leak_across_manager_lifecycles():
    """Leadership churn restarts the manager; recorder writer threads must
    terminate with their controller instead of accumulating."""
    import threading
    import time

    from agac.cloudprovider.aws.client import FakeCloudFactory
    from agac.cloudprovider.fake import FakeAWSBackend
    from agac.manager import ControllerConfig, Manager

    def cycle():
        client = InMemoryKubeClient()
        stop = threading.Event()
        manager = Manager()
        manager.run(client, ControllerConfig(), FakeCloudFactory(FakeAWSBackend()),
                    stop, resync_period=300.0, block=False)
        assert manager.wait_until_ready()
        stop.set()
        time.sleep(0.05)

    cycle()  # warm-up lifecycle (imports, etc.)
    time.sleep(0.3)
    baseline = threading.active_count()
    for _ in range(4):
        cycle()
    deadline = time.monotonic() + 15
    while threading.active_count() > baseline + 10:
        if time.monotonic() > deadline:
            break
        time.sleep(0.05)
    # workers exit on queue shutdown, informer loops on stop, recorder
    # writers on the sentinel: 4 extra lifecycles spawn ~12 threads EACH
    # (~48 if leaked), so a +10 margin still detects any real leak while
    # tolerating slow-to-exit watch polls under parallel test load
    assert threading.active_count() <= baseline + 10, (
        f"thread leak: baseline {baseline}, now {threading.active_count()}"
    )
