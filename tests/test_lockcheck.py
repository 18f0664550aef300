"""The lock-order checker itself (agac/lockcheck.py): detects inversions,
ignores re-entrancy and consistent ordering, and the real framework runs
clean under it (the `make test-race` tier)."""

import threading

import pytest

from agac import lockcheck


@pytest.fixture
def checker():
    # preserve any session-wide instrumentation (AGAC_LOCKCHECK=1 installs
    # in conftest): snapshot the global graph, run this test isolated, then
    # restore — so these self-tests never weaken the session-level check
    was_installed = lockcheck._installed
    prior_edges = dict(lockcheck._edges)
    lockcheck.reset()
    lockcheck.install()
    yield lockcheck
    lockcheck.check()  # note: tests that EXPECT a cycle reset before exit
    if not was_installed:
        lockcheck.uninstall()
    lockcheck.reset()
    with lockcheck._graph_lock:
        lockcheck._edges.update(prior_edges)


def test_consistent_order_is_clean(checker):
    a = threading.Lock()
    b = threading.Lock()
    for _ in range(3):
        with a:
            with b:
                pass
    checker.check()  # a->b only, no cycle


def test_inversion_is_detected(checker):
    a = threading.Lock()
    b = threading.Lock()
    with a:
        with b:
            pass
    with b:
        with a:
            pass
    with pytest.raises(lockcheck.LockOrderViolation):
        checker.check()
    checker.reset()


def test_three_way_cycle_detected(checker):
    a = threading.Lock()
    b = threading.Lock()
    c = threading.Lock()
    with a:
        with b:
            pass
    with b:
        with c:
            pass
    with c:
        with a:
            pass
    with pytest.raises(lockcheck.LockOrderViolation):
        checker.check()
    checker.reset()


def test_rlock_reentrancy_is_not_an_edge(checker):
    r = threading.RLock()
    with r:
        with r:  # re-entrant, legal
            pass
    checker.check()
    assert not any(site == site2 for site, site2 in checker.edges())


def test_cross_thread_orders_merge(checker):
    """Thread 1 takes a->b, thread 2 takes b->a: neither thread alone
    deadlocks, but together they can — the checker sees the merged graph."""
    a = threading.Lock()
    b = threading.Lock()

    def t1():
        with a:
            with b:
                pass

    def t2():
        with b:
            with a:
                pass

    th1 = threading.Thread(target=t1)
    th1.start()
    th1.join()
    th2 = threading.Thread(target=t2)
    th2.start()
    th2.join()
    with pytest.raises(lockcheck.LockOrderViolation):
        checker.check()
    checker.reset()


def test_framework_runs_clean_under_lockcheck(checker):
    """A full three-controller reconcile pass (store, informers, queues,
    backend, recorder) produces an acyclic lock-order graph."""
    import time

    from agac.apis import core as corev1
    from agac.apis.meta import ObjectMeta
    from agac.cloudprovider.aws.client import FakeCloudFactory
    from agac.cloudprovider.fake import FakeAWSBackend
    from agac.kube.client import InMemoryKubeClient
    from agac.manager import ControllerConfig, Manager

    backend = FakeAWSBackend()
    client = InMemoryKubeClient()
    stop = threading.Event()
    manager = Manager()
    manager.run(client, ControllerConfig(), FakeCloudFactory(backend), stop,
                resync_period=0.2, block=False)
    try:
        assert manager.wait_until_ready()
        backend.route53.create_hosted_zone("lc.example.com")
        lb = backend.elbv2.create_load_balancer("lc", region="us-east-1")
        client.create(corev1.Service(
            metadata=ObjectMeta(
                name="lc", namespace="default",
                annotations={
                    "service.beta.kubernetes.io/aws-load-balancer-type": "nlb",
                    "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed": "true",
                    "aws-global-accelerator-controller.h3poteto.dev/route53-hostname": "app.lc.example.com",
                },
            ),
            spec=corev1.ServiceSpec(
                type="LoadBalancer",
                ports=[corev1.ServicePort(port=80, protocol="TCP")],
            ),
            status=corev1.ServiceStatus(
                load_balancer=corev1.LoadBalancerStatus(
                    ingress=[corev1.LoadBalancerIngress(hostname=lb.dns_name)]
                )
            ),
        ))
        deadline = time.monotonic() + 15
        while not backend.ga.list_accelerators()[0]:
            assert time.monotonic() < deadline
            time.sleep(0.02)
        time.sleep(0.5)  # let a couple of resync sweeps run too
        checker.check()
        # the checker was really active: the framework's own locks are
        # wrapped (the backend lock is created in agac code post-install)
        from agac.lockcheck import _CheckedLock

        assert isinstance(backend.lock, _CheckedLock)
        # an empty edge set is itself the finding: no agac code path holds
        # two project locks at once, so no ordering can invert
    finally:
        stop.set()
