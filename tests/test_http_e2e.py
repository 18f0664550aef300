"""e2e over a real network boundary: the controller manager + informers +
leader election running through RestKubeClient against the HTTP API server
(the hermetic analogue of the reference's kind e2e tier)."""

import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.kube.httpapi import APIServer
from agac.kube.rest import RestKubeClient
from agac.kube.store import APIStore, ConflictError, NotFoundError
from agac.manager import ControllerConfig, Manager

MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
LB_TYPE = "service.beta.kubernetes.io/aws-load-balancer-type"


def wait_until(pred, timeout=15.0, interval=0.02):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        try:
            if pred():
                return True
        except Exception:
            pass
        time.sleep(interval)
    return False


@pytest.fixture
def api():
    server = APIServer(APIStore())
    server.start()
    yield server
    server.shutdown()


@pytest.fixture
def client(api):
    return RestKubeClient(api.url)


def mk_service(name="web", annotations=None, hostname="x-123abc.elb.us-east-1.amazonaws.com"):
    return corev1.Service(
        metadata=ObjectMeta(
            name=name, namespace="default", annotations=annotations or {}
        ),
        spec=corev1.ServiceSpec(
            type="LoadBalancer",
            ports=[corev1.ServicePort(port=80, protocol="TCP")],
        ),
        status=corev1.ServiceStatus(
            load_balancer=corev1.LoadBalancerStatus(
                ingress=[corev1.LoadBalancerIngress(hostname=hostname)]
            )
        ),
    )


class TestRestCRUD:
    def test_create_get_roundtrip(self, client):
        created = client.create(mk_service())
        assert created.metadata.resource_version
        got = client.get("Service", "default", "web")
        assert got.spec.ports[0].port == 80
        assert got.status.load_balancer.ingress[0].hostname.startswith("x-123abc")

    def test_not_found_is_typed(self, client):
        with pytest.raises(NotFoundError):
            client.get("Service", "default", "ghost")

    def test_conflict_is_typed(self, client):
        client.create(mk_service())
        a = client.get("Service", "default", "web")
        b = client.get("Service", "default", "web")
        a.metadata.annotations["a"] = "1"
        client.update(a)
        b.metadata.annotations["b"] = "2"
        with pytest.raises(ConflictError):
            client.update(b)

    def test_update_status_via_subresource(self, client):
        client.create(mk_service())
        obj = client.get("Service", "default", "web")
        obj.status.load_balancer.ingress[0].hostname = "changed.elb.us-east-1.amazonaws.com"
        client.update_status(obj)
        got = client.get("Service", "default", "web")
        assert got.status.load_balancer.ingress[0].hostname.startswith("changed")

    def test_list_with_namespace_filter(self, client):
        client.create(mk_service("a"))
        svc_b = mk_service("b")
        svc_b.metadata.namespace = "other"
        client.create(svc_b)
        items, rv = client.list("Service")
        assert len(items) == 2 and rv > 0
        items, _ = client.list("Service", namespace="other")
        assert [o.metadata.name for o in items] == ["b"]

    def test_delete(self, client):
        client.create(mk_service())
        client.delete("Service", "default", "web")
        with pytest.raises(NotFoundError):
            client.get("Service", "default", "web")


class TestRestWatch:
    def test_watch_streams_events(self, client):
        _, rv = client.list("Service")
        watch = client.watch("Service", resource_version=rv)
        try:
            client.create(mk_service("streamed"))
            event = watch.get(timeout=10.0)
            assert event is not None
            assert event.type == "ADDED"
            assert event.obj.metadata.name == "streamed"

            obj = client.get("Service", "default", "streamed")
            obj.metadata.annotations["k"] = "v"
            client.update(obj)
            event = watch.get(timeout=10.0)
            assert event.type == "MODIFIED"

            client.delete("Service", "default", "streamed")
            event = watch.get(timeout=10.0)
            assert event.type == "DELETED"
        finally:
            watch.stop()


class TestManagerOverHTTP:
    def test_full_reconcile_over_the_wire(self, api, client):
        backend = FakeAWSBackend()
        factory = FakeCloudFactory(backend)
        stop = threading.Event()
        manager = Manager()
        manager.run(client, ControllerConfig(), factory, stop, resync_period=1.0, block=False)
        try:
            assert manager.wait_until_ready()
            lb = backend.elbv2.create_load_balancer("wired", region="us-east-1")
            client.create(
                mk_service(
                    "wired",
                    annotations={LB_TYPE: "nlb", MANAGED: "true"},
                    hostname=lb.dns_name,
                )
            )
            assert wait_until(lambda: len(backend.ga.list_accelerators()[0]) == 1)
            acc = backend.ga.list_accelerators()[0][0]
            assert acc.name == "service-wired-wired" or acc.name == "service-default-wired"
            # deletion over the wire tears the accelerator down
            client.delete("Service", "default", "wired")
            assert wait_until(lambda: backend.ga.list_accelerators()[0] == [])
        finally:
            stop.set()

    def test_leader_election_over_http(self, client):
        from agac.kube.leaderelection import LeaderElectionConfig, LeaderElector

        config = LeaderElectionConfig(
            lease_duration=0.5, renew_deadline=0.3, retry_period=0.05
        )
        elector = LeaderElector(client, "agac-leader", "default", identity="r1", config=config)
        stop = threading.Event()
        t = threading.Thread(target=elector.run, args=(stop,), daemon=True)
        t.start()
        assert wait_until(lambda: elector.is_leader.is_set())
        lease = client.get("Lease", "default", "agac-leader")
        assert lease.spec.holder_identity == "r1"
        stop.set()
        t.join(timeout=5.0)
