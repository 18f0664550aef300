"""Full-lifecycle e2e scenarios (the hermetic analogue of the reference's
local_e2e suite + the kind webhook e2e):

1. kind-e2e analogue: the webhook sits in the API store's admission path
   over real HTTP; ARN updates are rejected with 403, weight updates pass
   (reference e2e/e2e_test.go:77-103).
2. local_e2e analogue: NLB service → GA triple + Route53 records appear and
   are cleaned up end-to-end, controllers + webhook all live.
3. HA: two manager replicas under leader election; killing the leader fails
   over and reconciliation continues (BASELINE.json config 5).
"""

import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis import endpointgroupbinding as egb
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.fixture import endpoint_group_binding
from agac.kube.admission import AdmissionDeniedError, http_admission
from agac.kube.httpapi import APIServer
from agac.kube.leaderelection import LeaderElectionConfig, LeaderElector
from agac.kube.rest import RestKubeClient
from agac.kube.store import APIStore
from agac.manager import ControllerConfig, Manager
from agac.webhook.server import WebhookServer

MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
HOSTNAME_ANN = "aws-global-accelerator-controller.h3poteto.dev/route53-hostname"
LB_TYPE = "service.beta.kubernetes.io/aws-load-balancer-type"


def wait_until(pred, timeout=20.0, interval=0.02):
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
def stack():
    """API server + webhook wired into admission, REST client, AWS fake."""
    store = APIStore()
    webhook_server = WebhookServer(port=0)
    webhook_server.start()
    store.admission_webhooks.append(
        http_admission(
            kinds=["EndpointGroupBinding"],
            operations=["UPDATE"],
            url=f"http://127.0.0.1:{webhook_server.port}/validate-endpointgroupbinding",
        )
    )
    api = APIServer(store)
    api.start()
    client = RestKubeClient(api.url)
    backend = FakeAWSBackend()
    yield client, backend, api
    api.shutdown()
    webhook_server.shutdown()


def mk_lb_service(backend, name, annotations):
    lb = backend.elbv2.create_load_balancer(name, region="us-east-1")
    return (
        corev1.Service(
            metadata=ObjectMeta(name=name, namespace="default", annotations=annotations),
            spec=corev1.ServiceSpec(
                type="LoadBalancer",
                ports=[corev1.ServicePort(port=80, protocol="TCP")],
            ),
            status=corev1.ServiceStatus(
                load_balancer=corev1.LoadBalancerStatus(
                    ingress=[corev1.LoadBalancerIngress(hostname=lb.dns_name)]
                )
            ),
        ),
        lb,
    )


class TestWebhookAdmissionPath:
    def test_arn_update_rejected_weight_allowed(self, stack):
        client, _, _ = stack
        binding = endpoint_group_binding(name="guarded")
        client.create(binding)

        stored = client.get("EndpointGroupBinding", "default", "guarded")
        stored.spec.weight = 7
        updated = client.update(stored)  # weight change allowed
        assert updated.spec.weight == 7

        stored = client.get("EndpointGroupBinding", "default", "guarded")
        stored.spec.endpoint_group_arn = "arn:aws:globalaccelerator::1:tampered"
        with pytest.raises(AdmissionDeniedError, match="immutable"):
            client.update(stored)
        # object unchanged
        again = client.get("EndpointGroupBinding", "default", "guarded")
        assert again.spec.endpoint_group_arn == binding.spec.endpoint_group_arn


class TestFullLifecycle:
    def test_service_to_ga_and_route53_and_cleanup(self, stack):
        client, backend, _ = stack
        backend.route53.create_hosted_zone("example.com")
        stop = threading.Event()
        manager = Manager()
        manager.run(
            client,
            ControllerConfig(),
            FakeCloudFactory(backend, ga_missing_retry=0.05),
            stop,
            resync_period=2.0,
            block=False,
        )
        try:
            assert manager.wait_until_ready()
            svc, lb = mk_lb_service(
                backend,
                "prod",
                {LB_TYPE: "nlb", MANAGED: "true", HOSTNAME_ANN: "www.example.com"},
            )
            client.create(svc)

            # GA triple appears (local_e2e waitUntilGlobalAccelerator analogue)
            def full_triple():
                accs, _ = backend.ga.list_accelerators()
                if len(accs) != 1:
                    return False
                ls, _ = backend.ga.list_listeners(accs[0].accelerator_arn)
                if len(ls) != 1:
                    return False
                gs, _ = backend.ga.list_endpoint_groups(ls[0].listener_arn)
                return len(gs) == 1

            # wait for the whole triple: the accelerator appears before its
            # listener/endpoint group mid-create
            assert wait_until(full_triple)
            acc = backend.ga.list_accelerators()[0][0]
            listeners, _ = backend.ga.list_listeners(acc.accelerator_arn)
            assert [p.from_port for p in listeners[0].port_ranges] == [80]
            groups, _ = backend.ga.list_endpoint_groups(listeners[0].listener_arn)
            assert groups[0].endpoint_descriptions[0].endpoint_id == lb.load_balancer_arn

            # Route53 records appear (waitUntilRoute53 analogue)
            def records():
                zones, _ = backend.route53.list_hosted_zones()
                recs, _ = backend.route53.list_resource_record_sets(zones[0].id)
                return {(r.name, r.type) for r in recs}

            assert wait_until(
                lambda: records()
                == {("www.example.com.", "A"), ("www.example.com.", "TXT")}
            )

            # deletion cleans everything up (waitUntilCleanup analogue)
            client.delete("Service", "default", "prod")
            assert wait_until(lambda: backend.ga.list_accelerators()[0] == [])
            assert wait_until(lambda: records() == set())
        finally:
            stop.set()


class TestLeaderFailoverWithReconcile:
    def _replica(self, client, backend, identity):
        """A controller replica: leader election wrapping a manager, like
        cmd/controller/controller.go:73-80."""
        stop = threading.Event()
        holder = {}

        def run_manager(stop_leading):
            manager = Manager()
            manager.run(
                client,
                ControllerConfig(),
                FakeCloudFactory(backend),
                stop_leading,
                resync_period=1.0,
                block=True,
            )

        elector = LeaderElector(
            client,
            name="aws-global-accelerator-controller",
            namespace="default",
            identity=identity,
            config=LeaderElectionConfig(
                lease_duration=0.6, renew_deadline=0.4, retry_period=0.05
            ),
            on_started_leading=run_manager,
        )
        thread = threading.Thread(target=elector.run, args=(stop,), daemon=True)
        thread.start()
        return elector, stop, thread

    def test_failover_keeps_reconciling(self, stack):
        client, backend, _ = stack
        r1, stop1, t1 = self._replica(client, backend, "replica-1")
        r2, stop2, t2 = self._replica(client, backend, "replica-2")
        try:
            assert wait_until(lambda: r1.is_leader.is_set() or r2.is_leader.is_set())
            leader, follower = (r1, r2) if r1.is_leader.is_set() else (r2, r1)
            assert not follower.is_leader.is_set()

            svc, _ = mk_lb_service(backend, "ha-1", {LB_TYPE: "nlb", MANAGED: "true"})
            client.create(svc)
            assert wait_until(lambda: len(backend.ga.list_accelerators()[0]) == 1)

            # kill the leader; the follower must take over
            (stop1 if leader is r1 else stop2).set()
            assert wait_until(lambda: follower.is_leader.is_set(), timeout=10.0)

            # and reconciliation continues under the new leader
            svc2, _ = mk_lb_service(backend, "ha-2", {LB_TYPE: "nlb", MANAGED: "true"})
            client.create(svc2)
            assert wait_until(lambda: len(backend.ga.list_accelerators()[0]) == 2)
        finally:
            stop1.set()
            stop2.set()
            t1.join(timeout=5)
            t2.join(timeout=5)
