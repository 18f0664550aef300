"""End-to-end controller integration: the full Manager (all three
controllers + shared informers) running against the in-memory kube API and
the stateful AWS fake.  This is the hermetic analogue of the reference's
local_e2e suite (BASELINE.json configs 1-4)."""

import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis import endpointgroupbinding as egb
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws import types as t
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.controller.endpointgroupbinding import EndpointGroupBindingController
from agac.kube.client import InMemoryKubeClient
from agac.manager import ControllerConfig, Manager

REGION = "us-east-1"
CLUSTER = "default"
MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
HOSTNAME_ANN = "aws-global-accelerator-controller.h3poteto.dev/route53-hostname"
LB_TYPE_ANN = "service.beta.kubernetes.io/aws-load-balancer-type"


def wait_until(pred, timeout=10.0, interval=0.02):
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
def env(monkeypatch):
    monkeypatch.setattr(EndpointGroupBindingController, "delete_drain_requeue", 0.02)
    client = InMemoryKubeClient()
    backend = FakeAWSBackend()
    factory = FakeCloudFactory(
        backend, lb_not_active_retry=0.05, ga_missing_retry=0.05
    )
    stop = threading.Event()
    manager = Manager()
    manager.run(
        client,
        ControllerConfig(),
        factory,
        stop,
        resync_period=0.5,
        block=False,
    )
    assert manager.wait_until_ready()
    yield client, backend, manager
    stop.set()


def mk_lb_service(backend, name="web", ns="default", annotations=None, lb_state="active"):
    lb = backend.elbv2.create_load_balancer(name, region=REGION, state=lb_state)
    svc = corev1.Service(
        metadata=ObjectMeta(
            name=name,
            namespace=ns,
            annotations={LB_TYPE_ANN: "nlb", **(annotations or {})},
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
    )
    return svc, lb


def list_accelerators(backend):
    accs, _ = backend.ga.list_accelerators()
    return accs


def events(client, reason):
    items, _ = client.list("Event")
    return [e for e in items if e.reason == reason]


class TestGlobalAcceleratorServicePath:
    def test_service_creates_accelerator_triple(self, env):
        client, backend, _ = env
        svc, lb = mk_lb_service(backend, annotations={MANAGED: "true"})
        client.create(svc)
        assert wait_until(lambda: len(list_accelerators(backend)) == 1)
        acc = list_accelerators(backend)[0]
        assert acc.name == "service-default-web"
        tags = {x.key: x.value for x in backend.ga.list_tags_for_resource(acc.accelerator_arn)}
        assert tags["aws-global-accelerator-owner"] == "service/default/web"
        assert tags["aws-global-accelerator-cluster"] == CLUSTER
        assert wait_until(lambda: len(events(client, "GlobalAcceleratorCreated")) == 1)

    def test_service_without_annotation_ignored(self, env):
        client, backend, _ = env
        svc, _ = mk_lb_service(backend)
        client.create(svc)
        time.sleep(0.3)
        assert list_accelerators(backend) == []

    def test_annotation_removal_deletes_accelerator(self, env):
        client, backend, _ = env
        svc, _ = mk_lb_service(backend, annotations={MANAGED: "true"})
        client.create(svc)
        assert wait_until(lambda: len(list_accelerators(backend)) == 1)
        stored = client.get("Service", "default", "web")
        del stored.metadata.annotations[MANAGED]
        client.update(stored)
        assert wait_until(lambda: list_accelerators(backend) == [])
        assert wait_until(lambda: len(events(client, "GlobalAcceleratorDeleted")) >= 1)

    def test_service_delete_cleans_up(self, env):
        client, backend, _ = env
        svc, _ = mk_lb_service(backend, annotations={MANAGED: "true"})
        client.create(svc)
        assert wait_until(lambda: len(list_accelerators(backend)) == 1)
        client.delete("Service", "default", "web")
        assert wait_until(lambda: list_accelerators(backend) == [])

    def test_provisioning_lb_retries_until_active(self, env):
        client, backend, _ = env
        svc, _ = mk_lb_service(
            backend, annotations={MANAGED: "true"}, lb_state="provisioning"
        )
        client.create(svc)
        time.sleep(0.2)
        assert list_accelerators(backend) == []
        backend.elbv2.set_state("web", "active")
        # the 0.05s retry requeue picks it up without any new k8s event
        assert wait_until(lambda: len(list_accelerators(backend)) == 1)

    def test_port_change_propagates_to_listener(self, env):
        client, backend, _ = env
        svc, _ = mk_lb_service(backend, annotations={MANAGED: "true"})
        client.create(svc)
        assert wait_until(lambda: len(list_accelerators(backend)) == 1)
        stored = client.get("Service", "default", "web")
        stored.spec.ports.append(corev1.ServicePort(port=443, protocol="TCP"))
        client.update(stored)

        def listener_ports():
            acc = list_accelerators(backend)[0]
            listeners, _ = backend.ga.list_listeners(acc.accelerator_arn)
            return sorted(p.from_port for p in listeners[0].port_ranges)

        assert wait_until(lambda: listener_ports() == [80, 443])


class TestGlobalAcceleratorIngressPath:
    def test_alb_ingress_creates_accelerator(self, env):
        client, backend, _ = env
        lb = backend.elbv2.create_load_balancer(
            "myingress", region=REGION, lb_type="application"
        )
        ingress = corev1.Ingress(
            metadata=ObjectMeta(
                name="ing",
                namespace="default",
                annotations={
                    MANAGED: "true",
                    "alb.ingress.kubernetes.io/listen-ports": '[{"HTTP": 80}]',
                },
            ),
            spec=corev1.IngressSpec(ingress_class_name="alb"),
            status=corev1.IngressStatus(
                load_balancer=corev1.IngressLoadBalancerStatus(
                    ingress=[corev1.IngressLoadBalancerIngress(hostname=lb.dns_name)]
                )
            ),
        )
        client.create(ingress)

        def accelerator_with_listener():
            accs = list_accelerators(backend)
            if len(accs) != 1:
                return False
            ls, _ = backend.ga.list_listeners(accs[0].accelerator_arn)
            return len(ls) == 1  # the triple is created in steps

        assert wait_until(accelerator_with_listener)
        acc = list_accelerators(backend)[0]
        assert acc.name == "ingress-default-ing"
        listeners, _ = backend.ga.list_listeners(acc.accelerator_arn)
        assert [p.from_port for p in listeners[0].port_ranges] == [80]


class TestRoute53Path:
    def test_hostname_annotation_creates_records(self, env):
        client, backend, _ = env
        backend.route53.create_hosted_zone("example.com")
        svc, _ = mk_lb_service(
            backend,
            annotations={MANAGED: "true", HOSTNAME_ANN: "www.example.com"},
        )
        client.create(svc)

        def records():
            zones, _ = backend.route53.list_hosted_zones()
            recs, _ = backend.route53.list_resource_record_sets(zones[0].id)
            return {(r.name, r.type) for r in recs}

        # route53 controller waits (ga_missing_retry=0.05) until the GA
        # controller has created the accelerator, then creates both records
        assert wait_until(
            lambda: records()
            == {("www.example.com.", "A"), ("www.example.com.", "TXT")}
        )

    def test_annotation_removal_deletes_records(self, env):
        client, backend, _ = env
        backend.route53.create_hosted_zone("example.com")
        svc, _ = mk_lb_service(
            backend,
            annotations={MANAGED: "true", HOSTNAME_ANN: "www.example.com"},
        )
        client.create(svc)

        def record_count():
            zones, _ = backend.route53.list_hosted_zones()
            recs, _ = backend.route53.list_resource_record_sets(zones[0].id)
            return len(recs)

        assert wait_until(lambda: record_count() == 2)
        stored = client.get("Service", "default", "web")
        del stored.metadata.annotations[HOSTNAME_ANN]
        client.update(stored)
        assert wait_until(lambda: record_count() == 0)


class TestEndpointGroupBindingPath:
    def seed_group(self, backend):
        acc = backend.ga.create_accelerator("external")
        listener = backend.ga.create_listener(
            acc.accelerator_arn, [t.PortRange(80, 80)], "TCP"
        )
        return backend.ga.create_endpoint_group(listener.listener_arn, REGION)

    def test_binding_lifecycle(self, env):
        client, backend, _ = env
        group = self.seed_group(backend)
        svc, lb = mk_lb_service(backend, name="bound")
        client.create(svc)
        binding = egb.EndpointGroupBinding(
            metadata=ObjectMeta(name="b1", namespace="default"),
            spec=egb.EndpointGroupBindingSpec(
                endpoint_group_arn=group.endpoint_group_arn,
                weight=50,
                service_ref=egb.ServiceReference(name="bound"),
            ),
        )
        client.create(binding)

        # finalizer installed, endpoint attached, status synced
        assert wait_until(
            lambda: client.get("EndpointGroupBinding", "default", "b1").metadata.finalizers
            == ["operator.h3poteto.dev/endpointgroupbindings"]
        )
        assert wait_until(
            lambda: [
                d.endpoint_id
                for d in backend.ga.describe_endpoint_group(
                    group.endpoint_group_arn
                ).endpoint_descriptions
            ]
            == [lb.load_balancer_arn]
        )
        assert wait_until(
            lambda: client.get(
                "EndpointGroupBinding", "default", "b1"
            ).status.endpoint_ids
            == [lb.load_balancer_arn]
        )
        desc = backend.ga.describe_endpoint_group(group.endpoint_group_arn)
        assert desc.endpoint_descriptions[0].weight == 50

        # weight update propagates (spec change bumps generation)
        stored = client.get("EndpointGroupBinding", "default", "b1")
        stored.spec.weight = 200
        client.update(stored)
        assert wait_until(
            lambda: backend.ga.describe_endpoint_group(group.endpoint_group_arn)
            .endpoint_descriptions[0]
            .weight
            == 200
        )

        # delete drains the endpoint then removes the finalizer + object
        client.delete("EndpointGroupBinding", "default", "b1")
        assert wait_until(
            lambda: backend.ga.describe_endpoint_group(
                group.endpoint_group_arn
            ).endpoint_descriptions
            == []
        )

        def gone():
            try:
                client.get("EndpointGroupBinding", "default", "b1")
                return False
            except Exception:
                return True

        assert wait_until(gone)

    def test_binding_with_missing_endpoint_group_unblocks_delete(self, env):
        client, backend, _ = env
        group = self.seed_group(backend)
        svc, lb = mk_lb_service(backend, name="bound2")
        client.create(svc)
        binding = egb.EndpointGroupBinding(
            metadata=ObjectMeta(name="b2", namespace="default"),
            spec=egb.EndpointGroupBindingSpec(
                endpoint_group_arn=group.endpoint_group_arn,
                service_ref=egb.ServiceReference(name="bound2"),
            ),
        )
        client.create(binding)
        assert wait_until(
            lambda: client.get("EndpointGroupBinding", "default", "b2").status.endpoint_ids
            == [lb.load_balancer_arn]
        )
        # the endpoint group disappears out from under us
        backend.ga.delete_endpoint_group(group.endpoint_group_arn)
        client.delete("EndpointGroupBinding", "default", "b2")

        def gone():
            try:
                client.get("EndpointGroupBinding", "default", "b2")
                return False
            except Exception:
                return True

        assert wait_until(gone)


class TestEventReasonParity:
    def test_service_route53_event_keeps_reference_typo(self, env):
        """r53/service.go:105 emits reason 'Route53RecourdCreated' (sic);
        the ingress path uses the corrected spelling.  Both are observable
        API surface and must match the reference byte-for-byte."""
        client, backend, _ = env
        backend.route53.create_hosted_zone("example.com")
        svc, _ = mk_lb_service(
            backend,
            name="typoed",
            annotations={MANAGED: "true", HOSTNAME_ANN: "t.example.com"},
        )
        client.create(svc)
        assert wait_until(
            lambda: any(
                e.reason == "Route53RecourdCreated"
                for e in client.list("Event")[0]
            )
        )
        reasons = {e.reason for e in client.list("Event")[0]}
        assert "Route53RecordCreated" not in reasons  # that's the ingress path
