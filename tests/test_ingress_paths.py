"""Ingress-flavored controller paths not covered by the main integration
suite: Route53 for ALB Ingresses, EndpointGroupBinding via ingressRef,
multi-hostname annotations, internal-ALB hostname parsing end-to-end."""

import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis import endpointgroupbinding as egb
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws import types as t
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.kube.client import InMemoryKubeClient
from agac.manager import ControllerConfig, Manager

MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
HOSTNAME_ANN = "aws-global-accelerator-controller.h3poteto.dev/route53-hostname"


def wait_until(pred, timeout=15.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        try:
            if pred():
                return True
        except Exception:
            pass
        time.sleep(0.02)
    return False


@pytest.fixture
def env():
    client = InMemoryKubeClient()
    backend = FakeAWSBackend()
    factory = FakeCloudFactory(backend, ga_missing_retry=0.05)
    stop = threading.Event()
    manager = Manager()
    manager.run(client, ControllerConfig(), factory, stop, resync_period=0.5, block=False)
    assert manager.wait_until_ready()
    yield client, backend
    stop.set()


def mk_alb_ingress(backend, name="ing", annotations=None, scheme="internet-facing"):
    lb = backend.elbv2.create_load_balancer(
        name, region="ap-northeast-1", lb_type="application", scheme=scheme
    )
    ingress = corev1.Ingress(
        metadata=ObjectMeta(
            name=name,
            namespace="default",
            annotations={
                "alb.ingress.kubernetes.io/listen-ports": '[{"HTTP": 80}]',
                **(annotations or {}),
            },
        ),
        spec=corev1.IngressSpec(ingress_class_name="alb"),
        status=corev1.IngressStatus(
            load_balancer=corev1.IngressLoadBalancerStatus(
                ingress=[corev1.IngressLoadBalancerIngress(hostname=lb.dns_name)]
            )
        ),
    )
    return ingress, lb


def zone_records(backend, zone):
    recs, _ = backend.route53.list_resource_record_sets(zone.id)
    return {(r.name, r.type) for r in recs}


class TestRoute53IngressPath:
    def test_ingress_hostnames_reconciled(self, env):
        client, backend = env
        zone = backend.route53.create_hosted_zone("example.com")
        ingress, _ = mk_alb_ingress(
            backend,
            annotations={
                MANAGED: "true",
                HOSTNAME_ANN: "app.example.com,api.example.com",
            },
        )
        client.create(ingress)
        assert wait_until(
            lambda: zone_records(backend, zone)
            == {
                ("app.example.com.", "A"),
                ("app.example.com.", "TXT"),
                ("api.example.com.", "A"),
                ("api.example.com.", "TXT"),
            }
        )
        # TXT ownership value carries resource=ingress
        recs, _ = backend.route53.list_resource_record_sets(zone.id)
        txt = next(r for r in recs if r.type == "TXT")
        assert "ingress/default/ing" in txt.resource_records[0].value
        # reason parity: ingress path uses the non-typo reason
        events, _ = client.list("Event")
        assert wait_until(
            lambda: any(
                e.reason == "Route53RecordCreated" for e in client.list("Event")[0]
            )
        )

    def test_internal_alb_hostname_end_to_end(self, env):
        client, backend = env
        ingress, lb = mk_alb_ingress(
            backend, name="internal-ing",
            annotations={MANAGED: "true"}, scheme="internal",
        )
        assert lb.dns_name.startswith("internal-")
        client.create(ingress)
        assert wait_until(lambda: len(backend.ga.list_accelerators()[0]) == 1)
        acc = backend.ga.list_accelerators()[0][0]
        tags = {x.key: x.value for x in backend.ga.list_tags_for_resource(acc.accelerator_arn)}
        assert tags["aws-global-accelerator-owner"] == "ingress/default/internal-ing"

    def test_hostname_annotation_removed_cleans_ingress_records(self, env):
        client, backend = env
        zone = backend.route53.create_hosted_zone("example.com")
        ingress, _ = mk_alb_ingress(
            backend,
            annotations={MANAGED: "true", HOSTNAME_ANN: "app.example.com"},
        )
        client.create(ingress)
        assert wait_until(lambda: len(zone_records(backend, zone)) == 2)
        stored = client.get("Ingress", "default", "ing")
        del stored.metadata.annotations[HOSTNAME_ANN]
        client.update(stored)
        assert wait_until(lambda: zone_records(backend, zone) == set())


class TestEGBIngressRef:
    def test_binding_via_ingress_ref(self, env):
        client, backend = env
        acc = backend.ga.create_accelerator("ext")
        listener = backend.ga.create_listener(
            acc.accelerator_arn, [t.PortRange(80, 80)], "TCP"
        )
        group = backend.ga.create_endpoint_group(listener.listener_arn, "ap-northeast-1")
        ingress, lb = mk_alb_ingress(backend, name="bound-ing")
        client.create(ingress)
        client.create(
            egb.EndpointGroupBinding(
                metadata=ObjectMeta(name="b-ing", namespace="default"),
                spec=egb.EndpointGroupBindingSpec(
                    endpoint_group_arn=group.endpoint_group_arn,
                    ingress_ref=egb.IngressReference(name="bound-ing"),
                ),
            )
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


class TestMultiLBService:
    def test_service_with_two_lb_ingress_entries(self, env):
        """A Service whose status carries two LB hostnames gets the triple
        ensured per hostname (reference loops svc.Status.LoadBalancer.Ingress)."""
        client, backend = env
        lb1 = backend.elbv2.create_load_balancer("multi-a", region="us-east-1")
        lb2 = backend.elbv2.create_load_balancer("multi-b", region="us-east-1")
        svc = corev1.Service(
            metadata=ObjectMeta(
                name="multi",
                namespace="default",
                annotations={
                    "service.beta.kubernetes.io/aws-load-balancer-type": "nlb",
                    MANAGED: "true",
                },
            ),
            spec=corev1.ServiceSpec(
                type="LoadBalancer",
                ports=[corev1.ServicePort(port=80, protocol="TCP")],
            ),
            status=corev1.ServiceStatus(
                load_balancer=corev1.LoadBalancerStatus(
                    ingress=[
                        corev1.LoadBalancerIngress(hostname=lb1.dns_name),
                        corev1.LoadBalancerIngress(hostname=lb2.dns_name),
                    ]
                )
            ),
        )
        client.create(svc)
        # the reference's semantics: ONE accelerator per owning resource —
        # the second hostname updates the same accelerator (tags match by
        # owner), repointing the endpoint at lb2
        assert wait_until(lambda: len(backend.ga.list_accelerators()[0]) == 1)
