"""Drop-in takeover: resources created by the *reference* controller (same
tag schema / TXT ownership format, byte-for-byte) must be ADOPTED by agac —
no duplicate accelerators, no Route53 record churn, drift repaired in
place.  This is the interoperability claim in docs/PARITY.md made
executable."""

import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws import types as t
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.kube.client import InMemoryKubeClient
from agac.manager import ControllerConfig, Manager

REGION = "us-east-1"
CLUSTER = "prod-cluster"
MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
HOSTNAME_ANN = "aws-global-accelerator-controller.h3poteto.dev/route53-hostname"
LB_TYPE = "service.beta.kubernetes.io/aws-load-balancer-type"


def wait_until(pred, timeout=15.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(0.02)
    return pred()


def reference_created_triple(backend, lb, svc_ns, svc_name, cluster, ports=(80,)):
    """Create the accelerator→listener→endpoint-group triple exactly as the
    reference would have (tag schema from global_accelerator.go:24-28,
    :654-700)."""
    acc = backend.ga.create_accelerator(
        name=f"service-{svc_ns}-{svc_name}",
        ip_address_type="DUAL_STACK",
        enabled=True,
        tags=[
            t.Tag("aws-global-accelerator-controller-managed", "true"),
            t.Tag("aws-global-accelerator-owner", f"service/{svc_ns}/{svc_name}"),
            t.Tag("aws-global-accelerator-target-hostname", lb.dns_name),
            t.Tag("aws-global-accelerator-cluster", cluster),
        ],
    )
    listener = backend.ga.create_listener(
        acc.accelerator_arn,
        [t.PortRange(p, p) for p in ports],
        "TCP",
    )
    backend.ga.create_endpoint_group(
        listener.listener_arn,
        REGION,
        endpoint_configurations=[
            t.EndpointConfiguration(
                endpoint_id=lb.load_balancer_arn,
                client_ip_preservation_enabled=False,
            )
        ],
    )
    return acc


def reference_created_records(backend, zone, hostname, acc, cluster, svc_ns, svc_name):
    """TXT + A-alias pair exactly as the reference writes them
    (route53.go:18-20, :240-289)."""
    owner = (
        f'"heritage=aws-global-accelerator-controller,cluster={cluster},'
        f'service/{svc_ns}/{svc_name}"'
    )
    backend.route53.change_resource_record_sets(
        zone.id,
        [
            t.Change(
                action="CREATE",
                record_set=t.ResourceRecordSet(
                    name=hostname, type="TXT", ttl=300,
                    resource_records=[t.ResourceRecord(value=owner)],
                ),
            ),
            t.Change(
                action="CREATE",
                record_set=t.ResourceRecordSet(
                    name=hostname, type="A",
                    alias_target=t.AliasTarget(
                        dns_name=acc.dns_name,
                        evaluate_target_health=True,
                        hosted_zone_id="Z2BJ6XQ5FK7U4H",
                    ),
                ),
            ),
        ],
    )


def mk_service(lb, name, ns="default", annotations=None, ports=(80,)):
    return corev1.Service(
        metadata=ObjectMeta(
            name=name, namespace=ns,
            annotations={LB_TYPE: "nlb", MANAGED: "true", **(annotations or {})},
        ),
        spec=corev1.ServiceSpec(
            type="LoadBalancer",
            ports=[corev1.ServicePort(port=p, protocol="TCP") for p in ports],
        ),
        status=corev1.ServiceStatus(
            load_balancer=corev1.LoadBalancerStatus(
                ingress=[corev1.LoadBalancerIngress(hostname=lb.dns_name)]
            )
        ),
    )


@pytest.fixture
def stack():
    client = InMemoryKubeClient()
    backend = FakeAWSBackend()
    stop = threading.Event()
    manager = Manager()
    config = ControllerConfig()
    config.global_accelerator.cluster_name = CLUSTER
    config.route53.cluster_name = CLUSTER
    manager.run(
        client, config,
        FakeCloudFactory(backend, ga_missing_retry=0.05),
        stop, resync_period=0.3, block=False,
    )
    assert manager.wait_until_ready()
    yield client, backend
    stop.set()


def test_adopts_reference_accelerator_without_duplicate(stack):
    client, backend = stack
    lb = backend.elbv2.create_load_balancer("legacy", region=REGION)
    reference_created_triple(backend, lb, "default", "legacy", CLUSTER)
    backend.ga.call_counts.clear()

    client.create(mk_service(lb, "legacy"))
    # give the controller time to reconcile (synced = one accelerator, ports match)
    assert wait_until(
        lambda: backend.ga.call_counts.get("list_accelerators", 0) >= 1
    )
    time.sleep(0.3)
    accs, _ = backend.ga.list_accelerators()
    assert len(accs) == 1, "must adopt, not duplicate"
    assert backend.ga.call_counts.get("create_accelerator", 0) == 0


def test_repairs_drift_on_adopted_accelerator(stack):
    client, backend = stack
    lb = backend.elbv2.create_load_balancer("drifted", region=REGION)
    # reference left the listener on port 80; the service now wants 8443
    reference_created_triple(backend, lb, "default", "drifted", CLUSTER, ports=(80,))
    client.create(mk_service(lb, "drifted", ports=(8443,)))

    def listener_updated():
        accs, _ = backend.ga.list_accelerators()
        if len(accs) != 1:
            return False
        listeners, _ = backend.ga.list_listeners(accs[0].accelerator_arn)
        return [p.from_port for p in listeners[0].port_ranges] == [8443]

    assert wait_until(listener_updated)


def test_adopts_reference_route53_records_without_churn(stack):
    client, backend = stack
    zone = backend.route53.create_hosted_zone("legacy.example.com")
    lb = backend.elbv2.create_load_balancer("webby", region=REGION)
    acc = reference_created_triple(backend, lb, "default", "webby", CLUSTER)
    reference_created_records(
        backend, zone, "app.legacy.example.com", acc, CLUSTER, "default", "webby"
    )

    client.create(
        mk_service(lb, "webby", annotations={HOSTNAME_ANN: "app.legacy.example.com"})
    )
    # wait for at least one route53 reconcile to complete (an Event or a scan)
    assert wait_until(
        lambda: any(
            e.reason in ("Route53RecourdCreated",)
            for e in client.list("Event")[0]
        )
        is False
        and backend.ga.call_counts.get("list_accelerators", 0) >= 1,
        timeout=5.0,
    )
    time.sleep(0.5)
    # records unchanged: still exactly the TXT+A pair, alias still points at
    # the reference-created accelerator
    recs, _ = backend.route53.list_resource_record_sets(zone.id)
    assert len(recs) == 2
    alias = next(r for r in recs if r.type == "A")
    assert alias.alias_target.dns_name == acc.dns_name + "."
    # and no Created event was emitted (nothing was created)
    assert all(
        e.reason != "Route53RecourdCreated" for e in client.list("Event")[0]
    )


def test_adopts_reference_ingress_accelerator(stack):
    """Ingress-owned takeover: owner tag resource prefix 'ingress/'."""
    client, backend = stack
    lb = backend.elbv2.create_load_balancer(
        "legacy-ing", region=REGION, lb_type="application"
    )
    acc = backend.ga.create_accelerator(
        name="ingress-default-legacy-ing",
        tags=[
            t.Tag("aws-global-accelerator-controller-managed", "true"),
            t.Tag("aws-global-accelerator-owner", "ingress/default/legacy-ing"),
            t.Tag("aws-global-accelerator-target-hostname", lb.dns_name),
            t.Tag("aws-global-accelerator-cluster", CLUSTER),
        ],
    )
    listener = backend.ga.create_listener(
        acc.accelerator_arn, [t.PortRange(80, 80)], "TCP"
    )
    backend.ga.create_endpoint_group(
        listener.listener_arn, REGION,
        endpoint_configurations=[
            t.EndpointConfiguration(endpoint_id=lb.load_balancer_arn)
        ],
    )
    backend.ga.call_counts.clear()

    client.create(
        corev1.Ingress(
            metadata=ObjectMeta(
                name="legacy-ing", namespace="default",
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
    )
    assert wait_until(lambda: backend.ga.call_counts.get("list_accelerators", 0) >= 1)
    time.sleep(0.3)
    accs, _ = backend.ga.list_accelerators()
    assert len(accs) == 1
    assert backend.ga.call_counts.get("create_accelerator", 0) == 0
