"""Two controller instances with different --cluster-name sharing one AWS
account: the cluster ownership tag must keep their resources isolated
(each manages only its own accelerators/records; neither touches the
other's on cleanup)."""

import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.controller.globalaccelerator import GlobalAcceleratorConfig
from agac.controller.route53 import Route53Config
from agac.kube.client import InMemoryKubeClient
from agac.manager import ControllerConfig, Manager

MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
LB_TYPE = "service.beta.kubernetes.io/aws-load-balancer-type"


def wait_until(pred, timeout=15.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(0.02)
    return pred()


def start_cluster(backend, cluster_name):
    """One 'cluster': own kube API, own controllers, shared AWS account."""
    client = InMemoryKubeClient()
    stop = threading.Event()
    manager = Manager()
    config = ControllerConfig(
        global_accelerator=GlobalAcceleratorConfig(cluster_name=cluster_name),
        route53=Route53Config(cluster_name=cluster_name),
    )
    manager.run(client, config, FakeCloudFactory(backend), stop, resync_period=0.3, block=False)
    assert manager.wait_until_ready()
    return client, stop


def mk_service(backend, name, lb_name=None):
    # LB names are unique per AWS account+region (enforced by the fake),
    # so each cluster's cloud controller would have provisioned a distinct
    # LB even for same-named Services
    lb = backend.elbv2.create_load_balancer(lb_name or name, region="us-east-1")
    return corev1.Service(
        metadata=ObjectMeta(
            name=name, namespace="default",
            annotations={LB_TYPE: "nlb", MANAGED: "true"},
        ),
        spec=corev1.ServiceSpec(
            type="LoadBalancer", ports=[corev1.ServicePort(port=80, protocol="TCP")]
        ),
        status=corev1.ServiceStatus(
            load_balancer=corev1.LoadBalancerStatus(
                ingress=[corev1.LoadBalancerIngress(hostname=lb.dns_name)]
            )
        ),
    )


def owners(backend):
    """{accelerator arn -> (owner tag, cluster tag)}"""
    accs, _ = backend.ga.list_accelerators()
    out = {}
    for acc in accs:
        tags = {x.key: x.value for x in backend.ga.list_tags_for_resource(acc.accelerator_arn)}
        out[acc.accelerator_arn] = (
            tags.get("aws-global-accelerator-owner"),
            tags.get("aws-global-accelerator-cluster"),
        )
    return out


def test_two_clusters_do_not_interfere():
    backend = FakeAWSBackend()
    client_a, stop_a = start_cluster(backend, "cluster-a")
    client_b, stop_b = start_cluster(backend, "cluster-b")
    try:
        # same-named service in both clusters (realistic collision case)
        client_a.create(mk_service(backend, "web", lb_name="web-cluster-a"))
        client_b.create(mk_service(backend, "web", lb_name="web-cluster-b"))
        assert wait_until(lambda: len(owners(backend)) == 2)
        clusters = {c for _, c in owners(backend).values()}
        assert clusters == {"cluster-a", "cluster-b"}

        # deleting cluster-a's service must only remove cluster-a's accelerator
        client_a.delete("Service", "default", "web")
        assert wait_until(
            lambda: [c for _, c in owners(backend).values()] == ["cluster-b"]
        )

        # cluster-b's object survives and still reconciles
        svc = client_b.get("Service", "default", "web")
        svc.spec.ports[0].port = 443
        client_b.update(svc)

        def b_listener_updated():
            (arn,) = [a for a, (_, c) in owners(backend).items() if c == "cluster-b"]
            listeners, _ = backend.ga.list_listeners(arn)
            return [p.from_port for p in listeners[0].port_ranges] == [443]

        assert wait_until(b_listener_updated)
    finally:
        stop_a.set()
        stop_b.set()
