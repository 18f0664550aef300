"""Opt-in cloud drift repair (--cloud-resync-minutes).

The reference never re-enqueues unchanged objects (its DeepEqual update
guard drops every resync pair — ga/controller.go:99-101), so cloud-side
drift on an untouched k8s object is never repaired.  These tests pin BOTH
behaviors: the parity default (drift persists) and the opt-in periodic
re-enqueue (drift converges within one period).  See docs/PARITY.md
§resync.
"""

import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws import types as awstypes
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.controller.endpointgroupbinding import EndpointGroupBindingConfig
from agac.controller.globalaccelerator import GlobalAcceleratorConfig
from agac.controller.route53 import Route53Config
from agac.kube.client import InMemoryKubeClient
from agac.manager import ControllerConfig, Manager

REGION = "us-east-1"


def managed_service(name, lb):
    return corev1.Service(
        metadata=ObjectMeta(
            name=name,
            namespace="default",
            annotations={
                "service.beta.kubernetes.io/aws-load-balancer-type": "nlb",
                "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed": "true",
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
    )


def start_stack(cloud_resync_period: float):
    backend = FakeAWSBackend()
    client = InMemoryKubeClient()
    stop = threading.Event()
    manager = Manager()
    config = ControllerConfig(
        global_accelerator=GlobalAcceleratorConfig(
            cloud_resync_period=cloud_resync_period
        ),
        route53=Route53Config(cloud_resync_period=cloud_resync_period),
        endpoint_group_binding=EndpointGroupBindingConfig(
            cloud_resync_period=cloud_resync_period
        ),
    )
    manager.run(client, config, FakeCloudFactory(backend), stop,
                resync_period=300.0, block=False)
    assert manager.wait_until_ready()
    return backend, client, stop


def wait_for(predicate, what, timeout=10.0):
    deadline = time.monotonic() + timeout
    while not predicate():
        if time.monotonic() > deadline:
            raise TimeoutError(f"{what} did not happen within {timeout}s")
        time.sleep(0.02)


def listener_ports(backend):
    accs, _ = backend.ga.list_accelerators()
    if not accs:
        return None
    listeners, _ = backend.ga.list_listeners(accs[0].accelerator_arn)
    if not listeners:
        return None
    return [p.from_port for p in listeners[0].port_ranges]


def test_drift_repaired_with_cloud_resync_enabled():
    backend, client, stop = start_stack(cloud_resync_period=0.2)
    try:
        lb = backend.elbv2.create_load_balancer("drift", region=REGION)
        client.create(managed_service("drift", lb))
        wait_for(lambda: listener_ports(backend) == [80], "initial converge")

        # mutate the cloud behind the controller's back; the k8s object is
        # NOT touched
        accs, _ = backend.ga.list_accelerators()
        listeners, _ = backend.ga.list_listeners(accs[0].accelerator_arn)
        backend.ga.update_listener(
            listeners[0].listener_arn, port_ranges=[awstypes.PortRange(9999, 9999)]
        )
        assert listener_ports(backend) == [9999]
        # the periodic re-enqueue repairs it within a couple of periods
        wait_for(lambda: listener_ports(backend) == [80], "drift repair",
                 timeout=5.0)
    finally:
        stop.set()


def test_drift_persists_with_default_config():
    """Parity default (0 = disabled): the reference never repairs drift on
    an unchanged object, and neither do we unless opted in."""
    backend, client, stop = start_stack(cloud_resync_period=0.0)
    try:
        lb = backend.elbv2.create_load_balancer("stuck", region=REGION)
        client.create(managed_service("stuck", lb))
        wait_for(lambda: listener_ports(backend) == [80], "initial converge")

        accs, _ = backend.ga.list_accelerators()
        listeners, _ = backend.ga.list_listeners(accs[0].accelerator_arn)
        backend.ga.update_listener(
            listeners[0].listener_arn, port_ranges=[awstypes.PortRange(9999, 9999)]
        )
        time.sleep(0.8)  # several would-be resync periods
        assert listener_ports(backend) == [9999], (
            "drift unexpectedly repaired — parity default changed"
        )
        # but an object EDIT still repairs it (the reference's only path)
        svc = client.get("Service", "default", "stuck")
        svc.spec.ports[0].port = 81
        client.update(svc)
        wait_for(lambda: listener_ports(backend) == [81], "edit-triggered repair")
    finally:
        stop.set()


def test_route53_drift_repaired():
    backend, client, stop = start_stack(cloud_resync_period=0.2)
    try:
        zone = backend.route53.create_hosted_zone("drift.example.com")
        lb = backend.elbv2.create_load_balancer("r53drift", region=REGION)
        svc = managed_service("r53drift", lb)
        svc.metadata.annotations[
            "aws-global-accelerator-controller.h3poteto.dev/route53-hostname"
        ] = "app.drift.example.com"
        client.create(svc)

        def alias_dns():
            recs, _ = backend.route53.list_resource_record_sets(zone.id)
            for r in recs:
                if r.type == "A" and r.alias_target is not None:
                    return r.alias_target.dns_name
            return None

        # 90s: covers the reference's 60s GA-missing requeue if the
        # route53 reconcile races ahead of the accelerator creation
        wait_for(lambda: alias_dns() is not None, "initial route53 converge",
                 timeout=90.0)
        good = alias_dns()

        # drift the alias record out from under the controller
        import agac.cloudprovider.aws.types as t

        backend.route53.change_resource_record_sets(zone.id, [t.Change(
            action="UPSERT",
            record_set=t.ResourceRecordSet(
                name="app.drift.example.com.", type="A",
                alias_target=t.AliasTarget(
                    dns_name="wrong.awsglobalaccelerator.com",
                    evaluate_target_health=True,
                    hosted_zone_id="Z2BJ6XQ5FK7U4H",
                ),
            ),
        )])
        wait_for(lambda: alias_dns() == good, "route53 drift repair", timeout=5.0)
    finally:
        stop.set()


def test_cli_flag_maps_to_all_three_configs(monkeypatch):
    """--cloud-resync-minutes reaches every controller config (in seconds)."""
    from click.testing import CliRunner

    from agac import cli as climod

    captured = {}

    class FakeManager:
        def run(self, kube_client, config, cloud_factory, stop, **kw):
            captured["config"] = config
            stop.set()

    monkeypatch.setattr("agac.manager.Manager", FakeManager)
    result = CliRunner().invoke(climod.cli, [
        "controller", "--api", "memory", "--no-leader-elect",
        "--cloud-resync-minutes", "2.5",
    ])
    assert result.exit_code == 0, result.output
    config = captured["config"]
    assert config.global_accelerator.cloud_resync_period == 150.0
    assert config.route53.cloud_resync_period == 150.0
    assert config.endpoint_group_binding.cloud_resync_period == 150.0
