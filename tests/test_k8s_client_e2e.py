"""Full-manager e2e over the PRODUCTION wire client (`--api k8s`):
K8sKubeClient → HTTP → the k8s-style apiserver routes, with a tiny LIST
page size so every informer sync exercises limit/continue pagination and
idle bookmarks flow on the watches.  This is the closest available
analogue to running against a real cluster (docs/REAL_CLUSTER.md) — the
same client class, wire paths, pagination, bookmark, retry and
fieldManager behavior the production deployment uses.
"""

import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis import endpointgroupbinding as egbapi
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.kube.httpapi import APIServer
from agac.kube.k8s import K8sKubeClient
from agac.kube.kubeconfig import RestConfig
from agac.kube.store import APIStore, NotFoundError
from agac.manager import ControllerConfig, Manager

REGION = "us-east-1"


@pytest.fixture
def env():
    server = APIServer(APIStore(), watch_idle_seconds=0.1)
    server.start()
    # page_size=2 forces multi-page informer syncs even for small fixtures
    client = K8sKubeClient(RestConfig(host=server.url), page_size=2)
    backend = FakeAWSBackend()
    stop = threading.Event()
    manager = Manager()
    manager.run(client, ControllerConfig(), FakeCloudFactory(backend), stop,
                resync_period=300.0, block=False)
    assert manager.wait_until_ready()
    yield type("Env", (), {
        "server": server, "client": client, "backend": backend, "stop": stop,
    })
    stop.set()
    server.shutdown()


def wait_for(predicate, what, timeout=20.0):
    deadline = time.monotonic() + timeout
    while not predicate():
        if time.monotonic() > deadline:
            raise TimeoutError(f"{what} within {timeout}s")
        time.sleep(0.02)


def managed_service(name, lb, hostname=None):
    annotations = {
        "service.beta.kubernetes.io/aws-load-balancer-type": "nlb",
        "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed": "true",
    }
    if hostname:
        annotations[
            "aws-global-accelerator-controller.h3poteto.dev/route53-hostname"
        ] = hostname
    return corev1.Service(
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
    )


def test_full_scenario_over_production_client(env):
    """Service -> GA triple, route53-hostname -> TXT+A pair, EGB attach and
    finalizer drain — the smoke() scenario, through K8sKubeClient."""
    zone = env.backend.route53.create_hosted_zone("k8s.example.com")
    lb = env.backend.elbv2.create_load_balancer("k8se2e", region=REGION)
    env.client.create(managed_service("k8se2e", lb, "app.k8s.example.com"))

    def full_triple():
        accs, _ = env.backend.ga.list_accelerators()
        if len(accs) != 1:
            return False
        listeners, _ = env.backend.ga.list_listeners(accs[0].accelerator_arn)
        if len(listeners) != 1:
            return False
        groups, _ = env.backend.ga.list_endpoint_groups(listeners[0].listener_arn)
        return len(groups) == 1

    wait_for(full_triple, "accelerator triple")
    acc = env.backend.ga.list_accelerators()[0][0]
    listeners, _ = env.backend.ga.list_listeners(acc.accelerator_arn)
    groups, _ = env.backend.ga.list_endpoint_groups(listeners[0].listener_arn)

    def record_types():
        recs, _ = env.backend.route53.list_resource_record_sets(zone.id)
        return {(r.name, r.type) for r in recs}

    # 90s: if the route53 reconcile raced ahead of the accelerator it
    # requeues at the reference's 60s GA-missing interval
    wait_for(
        lambda: {("app.k8s.example.com.", "A"),
                 ("app.k8s.example.com.", "TXT")} <= record_types(),
        "route53 pair",
        timeout=90.0,
    )

    # EGB attach + drain through the CRD path
    lb2 = env.backend.elbv2.create_load_balancer("k8se2e2", region=REGION)
    env.client.create(corev1.Service(
        metadata=ObjectMeta(name="k8se2e2", namespace="default"),
        spec=corev1.ServiceSpec(type="LoadBalancer"),
        status=corev1.ServiceStatus(
            load_balancer=corev1.LoadBalancerStatus(
                ingress=[corev1.LoadBalancerIngress(hostname=lb2.dns_name)]
            )
        ),
    ))
    env.client.create(egbapi.EndpointGroupBinding(
        metadata=ObjectMeta(name="k8sbind", namespace="default"),
        spec=egbapi.EndpointGroupBindingSpec(
            endpoint_group_arn=groups[0].endpoint_group_arn,
            weight=32,
            service_ref=egbapi.ServiceReference(name="k8se2e2"),
        ),
    ))

    def attached():
        group = env.backend.ga.describe_endpoint_group(groups[0].endpoint_group_arn)
        return any(
            d.endpoint_id == lb2.load_balancer_arn and d.weight == 32
            for d in group.endpoint_descriptions
        )

    wait_for(attached, "EGB attach")

    # status round-trips through the wire AFTER the cloud attach (the
    # controller writes endpoint_ids + observedGeneration as a second
    # step) — wait for it rather than racing it
    def status_round_tripped():
        b = env.client.get("EndpointGroupBinding", "default", "k8sbind")
        return (
            b.status.endpoint_ids == [lb2.load_balancer_arn]
            and b.status.observed_generation == b.metadata.generation
        )

    wait_for(status_round_tripped, "EGB status round-trip")

    env.client.delete("EndpointGroupBinding", "default", "k8sbind")

    def drained():
        group = env.backend.ga.describe_endpoint_group(groups[0].endpoint_group_arn)
        if any(d.endpoint_id == lb2.load_balancer_arn
               for d in group.endpoint_descriptions):
            return False
        try:
            env.client.get("EndpointGroupBinding", "default", "k8sbind")
            return False
        except NotFoundError:
            return True

    wait_for(drained, "EGB finalizer drain")


def test_annotation_removal_cleans_up(env):
    lb = env.backend.elbv2.create_load_balancer("cleanup", region=REGION)
    env.client.create(managed_service("cleanup", lb))
    wait_for(lambda: len(env.backend.ga.list_accelerators()[0]) == 1, "create")

    svc = env.client.get("Service", "default", "cleanup")
    del svc.metadata.annotations[
        "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
    ]
    env.client.update(svc)
    wait_for(lambda: len(env.backend.ga.list_accelerators()[0]) == 0, "cleanup")


def test_paged_sync_populates_every_controller(env):
    """With page_size=2, a pre-populated store (several pages) still syncs
    every informer and reconciles every managed object."""
    for i in range(7):
        lb = env.backend.elbv2.create_load_balancer(f"page-{i}", region=REGION)
        env.client.create(managed_service(f"page-{i}", lb))
    wait_for(lambda: len(env.backend.ga.list_accelerators()[0]) == 7,
             "all 7 reconciled through 4 pages", timeout=60.0)


def test_leader_election_over_production_client(env):
    from agac.kube.leaderelection import LeaderElectionConfig, LeaderElector

    won = threading.Event()
    stop = threading.Event()
    elector = LeaderElector(
        env.client,
        name="k8s-e2e-lease",
        namespace="default",
        on_started_leading=lambda stop_leading: won.set(),
        config=LeaderElectionConfig(
            lease_duration=1.0, renew_deadline=0.5, retry_period=0.1
        ),
    )
    t = threading.Thread(target=elector.run, args=(stop,), daemon=True)
    t.start()
    try:
        assert won.wait(10.0), "never acquired the lease over the wire client"
        lease = env.client.get("Lease", "default", "k8s-e2e-lease")
        assert lease.spec.holder_identity == elector.identity
    finally:
        stop.set()
        t.join(timeout=5.0)


def test_apiserver_bounce_with_persisted_state():
    """The apiserver process dies mid-flight and comes back on the same
    port from a dumped snapshot (agac apiserver --state-file semantics):
    the controllers' informers must ride out the outage (connection
    refused -> backoff -> relist), not duplicate existing cloud state, and
    reconcile objects created after the restart."""
    server = APIServer(APIStore(), watch_idle_seconds=0.1)
    server.start()
    port = server.port
    client = K8sKubeClient(RestConfig(host=f"http://127.0.0.1:{port}"))
    backend = FakeAWSBackend()
    stop = threading.Event()
    manager = Manager()
    manager.run(client, ControllerConfig(), FakeCloudFactory(backend), stop,
                resync_period=300.0, block=False)
    try:
        assert manager.wait_until_ready()
        lb1 = backend.elbv2.create_load_balancer("bounce1", region=REGION)
        client.create(managed_service("bounce1", lb1))
        wait_for(lambda: len(backend.ga.list_accelerators()[0]) == 1, "initial")

        snapshot = server.store.dump()
        server.shutdown()
        time.sleep(0.5)  # informers hit connection-refused and back off

        server2 = APIServer(APIStore.load(snapshot), port=port,
                            watch_idle_seconds=0.1)
        server2.start()
        try:
            # new work created after the restart reconciles
            lb2 = backend.elbv2.create_load_balancer("bounce2", region=REGION)
            deadline = time.monotonic() + 20
            while True:
                try:
                    client.create(managed_service("bounce2", lb2))
                    break
                except Exception:
                    if time.monotonic() > deadline:
                        raise
                    time.sleep(0.1)
            wait_for(lambda: len(backend.ga.list_accelerators()[0]) == 2,
                     "post-restart reconcile", timeout=30.0)
            # and the pre-restart object was NOT duplicated
            names = sorted(a.name for a in backend.ga.list_accelerators()[0])
            assert names == ["service-default-bounce1", "service-default-bounce2"]
        finally:
            server2.shutdown()
    finally:
        stop.set()
        try:
            server.shutdown()
        except Exception:
            pass
