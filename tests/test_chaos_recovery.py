"""Chaos: controller restarts + concurrent writers under churn must still
converge (level-triggered recovery, SURVEY.md §5 failure detection).
Bounded to a few seconds of wall clock."""

import random
import threading
import time

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.kube.client import InMemoryKubeClient
from agac.kube.store import ConflictError
from agac.manager import ControllerConfig, Manager

MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
LB_TYPE = "service.beta.kubernetes.io/aws-load-balancer-type"
N_OBJECTS = 6


def wait_until(pred, timeout=30.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(0.02)
    return pred()


def start_manager(client, backend):
    stop = threading.Event()
    manager = Manager()
    manager.run(
        client, ControllerConfig(), FakeCloudFactory(backend), stop,
        resync_period=0.2, block=False,
    )
    assert manager.wait_until_ready()
    return stop


def test_convergence_through_restarts_and_conflicts():
    client = InMemoryKubeClient()
    backend = FakeAWSBackend()
    rng = random.Random(7)

    lbs = {}
    for i in range(N_OBJECTS):
        lb = backend.elbv2.create_load_balancer(f"chaos-{i}", region="us-east-1")
        lbs[f"chaos-{i}"] = lb
        client.create(
            corev1.Service(
                metadata=ObjectMeta(
                    name=f"chaos-{i}", namespace="default",
                    annotations={LB_TYPE: "nlb", MANAGED: "true"},
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
        )

    stop = start_manager(client, backend)
    chaos_done = threading.Event()
    final_ports = {}
    conflicts = [0]

    def churn_writer():
        """Random port flips with optimistic-concurrency retries."""
        for round_idx in range(12):
            name = f"chaos-{rng.randint(0, N_OBJECTS - 1)}"
            port = rng.choice([80, 443, 8080])
            for _ in range(5):
                try:
                    svc = client.get("Service", "default", name)
                    svc.spec.ports[0].port = port
                    client.update(svc)
                    final_ports[name] = port
                    break
                except ConflictError:
                    conflicts[0] += 1
            time.sleep(rng.uniform(0.0, 0.05))

    writers = [threading.Thread(target=churn_writer) for _ in range(3)]
    for w in writers:
        w.start()

    # restart the whole controller plane twice mid-churn
    for _ in range(2):
        time.sleep(0.2)
        stop.set()
        time.sleep(0.1)
        stop = start_manager(client, backend)

    for w in writers:
        w.join(timeout=30)
        assert not w.is_alive()
    chaos_done.set()

    try:
        # after the dust settles, every accelerator listener must match the
        # service's final spec (read the store for ground truth)
        def converged():
            accs, _ = backend.ga.list_accelerators()
            if len(accs) != N_OBJECTS:
                return False
            by_owner = {}
            for acc in accs:
                tags = {
                    x.key: x.value
                    for x in backend.ga.list_tags_for_resource(acc.accelerator_arn)
                }
                by_owner[tags["aws-global-accelerator-owner"]] = acc.accelerator_arn
            for i in range(N_OBJECTS):
                name = f"chaos-{i}"
                svc = client.get("Service", "default", name)
                want = svc.spec.ports[0].port
                arn = by_owner.get(f"service/default/{name}")
                if arn is None:
                    return False
                listeners, _ = backend.ga.list_listeners(arn)
                if len(listeners) != 1:
                    return False
                ports = [p.from_port for p in listeners[0].port_ranges]
                if ports != [want]:
                    return False
            return True

        assert wait_until(converged), "cloud state diverged from specs after chaos"
        # exactly one accelerator per service — restarts created no duplicates
        accs, _ = backend.ga.list_accelerators()
        assert len(accs) == N_OBJECTS
    finally:
        stop.set()
