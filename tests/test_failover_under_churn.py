"""Leader failover under live churn (BASELINE.json config 5: "leader-
election failover with 2 controller replicas"): two full controller
replicas share one HTTP apiserver and one AWS fake; objects churn
continuously while the active leader is killed mid-flight.  The standby
must take over and the fleet must converge exactly — one accelerator per
managed service with the final ports, no duplicates, no orphans."""

import threading
import time

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.kube.httpapi import APIServer
from agac.kube.leaderelection import LeaderElectionConfig, LeaderElector
from agac.kube.rest import RestKubeClient
from agac.kube.store import APIStore, ConflictError
from agac.manager import ControllerConfig, Manager

REGION = "us-east-1"
MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
LB_TYPE = "service.beta.kubernetes.io/aws-load-balancer-type"


class Replica:
    """One controller process: leader election wrapping a manager, like
    `agac controller --leader-elect` (cmd flow in agac/cli.py)."""

    def __init__(self, name, url, backend):
        self.name = name
        self.client = RestKubeClient(url)
        self.backend = backend
        self.outer_stop = threading.Event()
        self.leading = threading.Event()

        def on_started_leading(stop_leading):
            manager = Manager()
            manager.run(self.client, ControllerConfig(),
                        FakeCloudFactory(backend), stop_leading,
                        resync_period=300.0, block=False)
            self.leading.set()
            stop_leading.wait()

        self.elector = LeaderElector(
            self.client,
            name="failover-churn-lease",
            namespace="default",
            identity=name,
            on_started_leading=on_started_leading,
            config=LeaderElectionConfig(
                lease_duration=1.0, renew_deadline=0.5, retry_period=0.1,
                release_on_cancel=True,
            ),
        )
        self.thread = threading.Thread(
            target=self.elector.run, args=(self.outer_stop,), daemon=True
        )

    def start(self):
        self.thread.start()

    def kill(self):
        self.outer_stop.set()


def test_failover_under_churn():
    server = APIServer(APIStore(), watch_idle_seconds=0.1)
    server.start()
    backend = FakeAWSBackend()
    writer_client = RestKubeClient(server.url)

    n_services = 8
    lbs = {}
    for i in range(n_services):
        lbs[f"fo-{i}"] = backend.elbv2.create_load_balancer(f"fo-{i}", region=REGION)

    def push(name, port):
        for _ in range(20):
            try:
                try:
                    svc = writer_client.get("Service", "default", name)
                    svc.spec.ports = [corev1.ServicePort(port=port, protocol="TCP")]
                    writer_client.update(svc)
                except Exception as e:
                    from agac.kube.store import NotFoundError

                    if not isinstance(e, NotFoundError):
                        raise
                    writer_client.create(corev1.Service(
                        metadata=ObjectMeta(
                            name=name, namespace="default",
                            annotations={LB_TYPE: "nlb", MANAGED: "true"},
                        ),
                        spec=corev1.ServiceSpec(
                            type="LoadBalancer",
                            ports=[corev1.ServicePort(port=port, protocol="TCP")],
                        ),
                        status=corev1.ServiceStatus(
                            load_balancer=corev1.LoadBalancerStatus(
                                ingress=[corev1.LoadBalancerIngress(
                                    hostname=lbs[name].dns_name)]
                            )
                        ),
                    ))
                return
            except ConflictError:
                continue

    a = Replica("replica-a", server.url, backend)
    b = Replica("replica-b", server.url, backend)
    a.start()
    time.sleep(0.2)  # let A win deterministically
    b.start()
    try:
        assert a.leading.wait(10.0), "replica A never led"
        assert not b.leading.is_set()

        # churn phase 1 under A
        for i in range(n_services):
            push(f"fo-{i}", 8000 + i)
        time.sleep(0.5)

        # kill the leader mid-churn; keep churning THROUGH the failover
        a.kill()
        final_ports = {}
        for round_ in range(3):
            for i in range(n_services):
                port = 9000 + 10 * round_ + i
                push(f"fo-{i}", port)
                final_ports[f"fo-{i}"] = port
            time.sleep(0.3)

        assert b.leading.wait(15.0), "standby never took over"

        # convergence audit: exactly one accelerator per service carrying
        # the FINAL port; no duplicates from the dual-brain window
        def converged():
            accs, _ = backend.ga.list_accelerators()
            owners = {}
            for acc in accs:
                tags = {t.key: t.value for t in
                        backend.ga.list_tags_for_resource(acc.accelerator_arn)}
                owners.setdefault(tags.get("aws-global-accelerator-owner"), []).append(acc)
            if set(owners) != {f"service/default/fo-{i}" for i in range(n_services)}:
                return False
            for name, port in final_ports.items():
                entries = owners[f"service/default/{name}"]
                if len(entries) != 1:
                    raise AssertionError(
                        f"duplicate accelerators for {name}: {len(entries)}"
                    )
                listeners, _ = backend.ga.list_listeners(entries[0].accelerator_arn)
                if len(listeners) != 1:
                    return False
                if [p.from_port for p in listeners[0].port_ranges] != [port]:
                    return False
            return True

        deadline = time.monotonic() + 30
        while not converged():
            assert time.monotonic() < deadline, "failover convergence timed out"
            time.sleep(0.05)

        # the lease really changed hands
        lease = writer_client.get("Lease", "default", "failover-churn-lease")
        assert lease.spec.holder_identity == "replica-b"
    finally:
        a.kill()
        b.kill()
        time.sleep(0.3)
        server.shutdown()
