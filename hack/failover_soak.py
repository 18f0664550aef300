#!/usr/bin/env python3
"""Leadership-churn soak: two controller replicas (Lease leader election)
over one HTTP apiserver and one AWS fake, with continuous object churn
while the ACTIVE LEADER IS KILLED every --kill-every seconds and replaced
by a fresh standby.  After every failover, convergence is audited exactly
(one accelerator per managed service carrying the current ports).

Emits one JSON line per failover cycle and a final summary:
  {"failovers": N, "churn_updates": M, "convergence_failures": 0,
   "max_failover_to_converged_s": ..., "ok": true}
"""

from __future__ import annotations

import argparse
import json
import sys
import threading
import time

sys.path.insert(0, ".")

from agac.apis import core as corev1  # noqa: E402
from agac.apis.meta import ObjectMeta  # noqa: E402
from agac.cloudprovider.aws.client import FakeCloudFactory  # noqa: E402
from agac.cloudprovider.fake import FakeAWSBackend  # noqa: E402
from agac.kube.httpapi import APIServer  # noqa: E402
from agac.kube.leaderelection import (  # noqa: E402
    LeaderElectionConfig,
    LeaderElector,
)
from agac.kube.rest import RestKubeClient  # noqa: E402
from agac.kube.store import APIStore, ConflictError, NotFoundError  # noqa: E402
from agac.manager import ControllerConfig, Manager  # noqa: E402

MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
LB_TYPE = "service.beta.kubernetes.io/aws-load-balancer-type"
REGION = "us-east-1"


class Replica:
    def __init__(self, ident: str, url: str, backend):
        self.ident = ident
        self.client = RestKubeClient(url)
        self.stop = threading.Event()
        self.leading = threading.Event()

        def on_started_leading(stop_leading):
            manager = Manager()
            manager.run(self.client, ControllerConfig(),
                        FakeCloudFactory(backend), stop_leading,
                        resync_period=300.0, block=False)
            self.leading.set()
            stop_leading.wait()

        self.elector = LeaderElector(
            self.client, name="failover-soak", namespace="default",
            identity=ident, on_started_leading=on_started_leading,
            config=LeaderElectionConfig(
                lease_duration=2.0, renew_deadline=1.0, retry_period=0.2,
                release_on_cancel=True,
            ),
        )
        self.thread = threading.Thread(
            target=self.elector.run, args=(self.stop,), daemon=True
        )
        self.thread.start()


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--minutes", type=float, default=5.0)
    parser.add_argument("--kill-every", type=float, default=20.0)
    parser.add_argument("--objects", type=int, default=24)
    args = parser.parse_args()

    server = APIServer(APIStore(), watch_idle_seconds=0.5)
    server.start()
    backend = FakeAWSBackend()
    writer = RestKubeClient(server.url)
    lbs = {}
    for i in range(args.objects):
        lbs[f"fs-{i}"] = backend.elbv2.create_load_balancer(f"fs-{i}", region=REGION)

    ports = {}
    churn_count = {"n": 0}
    churn_stop = threading.Event()

    def push(name, port):
        for _ in range(30):
            try:
                try:
                    svc = writer.get("Service", "default", name)
                    svc.spec.ports = [corev1.ServicePort(port=port, protocol="TCP")]
                    writer.update(svc)
                except NotFoundError:
                    writer.create(corev1.Service(
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
                ports[name] = port
                churn_count["n"] += 1
                return
            except (ConflictError, ConnectionError, OSError):
                time.sleep(0.01)

    churn_pause = threading.Event()

    def churner():
        i = 0
        while not churn_stop.is_set():
            if churn_pause.is_set():
                time.sleep(0.02)
                continue
            name = f"fs-{i % args.objects}"
            push(name, 8000 + (i % 500))
            i += 1
            time.sleep(0.01)

    def converged() -> bool:
        accs, _ = backend.ga.list_accelerators()
        owners = {}
        for acc in accs:
            tags = {t.key: t.value for t in
                    backend.ga.list_tags_for_resource(acc.accelerator_arn)}
            owners.setdefault(tags.get("aws-global-accelerator-owner"), []).append(acc)
        want = {f"service/default/{n}" for n in ports}
        if set(owners) != want:
            return False
        for name, port in ports.items():
            entries = owners[f"service/default/{name}"]
            if len(entries) != 1:
                raise RuntimeError(f"DUPLICATE accelerators for {name}")
            listeners, _ = backend.ga.list_listeners(entries[0].accelerator_arn)
            if len(listeners) != 1:
                return False
            if [p.from_port for p in listeners[0].port_ranges] != [port]:
                return False
        return True

    replica_seq = 0

    def new_replica():
        nonlocal replica_seq
        replica_seq += 1
        return Replica(f"replica-{replica_seq}", server.url, backend)

    active = new_replica()
    standby = new_replica()
    deadline0 = time.monotonic() + 20.0
    while not (active.leading.is_set() or standby.leading.is_set()):
        assert time.monotonic() < deadline0, "no initial leader"
        time.sleep(0.05)
    if standby.leading.is_set():  # either replica may win the first lease
        active, standby = standby, active
    threading.Thread(target=churner, daemon=True).start()

    deadline = time.monotonic() + args.minutes * 60
    failovers = 0
    convergence_failures = 0
    max_converge_s = 0.0
    try:
        while time.monotonic() < deadline:
            time.sleep(args.kill_every)
            # kill whichever replica currently leads
            leader = active if active.leading.is_set() else standby
            other = standby if leader is active else active
            leader.stop.set()
            t0 = time.monotonic()
            took_over = other.leading.wait(30.0)
            if not took_over:
                convergence_failures += 1
                print(json.dumps({"failover": failovers + 1,
                                  "error": "standby never led"}))
                break
            # quiesce churn briefly to audit an exact state
            churn_pause.set()
            time.sleep(0.3)
            audit_deadline = time.monotonic() + 60
            ok = False
            while time.monotonic() < audit_deadline:
                try:
                    if converged():
                        ok = True
                        break
                except RuntimeError as e:
                    print(json.dumps({"fatal": str(e)}))
                    raise
                time.sleep(0.05)
            elapsed = time.monotonic() - t0
            max_converge_s = max(max_converge_s, elapsed)
            failovers += 1
            if not ok:
                convergence_failures += 1
            print(json.dumps({
                "failover": failovers, "new_leader": other.ident,
                "kill_to_converged_s": round(elapsed, 2), "converged": ok,
            }))
            sys.stdout.flush()
            # resume churn with a fresh standby (single long-lived churner)
            churn_pause.clear()
            active, standby = other, new_replica()
    finally:
        churn_stop.set()
        active.stop.set()
        standby.stop.set()
        time.sleep(0.3)
        summary = {
            "failovers": failovers,
            "churn_updates": churn_count["n"],
            "convergence_failures": convergence_failures,
            "max_kill_to_converged_s": round(max_converge_s, 2),
            "minutes": args.minutes,
            "objects": args.objects,
            "ok": convergence_failures == 0 and failovers > 0,
        }
        print(json.dumps(summary))
        sys.stdout.flush()
        server.shutdown()


if __name__ == "__main__":
    main()
