"""kubectl-apply-lite: the shipped sample manifests apply cleanly and
drive real reconciles (validates config/samples against the live stack)."""

import os
import threading
import time

import pytest

from agac.apis import core as corev1
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.kube.apply import apply_yaml
from agac.kube.client import InMemoryKubeClient
from agac.manager import ControllerConfig, Manager

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def read(sample):
    with open(os.path.join(REPO, "config", "samples", sample)) as f:
        return f.read()


def wait_until(pred, timeout=15.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(0.02)
    return pred()


def test_apply_create_update_unchanged_cycle():
    client = InMemoryKubeClient()
    text = read("nlb-public-service.yaml")
    assert apply_yaml(client, text) == [("created", "service/default/sample-nlb")]
    assert apply_yaml(client, text) == [("unchanged", "service/default/sample-nlb")]
    changed = text.replace("port: 80", "port: 8080")
    assert apply_yaml(client, changed) == [("configured", "service/default/sample-nlb")]
    assert client.get("Service", "default", "sample-nlb").spec.ports[0].port == 8080


def test_unknown_kinds_are_skipped():
    client = InMemoryKubeClient()
    results = apply_yaml(client, read("deployment.yaml"))
    assert results == [("skipped", "apps/v1/Deployment")]


def test_all_samples_apply():
    client = InMemoryKubeClient()
    sample_dir = os.path.join(REPO, "config", "samples")
    outcomes = []
    for name in sorted(os.listdir(sample_dir)):
        outcomes += apply_yaml(client, read(name))
    applied = [i for a, i in outcomes if a == "created"]
    # every Service/Ingress/EndpointGroupBinding sample lands
    assert any(i.startswith("service/") for i in applied)
    assert any(i.startswith("ingress/") for i in applied)
    assert any(i.startswith("endpointgroupbinding/") for i in applied)


def test_applied_sample_reconciles_end_to_end():
    """Apply the shipped NLB sample, then play cloud-controller-manager:
    provision its LB and set status — the GA controller must converge."""
    client = InMemoryKubeClient()
    backend = FakeAWSBackend()
    stop = threading.Event()
    manager = Manager()
    manager.run(client, ControllerConfig(), FakeCloudFactory(backend), stop,
                resync_period=0.5, block=False)
    try:
        assert manager.wait_until_ready()
        apply_yaml(client, read("nlb-public-service.yaml"))
        # no LB status yet → controller skips
        time.sleep(0.2)
        assert backend.ga.list_accelerators()[0] == []

        lb = backend.elbv2.create_load_balancer("sample-nlb", region="us-east-1")
        svc = client.get("Service", "default", "sample-nlb")
        svc.status = corev1.ServiceStatus(
            load_balancer=corev1.LoadBalancerStatus(
                ingress=[corev1.LoadBalancerIngress(hostname=lb.dns_name)]
            )
        )
        client.update_status(svc)
        assert wait_until(lambda: len(backend.ga.list_accelerators()[0]) == 1)
        acc = backend.ga.list_accelerators()[0][0]
        assert acc.name == "service-default-sample-nlb"
    finally:
        stop.set()


def test_apply_cli_command(tmp_path):
    from click.testing import CliRunner

    from agac.cli import cli
    from agac.kube.httpapi import APIServer
    from agac.kube.store import APIStore

    api = APIServer(APIStore())
    api.start()
    try:
        manifest = tmp_path / "svc.yaml"
        manifest.write_text(read("nlb-public-service.yaml"))
        result = CliRunner().invoke(
            cli, ["apply", "-f", str(manifest), "--master", api.url]
        )
        assert result.exit_code == 0, result.output
        assert "service/default/sample-nlb created" in result.output
        # idempotent second apply
        result = CliRunner().invoke(
            cli, ["apply", "-f", str(manifest), "--master", api.url]
        )
        assert "unchanged" in result.output
    finally:
        api.shutdown()


def test_label_change_reconfigures():
    client = InMemoryKubeClient()
    text = read("nlb-public-service.yaml")
    apply_yaml(client, text)
    labeled = text.replace(
        "metadata:\n  name: sample-nlb",
        "metadata:\n  name: sample-nlb\n  labels:\n    team: infra",
    )
    assert apply_yaml(client, labeled) == [("configured", "service/default/sample-nlb")]
    assert client.get("Service", "default", "sample-nlb").metadata.labels == {"team": "infra"}
