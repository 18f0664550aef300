"""Store snapshot/restore (checkpoint) and the `agac get` inspection
command, including the apiserver process surviving a restart with state."""

import json
import os
import signal
import subprocess
import sys
import time
import urllib.request

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.fixture import endpoint_group_binding
from agac.kube.store import APIStore

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class TestSnapshotRoundtrip:
    def test_dump_load_preserves_objects_and_rv(self):
        store = APIStore()
        store.create(
            corev1.Service(
                metadata=ObjectMeta(name="web", namespace="default"),
                spec=corev1.ServiceSpec(
                    type="LoadBalancer",
                    ports=[corev1.ServicePort(port=80, protocol="TCP")],
                ),
            )
        )
        store.create(endpoint_group_binding(name="b"))
        snapshot = store.dump()
        assert json.loads(json.dumps(snapshot)) == snapshot  # JSON-safe

        restored = APIStore.load(snapshot)
        svc = restored.get("Service", "default", "web")
        assert svc.spec.ports[0].port == 80
        binding = restored.get("EndpointGroupBinding", "default", "b")
        assert binding.spec.weight == 128

        # rv monotonicity continues past the snapshot
        old_rv = int(svc.metadata.resource_version)
        svc.metadata.annotations["post"] = "restore"
        updated = restored.update(svc)
        assert int(updated.metadata.resource_version) > old_rv

    def test_load_then_update_no_conflict(self):
        store = APIStore()
        store.create(corev1.Service(metadata=ObjectMeta(name="s", namespace="d")))
        restored = APIStore.load(store.dump())
        obj = restored.get("Service", "d", "s")
        obj.metadata.annotations["x"] = "y"
        restored.update(obj)  # must not raise ConflictError


def wait_http(url, timeout=15.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        try:
            with urllib.request.urlopen(url, timeout=2) as resp:
                return resp.status
        except Exception:
            time.sleep(0.1)
    raise TimeoutError(url)


class TestApiserverStatePersistence:
    def test_restart_with_state_file(self, tmp_path):
        state = tmp_path / "state.json"
        port = "18077"

        def start():
            return subprocess.Popen(
                [sys.executable, "-m", "agac.cli", "apiserver",
                 "--port", port, "--state-file", str(state)],
                cwd=REPO_ROOT, stdout=subprocess.PIPE,
                stderr=subprocess.STDOUT, text=True,
            )

        proc = start()
        try:
            assert wait_http(f"http://127.0.0.1:{port}/healthz") == 200
            body = json.dumps(
                {"metadata": {"name": "persisted"}, "spec": {"type": "ClusterIP"}}
            ).encode()
            req = urllib.request.Request(
                f"http://127.0.0.1:{port}/api/v1/namespaces/default/services",
                data=body, headers={"Content-Type": "application/json"}, method="POST",
            )
            with urllib.request.urlopen(req, timeout=5) as resp:
                assert resp.status == 201
            proc.send_signal(signal.SIGTERM)
            proc.wait(timeout=10)
            assert state.exists()

            proc = start()
            assert wait_http(f"http://127.0.0.1:{port}/healthz") == 200
            with urllib.request.urlopen(
                f"http://127.0.0.1:{port}/api/v1/namespaces/default/services/persisted",
                timeout=5,
            ) as resp:
                restored = json.loads(resp.read())
            assert restored["metadata"]["name"] == "persisted"
        finally:
            proc.kill()
            proc.wait(timeout=10)


class TestGetCommand:
    def test_get_lists_objects(self, tmp_path):
        from click.testing import CliRunner

        from agac.cli import cli
        from agac.kube.httpapi import APIServer

        store = APIStore()
        store.create(endpoint_group_binding(name="visible"))
        api = APIServer(store)
        api.start()
        try:
            result = CliRunner().invoke(
                cli, ["get", "egb", "--master", api.url]
            )
            assert result.exit_code == 0, result.output
            assert "visible" in result.output
            assert "endpoints=0" in result.output
        finally:
            api.shutdown()

    def test_get_unknown_kind_errors(self):
        from click.testing import CliRunner

        from agac.cli import cli

        result = CliRunner().invoke(cli, ["get", "pods"])
        assert result.exit_code != 0
        assert "unknown kind" in result.output


class TestGetAliases:
    def test_all_kind_aliases_resolve(self, tmp_path):
        from click.testing import CliRunner

        from agac.cli import cli
        from agac.kube.httpapi import APIServer

        store = APIStore()
        store.create(corev1.Service(metadata=ObjectMeta(name="s", namespace="d")))
        api = APIServer(store)
        api.start()
        try:
            for alias in ("svc", "services", "service"):
                result = CliRunner().invoke(cli, ["get", alias, "--master", api.url])
                assert result.exit_code == 0, result.output
                assert "s" in result.output
            for alias in ("ing", "egb", "leases", "events"):
                result = CliRunner().invoke(cli, ["get", alias, "--master", api.url])
                assert result.exit_code == 0, result.output
        finally:
            api.shutdown()

    def test_namespace_filter(self, tmp_path):
        from click.testing import CliRunner

        from agac.cli import cli
        from agac.kube.httpapi import APIServer

        store = APIStore()
        store.create(corev1.Service(metadata=ObjectMeta(name="a", namespace="ns1")))
        store.create(corev1.Service(metadata=ObjectMeta(name="b", namespace="ns2")))
        api = APIServer(store)
        api.start()
        try:
            result = CliRunner().invoke(
                cli, ["get", "svc", "-n", "ns1", "--master", api.url]
            )
            assert "a" in result.output and "b" not in result.output
        finally:
            api.shutdown()
