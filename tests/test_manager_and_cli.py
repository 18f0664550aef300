"""Manager registry, CLI surface, fixture, signals."""

import threading

from click.testing import CliRunner

from agac.cli import cli, resolve_kubeconfig
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.fixture import endpoint_group_binding
from agac.kube.client import InMemoryKubeClient
from agac.manager import ControllerConfig, Manager, new_controller_initializers


class TestManager:
    def test_initializer_registry_names(self):
        # reference manager.go:34-40 registers exactly these three
        assert set(new_controller_initializers()) == {
            "global-accelerator-controller",
            "route53-controller",
            "endpoint-group-binding-controller",
        }

    def test_manager_starts_and_stops_all_controllers(self):
        client = InMemoryKubeClient()
        stop = threading.Event()
        manager = Manager()
        manager.run(client, ControllerConfig(), FakeCloudFactory(), stop, block=False)
        assert manager.wait_until_ready()
        assert len(manager.controllers) == 3
        assert len(manager.threads) == 3
        stop.set()
        for thread in manager.threads:
            thread.join(timeout=5.0)
            assert not thread.is_alive()


class TestCLI:
    def test_version(self):
        result = CliRunner().invoke(cli, ["version"])
        assert result.exit_code == 0
        assert "aws-global-accelerator-controller" in result.output

    def test_help_lists_subcommands(self):
        result = CliRunner().invoke(cli, ["--help"])
        assert result.exit_code == 0
        for sub in ("controller", "webhook", "version", "apiserver"):
            assert sub in result.output

    def test_controller_flags_surface(self):
        result = CliRunner().invoke(cli, ["controller", "--help"])
        assert result.exit_code == 0
        for flag in ("--workers", "--cluster-name", "--kubeconfig", "--master"):
            assert flag in result.output

    def test_webhook_flags_surface(self):
        result = CliRunner().invoke(cli, ["webhook", "--help"])
        for flag in ("--tls-cert-file", "--tls-private-key-file", "--port", "--ssl"):
            assert flag in result.output

    def test_resolve_kubeconfig_precedence(self, monkeypatch, tmp_path):
        monkeypatch.setenv("KUBECONFIG", "/from/env")
        assert resolve_kubeconfig("/from/flag") == "/from/flag"
        assert resolve_kubeconfig("") == "/from/env"
        monkeypatch.delenv("KUBECONFIG")
        monkeypatch.setattr("os.path.expanduser", lambda p: str(tmp_path / "config"))
        assert resolve_kubeconfig("") == ""
        (tmp_path / "config").write_text("kind: Config")
        assert resolve_kubeconfig("") == str(tmp_path / "config")


class TestFixture:
    def test_default_shape(self):
        binding = endpoint_group_binding()
        assert binding.spec.endpoint_group_arn.startswith("arn:aws:globalaccelerator")
        assert binding.spec.weight == 128
        assert binding.spec.service_ref.name == "test-service"
        assert binding.spec.ingress_ref is None

    def test_ingress_variant(self):
        binding = endpoint_group_binding(ingress_name="my-ingress")
        assert binding.spec.ingress_ref.name == "my-ingress"
        assert binding.spec.service_ref is None


class TestDemo:
    def test_demo_runs_end_to_end(self):
        from click.testing import CliRunner

        result = CliRunner().invoke(cli, ["demo", "--objects", "2"])
        assert result.exit_code == 0, result.output
        assert "demo OK" in result.output
        assert "GlobalAcceleratorCreated" in result.output


class TestLogFormat:
    def test_json_log_format_flag(self):
        import json as jsonlib
        import subprocess
        import sys

        out = subprocess.run(
            [sys.executable, "-m", "agac.cli", "--log-format", "json", "-v", "version"],
            capture_output=True, text=True, timeout=60,
        )
        assert out.returncode == 0
        for line in out.stderr.strip().splitlines():
            if line.startswith("{"):
                entry = jsonlib.loads(line)
                assert {"ts", "level", "logger", "message"} <= set(entry)
