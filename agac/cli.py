"""Command-line interface.

click-based equivalent of the reference's cobra CLI (``cmd/``):

- ``agac controller`` — start the controller manager under leader election
  (flags: --workers/-w, --cluster-name/-c, --kubeconfig, --master,
  reference ``cmd/controller/controller.go:24-98``);
- ``agac webhook``   — start the admission webhook server (flags:
  --tls-cert-file, --tls-private-key-file, --port, --ssl,
  reference ``cmd/webhook/webhook.go:17-41``);
- ``agac version``   — print version info (reference ``cmd/version.go``).

kubeconfig resolution order matches viper wiring: flag → $KUBECONFIG →
~/.kube/config.  ``POD_NAMESPACE`` selects the leader-election namespace.
The kube backend is selected by --api: "memory" runs against an embedded
in-process API store (with the in-memory AWS fake — demo/e2e mode), "http"
connects to an agac API server (see ``agac.kube.rest``).
"""

from __future__ import annotations

import logging
import os
import sys

import click

from . import __version__

logger = logging.getLogger(__name__)


class _JsonFormatter(logging.Formatter):
    """One JSON object per line (for log pipelines)."""

    def format(self, record):
        import json as jsonlib

        entry = {
            "ts": self.formatTime(record, "%Y-%m-%dT%H:%M:%S"),
            "level": record.levelname,
            "logger": record.name,
            "message": record.getMessage(),
        }
        if record.exc_info:
            entry["exc"] = self.formatException(record.exc_info)
        return jsonlib.dumps(entry)


def _setup_logging(verbosity: int, log_format: str = "text"):
    level = logging.WARNING
    if verbosity == 1:
        level = logging.INFO
    elif verbosity >= 2:
        level = logging.DEBUG
    handler = logging.StreamHandler(sys.stderr)
    if log_format == "json":
        handler.setFormatter(_JsonFormatter())
    else:
        handler.setFormatter(
            logging.Formatter("%(asctime)s %(levelname).1s %(name)s %(message)s")
        )
    root = logging.getLogger()
    root.setLevel(level)
    root.addHandler(handler)


@click.group()
@click.option("-v", "--verbose", count=True, help="Increase log verbosity (klog-style).")
@click.option("--log-format", type=click.Choice(["text", "json"]), default="text", show_default=True)
def cli(verbose: int, log_format: str):
    """aws-global-accelerator-controller (agac)."""
    _setup_logging(verbose, log_format)


def resolve_kubeconfig(kubeconfig: str) -> str:
    """flag → $KUBECONFIG → ~/.kube/config (reference controller.go:84-98)."""
    if kubeconfig:
        return kubeconfig
    env = os.environ.get("KUBECONFIG", "")
    if env:
        return env
    default = os.path.expanduser("~/.kube/config")
    if os.path.exists(default):
        return default
    return ""


@cli.command()
@click.option("-w", "--workers", default=1, show_default=True, help="Concurrent workers number for controller.")
@click.option("-c", "--cluster-name", default="default", show_default=True, help="Owner cluster name which is used in resource tags.")
@click.option("--kubeconfig", default="", help="Path to a kubeconfig. Only required if out-of-cluster.")
@click.option("--master", default="", help="The address of the Kubernetes API server. Overrides any value in kubeconfig.")
@click.option("--api", type=click.Choice(["memory", "http", "k8s"]), default="memory", show_default=True, help="Kube API backend: embedded in-memory store, an agac HTTP API server, or a real Kubernetes API server (kubeconfig/in-cluster auth).")
@click.option("--cloud", type=click.Choice(["auto", "aws", "fake"]), default="auto", show_default=True, help="Cloud backend: auto (aws for http/k8s APIs, fake for memory), aws (boto3, required), fake (in-memory).")
@click.option("--metrics-port", default=0, help="Serve Prometheus metrics on this port (0 = disabled).")
@click.option("--health-port", default=0, help="Serve /healthz (liveness) and /readyz (caches synced) on this port (0 = disabled).")
@click.option("--cloud-resync-minutes", default=0.0, show_default=True, help="Re-enqueue ALL managed objects every N minutes even if unchanged, repairing cloud-side drift (0 = disabled, matching the reference: drift on unchanged objects is never repaired — docs/PARITY.md).")
@click.option("--leader-elect/--no-leader-elect", default=True, show_default=True)
def controller(workers, cluster_name, kubeconfig, master, api, cloud, metrics_port, health_port, cloud_resync_minutes, leader_elect):
    """Start controller."""
    from .controller.endpointgroupbinding import EndpointGroupBindingConfig
    from .controller.globalaccelerator import GlobalAcceleratorConfig
    from .controller.route53 import Route53Config
    from .kube.leaderelection import LeaderElector
    from .manager import ControllerConfig, Manager
    from .metrics import start_metrics_server
    from .signals import setup_signal_handler

    if api == "memory":
        from .kube.client import InMemoryKubeClient

        kube_client = InMemoryKubeClient()
        logger.info("Using embedded in-memory API store")
    elif api == "k8s":
        from .kube.k8s import K8sKubeClient
        from .kube.kubeconfig import build_config

        kc = resolve_kubeconfig(kubeconfig)
        if kc:
            logger.info("Using kubeconfig: %s", kc)
        else:
            logger.info("Using in-cluster config")
        kube_client = K8sKubeClient(build_config(master, kc))
    else:
        from .kube.rest import RestKubeClient

        server = master or os.environ.get("AGAC_API_SERVER", "")
        if not server:
            raise click.UsageError("--api http requires --master or $AGAC_API_SERVER")
        kube_client = RestKubeClient(server)

    # Cloud backend selection is explicit: a controller that silently
    # mutates a throwaway fake while looking healthy would be dangerous,
    # so "auto" only picks the fake for the embedded API.
    if cloud == "auto":
        cloud = "fake" if api == "memory" else "aws"
    if cloud == "aws":
        from .cloudprovider.aws.client import boto3_cloud_factory

        try:
            cloud_factory = boto3_cloud_factory()
        except RuntimeError as e:
            raise click.UsageError(
                f"{e} (use --cloud fake for a demo without AWS credentials)"
            ) from e
    else:
        from .cloudprovider.aws.client import FakeCloudFactory

        cloud_factory = FakeCloudFactory()
        logger.info("Using the in-memory AWS fake")

    if api in ("http", "k8s"):
        # fail fast with a clear message if the API server is unreachable
        try:
            kube_client.list("Lease", "default")
        except Exception as e:
            raise click.ClickException(
                f"cannot reach the {api} API server: {e}"
            ) from e

    if metrics_port:
        start_metrics_server(metrics_port)

    namespace = os.environ.get("POD_NAMESPACE", "default")
    cloud_resync = max(0.0, cloud_resync_minutes) * 60.0
    config = ControllerConfig(
        global_accelerator=GlobalAcceleratorConfig(
            workers=workers, cluster_name=cluster_name,
            cloud_resync_period=cloud_resync,
        ),
        route53=Route53Config(
            workers=workers, cluster_name=cluster_name,
            cloud_resync_period=cloud_resync,
        ),
        endpoint_group_binding=EndpointGroupBindingConfig(
            workers=workers, cloud_resync_period=cloud_resync,
        ),
    )
    stop = setup_signal_handler()

    current_manager = {}
    if health_port:
        from .health import HealthServer

        health = HealthServer(
            health_port,
            ready_fn=lambda: (
                current_manager.get("m") is not None
                and current_manager["m"].is_ready()
            ),
        )
        health.start()
        logger.info("Health endpoints on :%d (/healthz, /readyz)", health.port)

    def run_manager(stop_leading):
        manager = Manager()
        current_manager["m"] = manager
        try:
            manager.run(kube_client, config, cloud_factory, stop_leading, block=True)
        finally:
            current_manager.pop("m", None)

    if leader_elect:
        elector = LeaderElector(
            kube_client,
            name="aws-global-accelerator-controller",
            namespace=namespace,
            on_started_leading=run_manager,
            on_stopped_leading=lambda: logger.info("leadership ended"),
            on_new_leader=lambda ident: logger.info("new leader elected: %s", ident),
        )
        logger.info("leader election id: %s", elector.identity)
        elector.run(stop)
        # reference exits 0 after losing/releasing the lease
        sys.exit(0)
    else:
        run_manager(stop)


@cli.command()
@click.option("--tls-cert-file", default="", help="TLS certificate file path.")
@click.option("--tls-private-key-file", default="", help="TLS private key file path.")
@click.option("--port", default=8443, show_default=True, help="Listen port.")
@click.option("--ssl/--no-ssl", default=True, show_default=True, help="Enable TLS.")
def webhook(tls_cert_file, tls_private_key_file, port, ssl):
    """Start validating admission webhook server."""
    from .webhook.server import serve

    if not ssl:
        tls_cert_file = tls_private_key_file = ""
    serve(port, tls_cert_file, tls_private_key_file)


@cli.command()
def version():
    """Print version/revision/build (reference cmd/version.go injects these
    via -ldflags; here revision/build come from the environment or git)."""
    import subprocess

    revision = os.environ.get("AGAC_REVISION", "")
    if not revision:
        try:
            revision = subprocess.run(
                ["git", "rev-parse", "--short", "HEAD"],
                capture_output=True, text=True, timeout=5,
                cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            ).stdout.strip() or "unknown"
        except Exception:
            revision = "unknown"
    build = os.environ.get("AGAC_BUILD", "source")
    click.echo(f"aws-global-accelerator-controller (agac) {__version__}")
    click.echo(f"revision: {revision}")
    click.echo(f"build: {build}")


@cli.command("apiserver")
@click.option("--port", default=8001, show_default=True, help="Listen port.")
@click.option("--state-file", default="", help="Snapshot file: loaded on start if present, written on shutdown (checkpoint/resume).")
@click.option("--checkpoint-interval-seconds", default=0.0, show_default=True, help="Also snapshot --state-file every N seconds (crash resilience; 0 = shutdown-only).")
@click.option("--resolve-webhook-service", "webhook_services", multiple=True, metavar="NAME.NAMESPACE=URL", help="Resolve a ValidatingWebhookConfiguration service reference to a URL (clusters use <name>.<ns>.svc DNS; repeatable).")
@click.option("--token", default="", help="Require `Authorization: Bearer <token>` on every request except /healthz (static-token authn).")
@click.option("--tls-cert-file", default="", help="Serve HTTPS with this certificate (pair with --tls-private-key-file).")
@click.option("--tls-private-key-file", default="", help="TLS private key.")
def apiserver(port, state_file, checkpoint_interval_seconds, webhook_services, token, tls_cert_file, tls_private_key_file):
    """Serve the in-memory API store over HTTP (hermetic e2e backend)."""
    import json as jsonlib

    from .kube.httpapi import APIServer
    from .kube.store import APIStore
    from .signals import setup_signal_handler

    if state_file and os.path.exists(state_file):
        with open(state_file) as f:
            store = APIStore.load(jsonlib.load(f))
        logger.info("Restored state from %s", state_file)
    else:
        store = APIStore()

    if webhook_services:
        mapping = {}
        for entry in webhook_services:
            ref, _, url = entry.partition("=")
            if not url:
                raise click.UsageError(
                    f"--resolve-webhook-service wants NAME.NAMESPACE=URL, got {entry!r}"
                )
            mapping[ref] = url

        def resolver(service_ref):
            return mapping.get(f"{service_ref.name}.{service_ref.namespace}")

        store.webhook_service_resolver = resolver
        logger.info("Webhook service resolver: %s", ", ".join(mapping))

    server = APIServer(store, port, host="", bearer_token=token or None,
                       tls_cert_file=tls_cert_file,
                       tls_key_file=tls_private_key_file)
    server.start()
    logger.info("API server listening on :%d%s%s", server.port,
                " (TLS)" if server.ssl_enabled else "",
                " (bearer-token authn)" if token else "")
    stop = setup_signal_handler()
    if state_file and checkpoint_interval_seconds > 0:
        store.start_checkpointer(state_file, checkpoint_interval_seconds, stop)
        logger.info(
            "Checkpointing to %s every %.0fs", state_file, checkpoint_interval_seconds
        )
    stop.wait()
    if state_file:
        store.save_snapshot(state_file)
        logger.info("State saved to %s", state_file)
    server.shutdown()


@cli.command("get")
@click.argument("kind")
@click.option("-n", "--namespace", default=None, help="Namespace filter (default: all).")
@click.option("--master", default="", help="agac API server URL (native wire scheme).")
def get_cmd(kind, namespace, master):
    """List objects of KIND (service/ingress/endpointgroupbinding/lease/event)
    from an agac API server — kubectl-get for the embedded backends."""
    from .kube.client import class_for_kind
    from .kube.rest import RestKubeClient

    aliases = {
        "service": "Service", "services": "Service", "svc": "Service",
        "ingress": "Ingress", "ingresses": "Ingress", "ing": "Ingress",
        "endpointgroupbinding": "EndpointGroupBinding",
        "endpointgroupbindings": "EndpointGroupBinding", "egb": "EndpointGroupBinding",
        "lease": "Lease", "leases": "Lease",
        "event": "Event", "events": "Event",
        "validatingwebhookconfiguration": "ValidatingWebhookConfiguration",
        "validatingwebhookconfigurations": "ValidatingWebhookConfiguration",
        "vwc": "ValidatingWebhookConfiguration",
    }
    resolved = aliases.get(kind.lower())
    if resolved is None:
        raise click.UsageError(f"unknown kind {kind!r}; one of {sorted(set(aliases))}")
    class_for_kind(resolved)  # validate early
    server = master or os.environ.get("AGAC_API_SERVER", "http://127.0.0.1:8001")
    client = RestKubeClient(server)
    items, _ = client.list(resolved, namespace)
    click.echo(f"{'NAMESPACE':<16} {'NAME':<40} {'RV':<8} DETAIL")
    for obj in items:
        detail = ""
        if resolved == "EndpointGroupBinding":
            detail = f"endpoints={len(obj.status.endpoint_ids)} gen={obj.metadata.generation} observed={obj.status.observed_generation}"
        elif resolved == "Lease":
            detail = f"holder={obj.spec.holder_identity}"
        elif resolved == "ValidatingWebhookConfiguration":
            detail = ", ".join(
                f"{w.name}({w.failure_policy})" for w in obj.webhooks
            )
        elif resolved == "Event":
            detail = f"{obj.reason} x{obj.count}"
        click.echo(
            f"{obj.metadata.namespace:<16} {obj.metadata.name:<40} "
            f"{obj.metadata.resource_version:<8} {detail}"
        )


@cli.command("apply")
@click.option("-f", "--filename", multiple=True, required=True, type=click.Path(exists=True), help="YAML file(s) to apply (repeatable).")
@click.option("--master", default="", help="agac API server URL (native wire scheme).")
def apply_cmd(filename, master):
    """Apply YAML manifests to an agac API server (kubectl-apply-lite;
    unknown kinds are skipped)."""
    from .kube.apply import apply_yaml
    from .kube.rest import RestKubeClient

    server = master or os.environ.get("AGAC_API_SERVER", "http://127.0.0.1:8001")
    client = RestKubeClient(server)
    for path in filename:
        with open(path) as f:
            for action, ident in apply_yaml(client, f.read()):
                click.echo(f"{ident} {action}")


@cli.command()
@click.option("--objects", default=3, show_default=True, help="Sample services to reconcile.")
def demo(objects):
    """Guided demo: run the full stack in-process (embedded API + AWS fake),
    reconcile sample services into Global Accelerators + Route53 records,
    then tear one down — printing the resulting cloud state and events."""
    import threading
    import time

    from .apis import core as corev1
    from .apis.meta import ObjectMeta
    from .cloudprovider.aws.client import FakeCloudFactory
    from .cloudprovider.fake import FakeAWSBackend
    from .kube.client import InMemoryKubeClient
    from .manager import ControllerConfig, Manager

    MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
    HOSTNAME = "aws-global-accelerator-controller.h3poteto.dev/route53-hostname"
    LB_TYPE = "service.beta.kubernetes.io/aws-load-balancer-type"

    client = InMemoryKubeClient()
    backend = FakeAWSBackend()
    backend.route53.create_hosted_zone("demo.example.com")
    stop = threading.Event()
    manager = Manager()
    manager.run(
        client, ControllerConfig(),
        FakeCloudFactory(backend, ga_missing_retry=0.1),
        stop, resync_period=5.0, block=False,
    )
    manager.wait_until_ready()
    click.echo(f"controllers up; creating {objects} LoadBalancer Services "
               "with managed + route53 annotations...")
    try:
        for i in range(objects):
            lb = backend.elbv2.create_load_balancer(f"demo-{i}", region="us-east-1")
            client.create(
                corev1.Service(
                    metadata=ObjectMeta(
                        name=f"demo-{i}", namespace="default",
                        annotations={
                            LB_TYPE: "nlb", MANAGED: "true",
                            HOSTNAME: f"demo-{i}.demo.example.com",
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
            )

        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            accs, _ = backend.ga.list_accelerators()
            zones, _ = backend.route53.list_hosted_zones()
            records, _ = backend.route53.list_resource_record_sets(zones[0].id)
            if len(accs) == objects and len(records) == 2 * objects:
                break
            time.sleep(0.05)

        click.echo("\nGlobal Accelerators:")
        for acc in accs:
            listeners, _ = backend.ga.list_listeners(acc.accelerator_arn)
            ports = [p.from_port for p in listeners[0].port_ranges]
            click.echo(f"  {acc.name:<24} {acc.dns_name:<44} ports={ports}")
        click.echo("\nRoute53 records (demo.example.com):")
        for record in records:
            target = record.alias_target.dns_name if record.alias_target else \
                record.resource_records[0].value[:40] + "..."
            click.echo(f"  {record.type:<4} {record.name:<34} -> {target}")

        # EndpointGroupBinding segment: bind demo-1's LB to an externally
        # managed endpoint group
        from .apis import endpointgroupbinding as egb

        ext = backend.ga.create_accelerator("external-demo")
        from .cloudprovider.aws import types as awstypes

        ext_listener = backend.ga.create_listener(
            ext.accelerator_arn, [awstypes.PortRange(80, 80)], "TCP"
        )
        ext_group = backend.ga.create_endpoint_group(
            ext_listener.listener_arn, "us-east-1"
        )
        client.create(
            egb.EndpointGroupBinding(
                metadata=ObjectMeta(name="demo-binding", namespace="default"),
                spec=egb.EndpointGroupBindingSpec(
                    endpoint_group_arn=ext_group.endpoint_group_arn,
                    weight=128,
                    service_ref=egb.ServiceReference(name="demo-1"),
                ),
            )
        )
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            binding = client.get("EndpointGroupBinding", "default", "demo-binding")
            if binding.status.endpoint_ids:
                break
            time.sleep(0.05)
        click.echo(
            f"\nEndpointGroupBinding demo-binding: endpoints="
            f"{len(binding.status.endpoint_ids)} weight=128 "
            f"(finalizer={binding.metadata.finalizers})"
        )

        click.echo(f"\ndeleting Service demo-0 ...")
        client.delete("Service", "default", "demo-0")
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            accs, _ = backend.ga.list_accelerators()
            if len(accs) == objects - 1:
                break
            time.sleep(0.05)
        click.echo(f"accelerators after delete: {len(accs)} (was {objects})")

        click.echo("\nEvents:")
        events, _ = client.list("Event")
        for ev in events:
            click.echo(f"  {ev.involved_object.name:<10} {ev.reason:<26} x{ev.count}")
    finally:
        stop.set()
    click.echo("\ndemo OK")


def main():
    cli()


if __name__ == "__main__":
    main()
