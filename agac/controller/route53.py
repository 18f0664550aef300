"""Route53 controller.

Watches Services and Ingresses for the ``route53-hostname`` annotation
(reference ``pkg/controller/route53/``): a comma-separated hostname list is
reconciled into A-ALIAS records (pointing at the owning Global Accelerator)
plus TXT ownership records; removing the annotation or deleting the object
cleans the records up.  Note the reference's asymmetric filters: the
service paths require the LoadBalancer-service shape, the ingress add/update
paths only require the annotation (r53/controller.go:130-166) — mirrored
here.
"""

from __future__ import annotations

import logging
import threading
from dataclasses import dataclass

from .. import reconcile
from ..apis import ROUTE53_HOSTNAME_ANNOTATION
from ..apis import core as corev1
from ..apis.meta import meta_namespace_key, split_meta_namespace_key
from ..cloudprovider import detect_cloud_provider
from ..cloudprovider.aws import get_lb_name_from_hostname
from ..errors import new_no_retry_errorf
from ..kube.events import EventRecorder
from ..kube.informer import wait_for_cache_sync
from ..kube.workqueue import RateLimitingQueue
from .base import (
    make_queue_rate_limiter,
    has_hostname_annotation,
    hostname_annotation_changed,
    objects_equal,
    spawn_cloud_resync,
    spawn_workers,
    was_load_balancer_service,
)

logger = logging.getLogger(__name__)

CONTROLLER_AGENT_NAME = "route53-controller"


@dataclass
class Route53Config:
    workers: int = 1
    cluster_name: str = "default"
    # queue token-bucket rate (client-go default 10/100); raise for scale
    queue_qps: float = 10.0
    queue_burst: int = 100
    # per-item failure-backoff bounds (client-go defaults)
    queue_item_base_delay: float = 0.005
    queue_item_max_delay: float = 1000.0
    # opt-in drift repair (see docs/PARITY.md §resync); 0 = parity
    cloud_resync_period: float = 0.0


class Route53Controller:
    def __init__(self, kube_client, informer_factory, config, cloud_factory):
        self.cluster_name = config.cluster_name
        self.cloud_resync_period = config.cloud_resync_period
        self.kube_client = kube_client
        self.cloud_factory = cloud_factory
        # LB hostname -> accelerator ARN hint (tag-verified before use;
        # misses fall back to the full by-hostname scan)
        self._arn_hints = {}
        # hostname -> HostedZone hint: skips the parent-domain zone walk;
        # a stale entry (zone gone) is dropped by the resource manager and
        # the walk re-runs (see Route53Mixin.ensure_route53_for_service)
        self._zone_hints = {}
        self._hints_lock = threading.Lock()
        self.recorder = EventRecorder(kube_client, CONTROLLER_AGENT_NAME)
        self.service_queue = RateLimitingQueue(
            rate_limiter=make_queue_rate_limiter(config.queue_qps, config.queue_burst, config.queue_item_base_delay, config.queue_item_max_delay),
            name=CONTROLLER_AGENT_NAME + "-service",
        )
        self.ingress_queue = RateLimitingQueue(
            rate_limiter=make_queue_rate_limiter(config.queue_qps, config.queue_burst, config.queue_item_base_delay, config.queue_item_max_delay),
            name=CONTROLLER_AGENT_NAME + "-ingress",
        )

        service_informer = informer_factory.services()
        self.service_lister = service_informer.lister()
        self.service_informer = service_informer
        service_informer.add_event_handler(
            on_add=self._add_service_notification,
            on_update=self._update_service_notification,
            on_delete=self._delete_service_notification,
        )

        ingress_informer = informer_factory.ingresses()
        self.ingress_lister = ingress_informer.lister()
        self.ingress_informer = ingress_informer
        ingress_informer.add_event_handler(
            on_add=self._add_ingress_notification,
            on_update=self._update_ingress_notification,
            on_delete=self._delete_ingress_notification,
        )

    # -- notifications (reference r53/controller.go:87-166) -----------------
    def _add_service_notification(self, svc):
        if was_load_balancer_service(svc) and has_hostname_annotation(svc):
            self._enqueue_service(svc)

    def _update_service_notification(self, old, new):
        if objects_equal(old, new):
            return
        if was_load_balancer_service(new):
            if has_hostname_annotation(new) or hostname_annotation_changed(old, new):
                self._enqueue_service(new)

    def _delete_service_notification(self, svc):
        if was_load_balancer_service(svc):
            self._enqueue_service(svc)

    def _add_ingress_notification(self, ingress):
        if has_hostname_annotation(ingress):
            self._enqueue_ingress(ingress)

    def _update_ingress_notification(self, old, new):
        if objects_equal(old, new):
            return
        if has_hostname_annotation(new) or hostname_annotation_changed(old, new):
            self._enqueue_ingress(new)

    def _delete_ingress_notification(self, ingress):
        self._enqueue_ingress(ingress)

    def _enqueue_service(self, obj):
        self.service_queue.add_rate_limited(meta_namespace_key(obj))

    def _enqueue_ingress(self, obj):
        self.ingress_queue.add_rate_limited(meta_namespace_key(obj))

    def _hint_for(self, lb_hostname: str):
        with self._hints_lock:
            return self._arn_hints.get(lb_hostname)

    def _remember_hint(self, lb_hostname: str, cloud):
        """Record the accelerator the ensure call actually matched (exposed
        by the resource manager) — seeds and self-heals the hint without
        any extra API call."""
        arn = getattr(cloud, "_last_matched_accelerator_arn", None)
        if arn:
            with self._hints_lock:
                self._arn_hints[lb_hostname] = arn

    # -- run ----------------------------------------------------------------
    def run(self, threadiness: int, stop: threading.Event):
        try:
            self._run(threadiness, stop)
        finally:
            self.service_queue.shut_down()
            self.ingress_queue.shut_down()
            self.recorder.stop()

    def _run(self, threadiness: int, stop: threading.Event):
        logger.info("Starting Route53 controller")
        if not wait_for_cache_sync(stop, self.service_informer, self.ingress_informer):
            if stop.is_set():
                return  # shutdown requested before caches synced
            raise RuntimeError("failed to wait for caches to sync")
        spawn_workers(threadiness, self._run_service_worker, CONTROLLER_AGENT_NAME + "-service", stop)
        spawn_workers(threadiness, self._run_ingress_worker, CONTROLLER_AGENT_NAME + "-ingress", stop)
        spawn_cloud_resync(
            self.cloud_resync_period,
            stop,
            [
                (
                    self.service_lister.list,
                    lambda o: was_load_balancer_service(o) and has_hostname_annotation(o),
                    self._enqueue_service,
                ),
                (
                    self.ingress_lister.list,
                    has_hostname_annotation,
                    self._enqueue_ingress,
                ),
            ],
            CONTROLLER_AGENT_NAME,
        )
        stop.wait()

    def _run_service_worker(self):
        while reconcile.process_next_work_item(
            self.service_queue,
            self._key_to_service,
            self.process_service_delete,
            self.process_service_create_or_update,
        ):
            pass

    def _run_ingress_worker(self):
        while reconcile.process_next_work_item(
            self.ingress_queue,
            self._key_to_ingress,
            self.process_ingress_delete,
            self.process_ingress_create_or_update,
        ):
            pass

    def _key_to_service(self, key: str):
        ns, name = split_meta_namespace_key(key)
        return self.service_lister.get(name, namespace=ns)

    def _key_to_ingress(self, key: str):
        ns, name = split_meta_namespace_key(key)
        return self.ingress_lister.get(name, namespace=ns)

    # -- service processors (reference r53/service.go:29-111) ---------------
    def process_service_delete(self, key: str) -> reconcile.Result:
        logger.info("%s has been deleted", key)
        try:
            ns, name = split_meta_namespace_key(key)
        except ValueError:
            raise new_no_retry_errorf("invalid resource key: %s", key)
        cloud = self.cloud_factory("us-west-2")
        cloud.cleanup_record_set(self.cluster_name, "service", ns, name)
        return reconcile.Result()

    def process_service_create_or_update(self, svc) -> reconcile.Result:
        hostname = svc.metadata.annotations.get(ROUTE53_HOSTNAME_ANNOTATION)
        if hostname is None:
            cloud = self.cloud_factory("us-west-2")
            cloud.cleanup_record_set(
                self.cluster_name, "service", svc.metadata.namespace, svc.metadata.name
            )
            logger.info("Delete route53 records for Service %s", meta_namespace_key(svc))
            self.recorder.event(
                svc,
                corev1.EVENT_TYPE_NORMAL,
                "Route53RecordDeleted",
                "Route53 record sets are deleted",
            )
            return reconcile.Result()

        hostnames = hostname.split(",")
        for lb_ingress in svc.status.load_balancer.ingress:
            try:
                provider = detect_cloud_provider(lb_ingress.hostname)
            except ValueError as e:
                logger.error(str(e))
                continue
            if provider != "aws":
                logger.warning("Not implemented for %s", provider)
                continue
            _, region = get_lb_name_from_hostname(lb_ingress.hostname)
            cloud = self.cloud_factory(region)
            created, retry_after = cloud.ensure_route53_for_service(
                svc, lb_ingress, hostnames, self.cluster_name,
                hint_arn=self._hint_for(lb_ingress.hostname),
                zone_hints=self._zone_hints,
            )
            if retry_after == 0:
                self._remember_hint(lb_ingress.hostname, cloud)
            if retry_after > 0:
                return reconcile.Result(requeue=True, requeue_after=retry_after)
            if created:
                # the reason string keeps the reference's typo on the service
                # path for event-stream parity (r53/service.go:105)
                self.recorder.eventf(
                    svc,
                    corev1.EVENT_TYPE_NORMAL,
                    "Route53RecourdCreated",
                    "Route53 record set is created: %s",
                    hostnames,
                )
        return reconcile.Result()

    # -- ingress processors (reference r53/ingress.go:20-104) ---------------
    def process_ingress_delete(self, key: str) -> reconcile.Result:
        logger.info("%s has been deleted", key)
        try:
            ns, name = split_meta_namespace_key(key)
        except ValueError:
            raise new_no_retry_errorf("invalid resource key: %s", key)
        cloud = self.cloud_factory("us-west-2")
        cloud.cleanup_record_set(self.cluster_name, "ingress", ns, name)
        return reconcile.Result()

    def process_ingress_create_or_update(self, ingress) -> reconcile.Result:
        hostname = ingress.metadata.annotations.get(ROUTE53_HOSTNAME_ANNOTATION)
        if hostname is None:
            cloud = self.cloud_factory("us-west-2")
            cloud.cleanup_record_set(
                self.cluster_name,
                "ingress",
                ingress.metadata.namespace,
                ingress.metadata.name,
            )
            logger.info(
                "Delete route53 records for Ingress %s", meta_namespace_key(ingress)
            )
            self.recorder.event(
                ingress,
                corev1.EVENT_TYPE_NORMAL,
                "Route53RecordDeleted",
                "Route53 record sets are deleted",
            )
            return reconcile.Result()

        hostnames = hostname.split(",")
        for lb_ingress in ingress.status.load_balancer.ingress:
            try:
                provider = detect_cloud_provider(lb_ingress.hostname)
            except ValueError as e:
                logger.error(str(e))
                continue
            if provider != "aws":
                logger.warning("Not implemented for %s", provider)
                continue
            _, region = get_lb_name_from_hostname(lb_ingress.hostname)
            cloud = self.cloud_factory(region)
            created, retry_after = cloud.ensure_route53_for_ingress(
                ingress, lb_ingress, hostnames, self.cluster_name,
                hint_arn=self._hint_for(lb_ingress.hostname),
                zone_hints=self._zone_hints,
            )
            if retry_after == 0:
                self._remember_hint(lb_ingress.hostname, cloud)
            if retry_after > 0:
                return reconcile.Result(requeue=True, requeue_after=retry_after)
            if created:
                self.recorder.eventf(
                    ingress,
                    corev1.EVENT_TYPE_NORMAL,
                    "Route53RecordCreated",
                    "Route53 record set is created: %s",
                    hostnames,
                )
        return reconcile.Result()
