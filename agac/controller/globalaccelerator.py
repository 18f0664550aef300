"""GlobalAccelerator controller.

Watches Services and Ingresses (reference
``pkg/controller/globalaccelerator/``): objects carrying the
``global-accelerator-managed`` annotation get an accelerator→listener→
endpoint-group triple per LoadBalancer-status hostname; removing the
annotation or deleting the object tears the accelerators down.
Two independent rate-limited queues (service / ingress) with
``threadiness`` workers each.
"""

from __future__ import annotations

import logging
import threading
from dataclasses import dataclass

from .. import reconcile
from ..apis import AWS_GLOBAL_ACCELERATOR_MANAGED_ANNOTATION
from ..apis import core as corev1
from ..apis.meta import split_meta_namespace_key, meta_namespace_key
from ..cloudprovider import detect_cloud_provider
from ..cloudprovider.aws import get_lb_name_from_hostname
from ..errors import new_no_retry_errorf
from ..kube.events import EventRecorder
from ..kube.informer import wait_for_cache_sync
from ..kube.workqueue import RateLimitingQueue
from .base import (
    make_queue_rate_limiter,
    spawn_cloud_resync,
    has_managed_annotation,
    managed_annotation_changed,
    objects_equal,
    spawn_workers,
    was_alb_ingress,
    was_load_balancer_service,
)

logger = logging.getLogger(__name__)

CONTROLLER_AGENT_NAME = "global-accelerator-controller"


@dataclass
class GlobalAcceleratorConfig:
    workers: int = 1
    cluster_name: str = "default"
    # queue token-bucket rate (client-go default 10/100); raise for scale
    queue_qps: float = 10.0
    queue_burst: int = 100
    # per-item failure-backoff bounds (client-go defaults)
    queue_item_base_delay: float = 0.005
    queue_item_max_delay: float = 1000.0
    # opt-in drift repair: re-enqueue unchanged managed objects every N
    # seconds (0 = reference-parity behavior: cloud drift on an unchanged
    # object is never repaired — see docs/PARITY.md §resync)
    cloud_resync_period: float = 0.0


class GlobalAcceleratorController:
    def __init__(self, kube_client, informer_factory, config, cloud_factory):
        self.cluster_name = config.cluster_name
        self.cloud_resync_period = config.cloud_resync_period
        self.kube_client = kube_client
        self.cloud_factory = cloud_factory
        # (resource, ns, name) -> accelerator ARN hint.  Purely an API-cost
        # optimization: the resource manager re-verifies ownership tags
        # before trusting a hint, and any miss falls back to the full
        # ListAccelerators scan (see ensure_global_accelerator_for_service).
        self._arn_hints = {}
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

    # -- notifications (reference ga/controller.go:91-173) -----------------
    def _add_service_notification(self, svc):
        if was_load_balancer_service(svc) and has_managed_annotation(svc):
            logger.debug("Service %s is created", meta_namespace_key(svc))
            self._enqueue_service(svc)

    def _update_service_notification(self, old, new):
        if objects_equal(old, new):
            return
        if was_load_balancer_service(new):
            if has_managed_annotation(new) or managed_annotation_changed(old, new):
                self._enqueue_service(new)

    def _delete_service_notification(self, svc):
        if was_load_balancer_service(svc):
            logger.debug("Deleting Service %s", meta_namespace_key(svc))
            self._enqueue_service(svc)

    def _add_ingress_notification(self, ingress):
        if was_alb_ingress(ingress) and has_managed_annotation(ingress):
            self._enqueue_ingress(ingress)

    def _update_ingress_notification(self, old, new):
        if objects_equal(old, new):
            return
        if was_alb_ingress(new):
            if has_managed_annotation(new) or managed_annotation_changed(old, new):
                self._enqueue_ingress(new)

    def _delete_ingress_notification(self, ingress):
        # reference enqueues ingress deletes unconditionally (ga/controller.go:170)
        self._enqueue_ingress(ingress)

    def _enqueue_service(self, obj):
        self.service_queue.add_rate_limited(meta_namespace_key(obj))

    def _enqueue_ingress(self, obj):
        self.ingress_queue.add_rate_limited(meta_namespace_key(obj))

    # -- run (reference ga/controller.go:195-230) ---------------------------
    def run(self, threadiness: int, stop: threading.Event):
        try:
            self._run(threadiness, stop)
        finally:
            # teardown on EVERY exit path (incl. shutdown-before-synced):
            # queues release workers, the recorder's writer thread exits
            self.service_queue.shut_down()
            self.ingress_queue.shut_down()
            self.recorder.stop()

    def _run(self, threadiness: int, stop: threading.Event):
        logger.info("Starting GlobalAccelerator controller")
        if not wait_for_cache_sync(stop, self.service_informer, self.ingress_informer):
            if stop.is_set():
                return  # shutdown requested before caches synced
            raise RuntimeError("failed to wait for caches to sync")
        logger.info("Starting workers")
        spawn_workers(threadiness, self._run_service_worker, CONTROLLER_AGENT_NAME + "-service", stop)
        spawn_workers(threadiness, self._run_ingress_worker, CONTROLLER_AGENT_NAME + "-ingress", stop)
        spawn_cloud_resync(
            self.cloud_resync_period,
            stop,
            [
                (
                    self.service_lister.list,
                    lambda o: was_load_balancer_service(o) and has_managed_annotation(o),
                    self._enqueue_service,
                ),
                (
                    self.ingress_lister.list,
                    lambda o: was_alb_ingress(o) and has_managed_annotation(o),
                    self._enqueue_ingress,
                ),
            ],
            CONTROLLER_AGENT_NAME,
        )
        stop.wait()
        logger.info("Shutting down workers")

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

    # -- service processors (reference ga/service.go:28-126) ----------------
    def process_service_delete(self, key: str) -> reconcile.Result:
        logger.info("%s has been deleted", key)
        try:
            ns, name = split_meta_namespace_key(key)
        except ValueError:
            raise new_no_retry_errorf("invalid resource key: %s", key)
        self._cleanup_accelerators("service", ns, name)
        return reconcile.Result()

    def _cleanup_accelerators(self, resource: str, ns: str, name: str):
        with self._hints_lock:
            self._arn_hints.pop((resource, ns, name), None)
        cloud = self.cloud_factory("us-west-2")
        accelerators = cloud.list_global_accelerator_by_resource(
            self.cluster_name, resource, ns, name
        )
        for accelerator in accelerators:
            cloud.cleanup_global_accelerator(accelerator.accelerator_arn)

    def _hint_for(self, resource: str, ns: str, name: str):
        with self._hints_lock:
            return self._arn_hints.get((resource, ns, name))

    def _remember_hint(self, resource: str, ns: str, name: str, arn):
        with self._hints_lock:
            if arn:
                self._arn_hints[(resource, ns, name)] = arn

    def process_service_create_or_update(self, svc) -> reconcile.Result:
        if len(svc.status.load_balancer.ingress) < 1:
            logger.warning(
                "%s does not have ingress LoadBalancer, so skip it",
                meta_namespace_key(svc),
            )
            return reconcile.Result()

        if AWS_GLOBAL_ACCELERATOR_MANAGED_ANNOTATION not in svc.metadata.annotations:
            self._cleanup_accelerators(
                "service", svc.metadata.namespace, svc.metadata.name
            )
            logger.info(
                "Delete Global Accelerator for Service %s", meta_namespace_key(svc)
            )
            self.recorder.event(
                svc,
                corev1.EVENT_TYPE_NORMAL,
                "GlobalAcceleratorDeleted",
                "Global Accelerators are deleted",
            )
            return reconcile.Result()

        for lb_ingress in svc.status.load_balancer.ingress:
            try:
                provider = detect_cloud_provider(lb_ingress.hostname)
            except ValueError as e:
                logger.error(str(e))
                continue
            if provider != "aws":
                logger.warning("Not implemented for %s", provider)
                continue
            name, region = get_lb_name_from_hostname(lb_ingress.hostname)
            cloud = self.cloud_factory(region)
            arn, created, retry_after = cloud.ensure_global_accelerator_for_service(
                svc, lb_ingress, self.cluster_name, name, region,
                hint_arn=self._hint_for(
                    "service", svc.metadata.namespace, svc.metadata.name
                ),
            )
            self._remember_hint(
                "service", svc.metadata.namespace, svc.metadata.name, arn
            )
            if retry_after > 0:
                return reconcile.Result(requeue=True, requeue_after=retry_after)
            if created:
                self.recorder.eventf(
                    svc,
                    corev1.EVENT_TYPE_NORMAL,
                    "GlobalAcceleratorCreated",
                    "Global Acclerator is created: %s",
                    arn,
                )
        return reconcile.Result()

    # -- ingress processors (reference ga/ingress.go:29-130) ----------------
    def process_ingress_delete(self, key: str) -> reconcile.Result:
        logger.info("%s has been deleted", key)
        try:
            ns, name = split_meta_namespace_key(key)
        except ValueError:
            raise new_no_retry_errorf("invalid resource key: %s", key)
        self._cleanup_accelerators("ingress", ns, name)
        return reconcile.Result()

    def process_ingress_create_or_update(self, ingress) -> reconcile.Result:
        if len(ingress.status.load_balancer.ingress) < 1:
            logger.warning(
                "%s does not have ingress LoadBalancer, so skip it",
                meta_namespace_key(ingress),
            )
            return reconcile.Result()

        if AWS_GLOBAL_ACCELERATOR_MANAGED_ANNOTATION not in ingress.metadata.annotations:
            self._cleanup_accelerators(
                "ingress", ingress.metadata.namespace, ingress.metadata.name
            )
            logger.info(
                "Delete Global Accelerator for Ingress %s", meta_namespace_key(ingress)
            )
            self.recorder.event(
                ingress,
                corev1.EVENT_TYPE_NORMAL,
                "GlobalAcceleratorDeleted",
                "Global Accelerators are deleted",
            )
            return reconcile.Result()

        for lb_ingress in ingress.status.load_balancer.ingress:
            try:
                provider = detect_cloud_provider(lb_ingress.hostname)
            except ValueError as e:
                logger.error(str(e))
                continue
            if provider != "aws":
                logger.warning("Not implemented for %s", provider)
                continue
            name, region = get_lb_name_from_hostname(lb_ingress.hostname)
            cloud = self.cloud_factory(region)
            arn, created, retry_after = cloud.ensure_global_accelerator_for_ingress(
                ingress, lb_ingress, self.cluster_name, name, region,
                hint_arn=self._hint_for(
                    "ingress", ingress.metadata.namespace, ingress.metadata.name
                ),
            )
            self._remember_hint(
                "ingress", ingress.metadata.namespace, ingress.metadata.name, arn
            )
            if retry_after > 0:
                return reconcile.Result(requeue=True, requeue_after=retry_after)
            if created:
                self.recorder.eventf(
                    ingress,
                    corev1.EVENT_TYPE_NORMAL,
                    "GlobalAcceleratorCreated",
                    "Global Acclerator is created: %s",
                    arn,
                )
        return reconcile.Result()
