"""Lease-based leader election.

Replaces client-go's leaderelection + resourcelock.LeaseLock (the reference
wires it at ``pkg/leaderelection/leaderelection.go:21-84`` with lease 60s /
renew-deadline 15s / retry 5s, ReleaseOnCancel, and exit-on-lost-lease).
The Lease object lives in the same API store the controllers use, so a
2-replica failover is fully testable in-process.
"""

from __future__ import annotations

import logging
import threading
import time
import uuid
from dataclasses import dataclass
from typing import Callable, Optional

from ..apis import core as corev1
from ..apis.meta import ObjectMeta
from .store import ConflictError, NotFoundError

logger = logging.getLogger(__name__)


def _now() -> float:
    return time.time()


def _fmt(ts: float) -> str:
    """k8s MicroTime wire format: 2006-01-02T15:04:05.000000Z."""
    base = time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime(ts))
    return f"{base}.{int((ts % 1) * 1e6):06d}Z"


def _parse(ts: str) -> float:
    """Inverse of _fmt; tolerates missing fractional part."""
    import calendar

    body = ts.rstrip("Z")
    if "." in body:
        base, frac = body.split(".", 1)
        fraction = float("0." + frac)
    else:
        base, fraction = body, 0.0
    return calendar.timegm(time.strptime(base, "%Y-%m-%dT%H:%M:%S")) + fraction


@dataclass
class LeaderElectionConfig:
    lease_duration: float = 60.0
    renew_deadline: float = 15.0
    retry_period: float = 5.0
    release_on_cancel: bool = True


class LeaderElector:
    """Acquire/renew loop over a Lease object.

    Callbacks mirror client-go: ``on_started_leading(stop_leading_event)``
    runs in its own thread once the lock is acquired; ``on_stopped_leading``
    fires when the lease cannot be renewed (or on release);
    ``on_new_leader(identity)`` on observed leader changes.
    """

    def __init__(
        self,
        client,
        name: str,
        namespace: str,
        identity: Optional[str] = None,
        config: Optional[LeaderElectionConfig] = None,
        on_started_leading: Optional[Callable] = None,
        on_stopped_leading: Optional[Callable] = None,
        on_new_leader: Optional[Callable[[str], None]] = None,
    ):
        self.client = client
        self.name = name
        self.namespace = namespace
        self.identity = identity or str(uuid.uuid4())
        self.config = config or LeaderElectionConfig()
        self.on_started_leading = on_started_leading
        self.on_stopped_leading = on_stopped_leading
        self.on_new_leader = on_new_leader
        self._observed_leader: Optional[str] = None
        self._renew_time: float = 0.0
        self.is_leader = threading.Event()

    # -- lease record helpers ----------------------------------------------
    def _get_lease(self):
        return self.client.get("Lease", self.namespace, self.name)

    def _lease_expired(self, lease) -> bool:
        spec = lease.spec
        if not spec.holder_identity:
            return True
        if spec.renew_time is None:
            return True
        try:
            renewed_at = _parse(spec.renew_time)
        except Exception:
            return True
        duration = spec.lease_duration_seconds or self.config.lease_duration
        return _now() > renewed_at + duration

    def _try_acquire_or_renew(self) -> bool:
        now = _fmt(_now())
        try:
            lease = self._get_lease()
        except NotFoundError:
            lease = corev1.Lease(
                metadata=ObjectMeta(name=self.name, namespace=self.namespace),
                spec=corev1.LeaseSpec(
                    holder_identity=self.identity,
                    lease_duration_seconds=int(self.config.lease_duration),
                    acquire_time=now,
                    renew_time=now,
                    lease_transitions=0,
                ),
            )
            try:
                self.client.create(lease)
                return True
            except Exception:
                return False

        holder = lease.spec.holder_identity
        if holder and holder != self._observed_leader:
            self._observed_leader = holder
            if self.on_new_leader and holder != self.identity:
                self.on_new_leader(holder)

        if holder != self.identity and not self._lease_expired(lease):
            return False

        if holder != self.identity:
            lease.spec.lease_transitions += 1
            lease.spec.acquire_time = now
        lease.spec.holder_identity = self.identity
        lease.spec.renew_time = now
        lease.spec.lease_duration_seconds = int(self.config.lease_duration)
        try:
            self.client.update(lease)
            return True
        except (ConflictError, NotFoundError):
            return False

    def _release(self):
        try:
            lease = self._get_lease()
            if lease.spec.holder_identity == self.identity:
                lease.spec.holder_identity = None
                lease.spec.renew_time = None
                self.client.update(lease)
        except Exception:
            logger.debug("lease release failed", exc_info=True)

    # -- main loop ----------------------------------------------------------
    def run(self, stop: threading.Event):
        """Blocks: acquire → lead (renewing) → on failure/stop return.
        Mirrors leaderelection.RunOrDie's acquire/renew structure."""
        try:
            self._acquire(stop)
            if stop.is_set():
                return
            logger.info("%s became leader", self.identity)
            self.is_leader.set()
            stop_leading = threading.Event()
            lead_thread = None
            if self.on_started_leading:
                lead_thread = threading.Thread(
                    target=self.on_started_leading,
                    args=(stop_leading,),
                    name=f"leader-{self.name}",
                    daemon=True,
                )
                lead_thread.start()
            self._renew_loop(stop)
            stop_leading.set()
            self.is_leader.clear()
            if self.config.release_on_cancel and stop.is_set():
                self._release()
            if self.on_stopped_leading:
                self.on_stopped_leading()
            if lead_thread is not None:
                lead_thread.join(timeout=5.0)
        finally:
            self.is_leader.clear()

    def _acquire(self, stop: threading.Event):
        while not stop.is_set():
            if self._try_acquire_or_renew():
                self._renew_time = _now()
                return
            stop.wait(self.config.retry_period)

    def _renew_loop(self, stop: threading.Event):
        """Renew every retry_period; give up when renew_deadline passes
        without a successful renewal (lost lease)."""
        while not stop.is_set():
            if self._try_acquire_or_renew():
                self._renew_time = _now()
            elif _now() - self._renew_time > self.config.renew_deadline:
                logger.warning("%s lost the lease", self.identity)
                return
            stop.wait(self.config.retry_period)
