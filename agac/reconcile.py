"""Generic reconcile engine.

Mirrors the reference's ``pkg/reconcile/reconcile.go:18-91`` exactly:

- ``Result(requeue, requeue_after)`` drives requeue behavior;
- ``process_next_work_item(queue, key_to_obj, process_delete,
  process_create_or_update)`` pumps one item:
    * ``key_to_obj`` raising NotFound ⇒ the object is gone ⇒ ``process_delete``;
    * other ``key_to_obj`` errors ⇒ logged, item retried rate-limited;
    * processor raising ``NoRetryError`` ⇒ Forget (no retry);
    * other processor errors ⇒ AddRateLimited;
    * ``Result.requeue_after > 0`` ⇒ Forget + AddAfter;
    * ``Result.requeue`` ⇒ AddRateLimited;
    * success ⇒ Forget.

Every error is swallowed after logging (utilruntime.HandleError) — a worker
loop never dies on a failed reconcile.
"""

from __future__ import annotations

import logging
import time
from dataclasses import dataclass

from . import metrics
from .errors import is_no_retry
from .kube.store import is_not_found

logger = logging.getLogger(__name__)


@dataclass
class Result:
    requeue: bool = False
    requeue_after: float = 0.0  # seconds; 0 = no delayed requeue


def process_next_work_item(queue, key_to_obj, process_delete, process_create_or_update) -> bool:
    """Processes one queue item; returns False only when the queue shut down
    (reference ``reconcile.go:26-43``)."""
    key, shutdown = queue.get()
    if shutdown:
        return False
    try:
        _reconcile_handler(key, queue, key_to_obj, process_delete, process_create_or_update)
    except Exception:
        logger.exception("error processing %r", key)
    finally:
        queue.done(key)
    return True


def _reconcile_handler(key, queue, key_to_obj, process_delete, process_create_or_update):
    if not isinstance(key, str):
        queue.forget(key)
        logger.error("expected string in workqueue but got %r", key)
        return

    start = time.monotonic()
    outcome = "success"
    try:
        res = Result()
        err = None
        try:
            obj = key_to_obj(key)
        except Exception as lookup_err:
            if is_not_found(lookup_err):
                res, err = _run(process_delete, key)
            else:
                logger.error("unable to retrieve %r from store: %s", key, lookup_err)
                outcome = "error"
                return
        else:
            # The reference DeepCopies here (keyToService returns the shared
            # informer-cache pointer, reconcile.go:52).  Our listers already
            # return a private deep copy per Get (informer.cache_get), so a
            # second copy would be pure overhead — the process func owns obj.
            res, err = _run(process_create_or_update, obj)

        if err is not None:
            if is_no_retry(err):
                outcome = "no_retry_error"
                logger.error("error syncing %r: %s", key, err)
            else:
                outcome = "error"
                queue.add_rate_limited(key)
                logger.error("error syncing %r, and requeued: %s", key, err)
        elif res.requeue_after > 0:
            outcome = "requeue_after"
            queue.forget(key)
            queue.add_after(key, res.requeue_after)
            logger.info("Successfully synced %r, but requeued after %ss", key, res.requeue_after)
        elif res.requeue:
            outcome = "requeue"
            queue.add_rate_limited(key)
            logger.info("Successfully synced %r, but requeued", key)
        else:
            queue.forget(key)
            logger.debug("Successfully synced %r", key)
    finally:
        elapsed = time.monotonic() - start
        metrics.observe_reconcile(queue.name, outcome, elapsed)
        logger.debug("Finished syncing %r (%.6fs)", key, elapsed)


def _run(fn, arg):
    """Runs a processor, returning (Result, error) Go-style."""
    try:
        res = fn(arg)
        return (res if res is not None else Result()), None
    except Exception as e:  # noqa: BLE001 — the engine decides retry policy
        return Result(), e
