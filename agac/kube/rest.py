"""REST KubeClient: the same ``KubeClient`` interface as the in-memory
client, over HTTP against an ``agac.kube.httpapi`` API server (or anything
speaking the same wire scheme).  This is the out-of-process analogue of
client-go's typed clientset: informers, controllers, leader election and
the manager all run unchanged on top of it.
"""

from __future__ import annotations

import json
import logging
import queue
import threading
from typing import Optional

import requests

from ..apis.meta import from_dict, to_dict
from .client import KubeClient, class_for_kind
from .store import (
    AlreadyExistsError,
    APIError,
    ConflictError,
    GoneError,
    NotFoundError,
    WatchEvent,
)

from .admission import AdmissionDeniedError

logger = logging.getLogger(__name__)

_REASON_TO_ERROR = {
    "NotFound": NotFoundError,
    "AlreadyExists": AlreadyExistsError,
    "Conflict": ConflictError,
    "Gone": GoneError,
    "Forbidden": AdmissionDeniedError,
}


def _raise_for(response):
    if response.status_code < 400:
        return
    try:
        body = response.json()
        reason = body.get("reason", "InternalError")
        message = body.get("message", response.text)
    except ValueError:
        reason, message = "InternalError", response.text
    cls = _REASON_TO_ERROR.get(reason)
    if cls is not None:
        raise cls(message)
    raise APIError(message, response.status_code)


class _RestWatch:
    """Iterates ndjson watch events from a streaming HTTP response; the
    same interface as the in-memory store's watch handle."""

    def __init__(self, response):
        self._response = response
        self._queue: "queue.Queue[Optional[WatchEvent]]" = queue.Queue()
        self._stopped = False
        self.closed = False
        self._thread = threading.Thread(target=self._pump, daemon=True)
        self._thread.start()

    def _pump(self):
        try:
            for line in self._response.iter_lines():
                if self._stopped:
                    break
                if not line:
                    continue  # heartbeat
                payload = json.loads(line)
                if payload["type"] == "BOOKMARK":
                    self._queue.put(
                        WatchEvent("BOOKMARK", None, payload["resourceVersion"])
                    )
                    continue
                cls = class_for_kind(payload["object"]["kind"])
                self._queue.put(
                    WatchEvent(
                        payload["type"],
                        from_dict(cls, payload["object"]),
                        payload["resourceVersion"],
                    )
                )
        except Exception:
            if not self._stopped:
                logger.debug("watch stream ended", exc_info=True)
        finally:
            self._queue.put(None)

    def get(self, timeout: Optional[float] = None) -> Optional[WatchEvent]:
        if self.closed:
            return None
        try:
            item = self._queue.get(timeout=timeout)
        except queue.Empty:
            return None
        if item is None:
            self.closed = True
        return item

    def stop(self):
        self._stopped = True
        try:
            self._response.close()
        except Exception:
            pass

    def __iter__(self):
        return self

    def __next__(self):
        item = self._queue.get()
        if item is None:
            raise StopIteration
        return item


class RestKubeClient(KubeClient):
    def __init__(self, base_url: str, timeout: float = 10.0,
                 watch_timeout_seconds: float = 300.0):
        self.base_url = base_url.rstrip("/")
        self.timeout = timeout
        self.watch_timeout_seconds = watch_timeout_seconds
        self.session = requests.Session()

    # -- raw verbs ---------------------------------------------------------
    def _obj_body(self, obj) -> dict:
        d = to_dict(obj)
        d["kind"] = type(obj).kind
        d["apiVersion"] = type(obj).api_version
        return d

    def create(self, obj):
        r = self.session.post(
            f"{self.base_url}/apis/{type(obj).kind}",
            json=self._obj_body(obj),
            timeout=self.timeout,
        )
        _raise_for(r)
        return from_dict(type(obj), r.json())

    def get(self, kind: str, namespace: str, name: str):
        r = self.session.get(
            f"{self.base_url}/apis/{kind}/{namespace}/{name}", timeout=self.timeout
        )
        _raise_for(r)
        return from_dict(class_for_kind(kind), r.json())

    def list(self, kind: str, namespace: Optional[str] = None,
             page_size: int = 500):
        """Chunked list (limit/continue loop); a 410 on continuation
        restarts the list once from scratch."""
        cls = class_for_kind(kind)
        for attempt in (0, 1):
            items, rv, cont = [], 0, None
            try:
                while True:
                    params = {"limit": str(page_size)}
                    if namespace:
                        params["namespace"] = namespace
                    if cont:
                        params["continue"] = cont
                    r = self.session.get(
                        f"{self.base_url}/apis/{kind}", params=params,
                        timeout=self.timeout,
                    )
                    _raise_for(r)
                    body = r.json()
                    items.extend(from_dict(cls, item) for item in body["items"])
                    rv = body["resourceVersion"]
                    cont = body.get("continue")
                    if not cont:
                        return items, rv
            except GoneError:
                if attempt == 1:
                    raise
                logger.info("list %s continue token expired; restarting list", kind)
        return items, rv  # pragma: no cover

    def update(self, obj):
        meta = obj.metadata
        r = self.session.put(
            f"{self.base_url}/apis/{type(obj).kind}/{meta.namespace}/{meta.name}",
            json=self._obj_body(obj),
            timeout=self.timeout,
        )
        _raise_for(r)
        return from_dict(type(obj), r.json())

    def update_status(self, obj):
        meta = obj.metadata
        r = self.session.put(
            f"{self.base_url}/apis/{type(obj).kind}/{meta.namespace}/{meta.name}/status",
            json=self._obj_body(obj),
            timeout=self.timeout,
        )
        _raise_for(r)
        return from_dict(type(obj), r.json())

    def delete(self, kind: str, namespace: str, name: str):
        r = self.session.delete(
            f"{self.base_url}/apis/{kind}/{namespace}/{name}", timeout=self.timeout
        )
        _raise_for(r)
        return None

    def patch(self, kind: str, namespace: str, name: str, patch: dict,
              subresource=None):
        from .patch import MERGE_PATCH_CONTENT_TYPE

        url = f"{self.base_url}/apis/{kind}/{namespace}/{name}"
        if subresource:
            url += f"/{subresource}"
        r = self.session.patch(
            url, json=patch,
            headers={"Content-Type": MERGE_PATCH_CONTENT_TYPE},
            timeout=self.timeout,
        )
        _raise_for(r)
        from .client import class_for_kind

        return from_dict(class_for_kind(kind), r.json())

    def watch(self, kind: str, namespace: Optional[str] = None, resource_version=None):
        params = {"allowWatchBookmarks": "true"}
        if self.watch_timeout_seconds:
            params["timeoutSeconds"] = str(int(self.watch_timeout_seconds))
        if namespace:
            params["namespace"] = namespace
        if resource_version is not None:
            params["resourceVersion"] = resource_version
        # no read timeout on the streaming socket: heartbeats flow every 5s
        r = requests.get(
            f"{self.base_url}/watch/{kind}",
            params=params,
            stream=True,
            timeout=(self.timeout, 30.0),
        )
        _raise_for(r)
        return _RestWatch(r)
