"""KubeClient speaking the real Kubernetes REST wire format.

The production client (client-go's role in the reference): resource paths
from ``k8swire`` (core/v1, networking.k8s.io/v1, coordination.k8s.io/v1 and
the operator.h3poteto.dev/v1alpha1 CRD), bearer-token / client-cert auth
from a ``RestConfig``, ``?watch=true`` streaming with k8s event framing,
and ``Status``-object error translation.  Runs unchanged against a real
kube-apiserver or against ``agac.kube.httpapi`` (which serves the same
paths), which is how it is tested hermetically.
"""

from __future__ import annotations

import json
import logging
import queue
import threading
from typing import Optional

import requests

from ..apis.meta import from_dict, to_dict
from . import k8swire
from .client import KubeClient, class_for_kind
from .kubeconfig import RestConfig
from .store import WatchEvent

logger = logging.getLogger(__name__)


class _K8sWatch:
    """k8s-framed watch stream ({"type", "object"} lines)."""

    def __init__(self, response, kind: str):
        self._response = response
        self._cls = class_for_kind(kind)
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
                    continue
                payload = json.loads(line)
                obj_dict = payload["object"]
                rv_raw = (obj_dict.get("metadata") or {}).get("resourceVersion", "0")
                try:
                    rv = int(rv_raw)
                except (TypeError, ValueError):
                    rv = 0
                if payload["type"] == "BOOKMARK":
                    # metadata-only frame: advances the consumer's resume rv,
                    # never touches caches/handlers (client-go reflector
                    # bookmark handling)
                    self._queue.put(WatchEvent("BOOKMARK", None, rv))
                    continue
                obj = from_dict(self._cls, obj_dict)
                self._queue.put(WatchEvent(payload["type"], obj, rv))
        except Exception:
            if not self._stopped:
                logger.debug("k8s watch stream ended", exc_info=True)
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


def _raise_for(response):
    if response.status_code < 400:
        return
    try:
        status = response.json()
    except ValueError:
        status = {}
    raise k8swire.error_for_status(status, response.status_code)


class K8sKubeClient(KubeClient):
    """``page_size`` chunks every LIST (``?limit=&continue=`` loop, default
    500 — the client-go reflector's WatchListPageSize default); watches are
    opened with ``allowWatchBookmarks=true``.  429/503 responses with
    Retry-After are retried with backoff (apiserver priority&fairness
    shedding) up to ``max_retries`` times."""

    def __init__(self, config: RestConfig, timeout: float = 10.0,
                 page_size: int = 500, max_retries: int = 5,
                 field_manager: str = "aws-global-accelerator-controller",
                 watch_timeout_seconds: float = 300.0):
        self.config = config
        self.base_url = config.host
        self.timeout = timeout
        self.page_size = page_size
        self.max_retries = max_retries
        # declared owner of fields we write (apiserver server-side field
        # tracking; sent as ?fieldManager= on every mutating verb, the
        # client-go behavior the reference inherits)
        self.field_manager = field_manager
        # server-side watch deadline (client-go sends 5-10 min): the server
        # cleanly ends the stream and the informer re-watches from its
        # resume rv — keeps long-lived watches from going half-open
        self.watch_timeout_seconds = watch_timeout_seconds
        self.session = requests.Session()
        if config.token:
            self.session.headers["Authorization"] = f"Bearer {config.token}"
        if config.cert:
            self.session.cert = config.cert
        self.session.verify = config.verify

    def _request(self, method: str, url: str, **kwargs):
        """Issue a request, honoring Retry-After on 429/503 (retries never
        apply to streaming watches — those reconnect at the informer)."""
        import time as _time

        attempt = 0
        # per-request verify: requests gives REQUESTS_CA_BUNDLE/CURL_CA_BUNDLE
        # env precedence over session.verify, which would silently override
        # the kubeconfig's certificate-authority — the explicit kwarg wins
        kwargs.setdefault("verify", self.session.verify)
        while True:
            r = getattr(self.session, method)(url, **kwargs)
            if r.status_code not in (429, 503) or attempt >= self.max_retries:
                return r
            retry_after = r.headers.get("Retry-After")
            try:
                delay = min(float(retry_after), 30.0) if retry_after else 0.0
            except ValueError:
                delay = 0.0
            if not delay:
                delay = min(0.5 * 2**attempt, 8.0)
            logger.info(
                "apiserver %d for %s %s; retrying in %.1fs",
                r.status_code, method.upper(), url, delay,
            )
            r.close()
            _time.sleep(delay)
            attempt += 1

    # -- helpers -----------------------------------------------------------
    def _url(self, kind: str, namespace=None, name=None, subresource=None) -> str:
        gvr = k8swire.gvr_for_kind(kind)
        return self.base_url + gvr.path(namespace, name, subresource)

    def _obj_body(self, obj) -> dict:
        gvr = k8swire.gvr_for_kind(type(obj).kind)
        d = to_dict(obj)
        d["kind"] = gvr.kind
        d["apiVersion"] = gvr.api_version
        return d

    # -- verbs -------------------------------------------------------------
    def create(self, obj):
        kind = type(obj).kind
        r = self._request(
            "post",
            self._url(kind, obj.metadata.namespace or None),
            params={"fieldManager": self.field_manager},
            json=self._obj_body(obj),
            timeout=self.timeout,
        )
        _raise_for(r)
        return from_dict(type(obj), r.json())

    def get(self, kind: str, namespace: str, name: str):
        r = self._request(
            "get", self._url(kind, namespace or None, name), timeout=self.timeout
        )
        _raise_for(r)
        return from_dict(class_for_kind(kind), r.json())

    def list(self, kind: str, namespace: Optional[str] = None):
        """Paginated LIST: follows ``metadata.continue`` in ``page_size``
        chunks; every page of one logical list is served at the first
        page's resourceVersion.  A 410 (continue token expired) restarts
        the whole list from scratch — the client-go reflector's
        pagination contract (reference gets this via the informers at
        pkg/manager/manager.go:52-53)."""
        from .store import GoneError

        cls = class_for_kind(kind)
        for attempt in (0, 1):
            items = []
            params = {"limit": str(self.page_size)} if self.page_size else {}
            rv = 0
            try:
                while True:
                    r = self._request(
                        "get", self._url(kind, namespace), params=params,
                        timeout=self.timeout,
                    )
                    _raise_for(r)
                    body = r.json()
                    items.extend(from_dict(cls, item) for item in body.get("items", []))
                    meta = body.get("metadata") or {}
                    rv_raw = meta.get("resourceVersion", "0")
                    try:
                        rv = int(rv_raw)
                    except (TypeError, ValueError):
                        rv = 0
                    cont = meta.get("continue")
                    if not cont:
                        return items, rv
                    params = {"limit": str(self.page_size), "continue": cont}
            except GoneError:
                if attempt == 1:
                    raise
                logger.info("list %s continue token expired; restarting list", kind)
        return items, rv  # pragma: no cover - loop always returns/raises

    def update(self, obj):
        kind = type(obj).kind
        r = self._request(
            "put",
            self._url(kind, obj.metadata.namespace or None, obj.metadata.name),
            params={"fieldManager": self.field_manager},
            json=self._obj_body(obj),
            timeout=self.timeout,
        )
        _raise_for(r)
        return from_dict(type(obj), r.json())

    def update_status(self, obj):
        kind = type(obj).kind
        r = self._request(
            "put",
            self._url(kind, obj.metadata.namespace or None, obj.metadata.name, "status"),
            params={"fieldManager": self.field_manager},
            json=self._obj_body(obj),
            timeout=self.timeout,
        )
        _raise_for(r)
        return from_dict(type(obj), r.json())

    def delete(self, kind: str, namespace: str, name: str):
        r = self._request(
            "delete", self._url(kind, namespace or None, name), timeout=self.timeout
        )
        _raise_for(r)
        return None

    def patch(self, kind: str, namespace: str, name: str, patch: dict,
              subresource=None):
        from .patch import MERGE_PATCH_CONTENT_TYPE

        r = self._request(
            "patch",
            self._url(kind, namespace or None, name, subresource),
            params={"fieldManager": self.field_manager},
            json=patch,
            headers={"Content-Type": MERGE_PATCH_CONTENT_TYPE},
            timeout=self.timeout,
        )
        _raise_for(r)
        return from_dict(class_for_kind(kind), r.json())

    def watch(self, kind: str, namespace: Optional[str] = None, resource_version=None):
        params = {"watch": "true", "allowWatchBookmarks": "true"}
        if self.watch_timeout_seconds:
            params["timeoutSeconds"] = str(int(self.watch_timeout_seconds))
        if resource_version is not None:
            params["resourceVersion"] = str(resource_version)
        r = self.session.get(
            self._url(kind, namespace),
            params=params,
            stream=True,
            timeout=(self.timeout, 30.0),
            verify=self.session.verify,
        )
        _raise_for(r)
        return _K8sWatch(r, kind)
