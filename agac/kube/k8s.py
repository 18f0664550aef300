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
                obj = from_dict(self._cls, obj_dict)
                rv_raw = (obj_dict.get("metadata") or {}).get("resourceVersion", "0")
                try:
                    rv = int(rv_raw)
                except (TypeError, ValueError):
                    rv = 0
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
    def __init__(self, config: RestConfig, timeout: float = 10.0):
        self.config = config
        self.base_url = config.host
        self.timeout = timeout
        self.session = requests.Session()
        if config.token:
            self.session.headers["Authorization"] = f"Bearer {config.token}"
        if config.cert:
            self.session.cert = config.cert
        self.session.verify = config.verify

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
        r = self.session.post(
            self._url(kind, obj.metadata.namespace or None),
            json=self._obj_body(obj),
            timeout=self.timeout,
        )
        _raise_for(r)
        return from_dict(type(obj), r.json())

    def get(self, kind: str, namespace: str, name: str):
        r = self.session.get(
            self._url(kind, namespace or None, name), timeout=self.timeout
        )
        _raise_for(r)
        return from_dict(class_for_kind(kind), r.json())

    def list(self, kind: str, namespace: Optional[str] = None):
        r = self.session.get(self._url(kind, namespace), timeout=self.timeout)
        _raise_for(r)
        body = r.json()
        cls = class_for_kind(kind)
        items = [from_dict(cls, item) for item in body.get("items", [])]
        rv_raw = (body.get("metadata") or {}).get("resourceVersion", "0")
        try:
            rv = int(rv_raw)
        except (TypeError, ValueError):
            rv = 0
        return items, rv

    def update(self, obj):
        kind = type(obj).kind
        r = self.session.put(
            self._url(kind, obj.metadata.namespace or None, obj.metadata.name),
            json=self._obj_body(obj),
            timeout=self.timeout,
        )
        _raise_for(r)
        return from_dict(type(obj), r.json())

    def update_status(self, obj):
        kind = type(obj).kind
        r = self.session.put(
            self._url(kind, obj.metadata.namespace or None, obj.metadata.name, "status"),
            json=self._obj_body(obj),
            timeout=self.timeout,
        )
        _raise_for(r)
        return from_dict(type(obj), r.json())

    def delete(self, kind: str, namespace: str, name: str):
        r = self.session.delete(
            self._url(kind, namespace or None, name), timeout=self.timeout
        )
        _raise_for(r)
        return None

    def patch(self, kind: str, namespace: str, name: str, patch: dict,
              subresource=None):
        from .patch import MERGE_PATCH_CONTENT_TYPE

        r = self.session.patch(
            self._url(kind, namespace or None, name, subresource),
            json=patch,
            headers={"Content-Type": MERGE_PATCH_CONTENT_TYPE},
            timeout=self.timeout,
        )
        _raise_for(r)
        return from_dict(class_for_kind(kind), r.json())

    def watch(self, kind: str, namespace: Optional[str] = None, resource_version=None):
        params = {"watch": "true"}
        if resource_version is not None:
            params["resourceVersion"] = str(resource_version)
        r = self.session.get(
            self._url(kind, namespace),
            params=params,
            stream=True,
            timeout=(self.timeout, 30.0),
        )
        _raise_for(r)
        return _K8sWatch(r, kind)
