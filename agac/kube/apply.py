"""kubectl-apply-lite: create-or-update API objects from YAML documents.

The counterpart of the reference e2e's server-side-apply helper
(``e2e/pkg/util/manifests.go:35-70``): parse multi-document YAML, map
apiVersion/kind through the wire registry, and create or update each
object through any ``KubeClient``.  Unknown kinds (Deployments etc. in the
samples) are skipped with a notice rather than failing, like applying to a
cluster without those controllers installed would surface later.
"""

from __future__ import annotations

import logging
from typing import List, Tuple

import yaml

from ..apis.meta import from_dict
from . import k8swire
from .client import class_for_kind
from .store import NotFoundError

logger = logging.getLogger(__name__)


def parse_documents(text: str) -> List[dict]:
    return [doc for doc in yaml.safe_load_all(text) if isinstance(doc, dict)]


def apply_document(client, doc: dict) -> Tuple[str, str]:
    """Apply one decoded object; returns (action, 'kind/ns/name').
    action ∈ {created, configured, unchanged, skipped}."""
    kind = doc.get("kind", "")
    api_version = doc.get("apiVersion", "")
    gvr = k8swire.BY_KIND.get(kind)
    if gvr is None or gvr.api_version != api_version:
        return "skipped", f"{api_version}/{kind}"
    cls = class_for_kind(kind)
    obj = from_dict(cls, doc)
    if gvr.namespaced:
        namespace = obj.metadata.namespace or "default"
    else:
        namespace = ""  # cluster-scoped
    obj.metadata.namespace = namespace
    ident = f"{kind.lower()}/{namespace}/{obj.metadata.name}"
    try:
        existing = client.get(kind, namespace, obj.metadata.name)
    except NotFoundError:
        client.create(obj)
        return "created", ident
    from ..apis.meta import to_dict

    desired, live = to_dict(obj), to_dict(existing)
    unchanged = desired.get("spec") == live.get("spec") and all(
        desired.get("metadata", {}).get(k) == live.get("metadata", {}).get(k)
        for k in ("annotations", "labels")
    )
    if unchanged:
        return "unchanged", ident
    # PATCH the manifest's fields instead of full PUT replacement, so
    # server-populated fields (status, uid, creationTimestamp, fields other
    # managers own) survive — the merge-patch analogue of the reference
    # e2e's server-side apply (e2e/pkg/util/manifests.go:35-70); the k8s
    # wire client stamps ?fieldManager= on the request.
    patch = {
        "metadata": {
            "labels": desired.get("metadata", {}).get("labels"),
            "annotations": desired.get("metadata", {}).get("annotations"),
        },
        "spec": desired.get("spec"),
    }
    patch["metadata"] = {k: v for k, v in patch["metadata"].items() if v is not None}
    if not patch["metadata"]:
        del patch["metadata"]
    if patch.get("spec") is None:
        patch.pop("spec", None)
    client.patch(kind, namespace, obj.metadata.name, patch)
    return "configured", ident


def apply_yaml(client, text: str) -> List[Tuple[str, str]]:
    return [apply_document(client, doc) for doc in parse_documents(text)]
