"""JSON merge patch (RFC 7386).

The real kube-apiserver accepts ``application/merge-patch+json``; agac
supports it end-to-end (store, HTTP server, REST/k8s clients, apply) so
partial updates never clobber fields the sender doesn't model — the safe
verb for real-cluster writes.
"""

from __future__ import annotations

MERGE_PATCH_CONTENT_TYPE = "application/merge-patch+json"


def json_merge_patch(target, patch):
    """RFC 7386: dicts merge recursively, None deletes, everything else
    replaces.  Returns a new structure; inputs are not mutated."""
    if not isinstance(patch, dict):
        return patch
    if not isinstance(target, dict):
        target = {}
    result = dict(target)
    for key, value in patch.items():
        if value is None:
            result.pop(key, None)
        else:
            result[key] = json_merge_patch(result.get(key), value)
    return result
