"""API machinery: in-memory API server, typed clients, informers, workqueues,
event recording and leader election.

Replaces client-go + the k8s code-generator output of the reference
(``pkg/client/**``, ~1300 generated LoC) with hand-written, fully typed
Python equivalents that share one watch/cache/queue implementation.
"""

from .store import APIStore, ConflictError, NotFoundError, AlreadyExistsError, is_not_found
from .client import KubeClient, InMemoryKubeClient
from .workqueue import RateLimitingQueue, ItemExponentialFailureRateLimiter
from .informer import SharedInformerFactory, Informer, Lister, wait_for_cache_sync
from .leaderelection import LeaderElector, LeaderElectionConfig
from .admission import AdmissionDeniedError, http_admission, local_admission
from .patch import json_merge_patch
from .apply import apply_yaml

__all__ = [
    "APIStore",
    "ConflictError",
    "NotFoundError",
    "AlreadyExistsError",
    "is_not_found",
    "KubeClient",
    "InMemoryKubeClient",
    "RateLimitingQueue",
    "ItemExponentialFailureRateLimiter",
    "SharedInformerFactory",
    "Informer",
    "Lister",
    "wait_for_cache_sync",
    "LeaderElector",
    "LeaderElectionConfig",
    "AdmissionDeniedError",
    "http_admission",
    "local_admission",
    "json_merge_patch",
    "apply_yaml",
]
