"""Prometheus metrics.

The reference has no metrics endpoint (SURVEY.md §5 flags this as a gap);
this build closes it: reconcile outcomes/latency, workqueue depth and AWS
API call counters, exposed via ``start_metrics_server``.  All metric calls
are no-ops if prometheus_client is unavailable.
"""

from __future__ import annotations

try:
    from prometheus_client import Counter, Gauge, Histogram, start_http_server

    _AVAILABLE = True
except Exception:  # pragma: no cover - prometheus_client is installed here
    _AVAILABLE = False

if _AVAILABLE:
    RECONCILE_TOTAL = Counter(
        "agac_reconcile_total",
        "Reconcile attempts by queue and outcome",
        ["queue", "outcome"],
    )
    RECONCILE_DURATION = Histogram(
        "agac_reconcile_duration_seconds",
        "Reconcile latency by queue",
        ["queue"],
        buckets=(0.0005, 0.001, 0.005, 0.01, 0.05, 0.1, 0.5, 1.0, 5.0, 30.0),
    )
    AWS_API_CALLS = Counter(
        "agac_aws_api_calls_total",
        "AWS API operations issued",
        ["service", "operation"],
    )
    WORKQUEUE_DEPTH = Gauge(
        "agac_workqueue_depth",
        "Items currently queued (not yet picked up by a worker)",
        ["queue"],
    )
    HINT_TOTAL = Counter(
        "agac_hint_total",
        "ARN-hint verification outcomes (hit = O(1) discovery; "
        "miss/error = fall back to the full accelerator scan)",
        ["controller", "outcome"],
    )
    WORKQUEUE_QUEUE_DURATION = Histogram(
        "agac_workqueue_queue_duration_seconds",
        "Time items wait in a workqueue before a worker picks them up "
        "(client-go workqueue_queue_duration_seconds)",
        ["queue"],
        buckets=(1e-4, 1e-3, 1e-2, 0.1, 1.0, 10.0, 60.0),
    )
    WORKQUEUE_WORK_DURATION = Histogram(
        "agac_workqueue_work_duration_seconds",
        "Time a worker spends processing an item "
        "(client-go workqueue_work_duration_seconds)",
        ["queue"],
        buckets=(1e-4, 1e-3, 1e-2, 0.1, 1.0, 10.0, 60.0),
    )
    WORKQUEUE_RETRIES = Counter(
        "agac_workqueue_retries_total",
        "Rate-limited re-adds (client-go workqueue_retries_total)",
        ["queue"],
    )
    WEBHOOK_REVIEWS = Counter(
        "agac_webhook_reviews_total",
        "AdmissionReview verdicts served by the webhook process",
        ["operation", "verdict"],
    )


# prometheus_client's .labels() re-validates and re-hashes on every call;
# the label sets here are tiny and hot (every reconcile/AWS op), so memoize
# the child metric objects
_label_cache: dict = {}


def _child(metric, *labels):
    key = (id(metric), labels)
    child = _label_cache.get(key)
    if child is None:
        child = _label_cache[key] = metric.labels(*labels)
    return child


def observe_reconcile(queue_name: str, outcome: str, seconds: float):
    if _AVAILABLE:
        q = queue_name or "unknown"
        _child(RECONCILE_TOTAL, q, outcome).inc()
        _child(RECONCILE_DURATION, q).observe(seconds)


def observe_hint(controller: str, outcome: str):
    """outcome: hit | stale | error (stale = tags no longer match)."""
    if _AVAILABLE:
        _child(HINT_TOTAL, controller, outcome).inc()


def observe_aws_call(service: str, operation: str):
    if _AVAILABLE:
        _child(AWS_API_CALLS, service, operation).inc()


def set_queue_depth(queue_name: str, depth: int):
    if _AVAILABLE and queue_name:
        _child(WORKQUEUE_DEPTH, queue_name).set(depth)


def observe_queue_latency(queue_name: str, seconds: float):
    if _AVAILABLE and queue_name:
        _child(WORKQUEUE_QUEUE_DURATION, queue_name).observe(seconds)


def observe_work_duration(queue_name: str, seconds: float):
    if _AVAILABLE and queue_name:
        _child(WORKQUEUE_WORK_DURATION, queue_name).observe(seconds)


def count_queue_retry(queue_name: str):
    if _AVAILABLE and queue_name:
        _child(WORKQUEUE_RETRIES, queue_name).inc()


def observe_webhook_review(operation: str, verdict: str):
    if _AVAILABLE:
        _child(WEBHOOK_REVIEWS, operation or "UNKNOWN", verdict).inc()


def start_metrics_server(port: int):
    if _AVAILABLE:
        start_http_server(port)
