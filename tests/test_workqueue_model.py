"""Bounded model-checking of the workqueue's client-go semantics.

The dirty/processing coalescing contract (client-go util/workqueue) is the
subtlest piece of machinery in the framework: an Add during processing must
re-deliver exactly once after Done; duplicate Adds coalesce; an item is
never handed to two workers at once.  Instead of sampling interleavings
with threads, enumerate EVERY call order up to a bounded length
single-threaded (the queue's behavior depends only on call order) and
compare against an executable reference model transcribed from the
client-go contract.
"""

import itertools

import pytest

from agac.kube.workqueue import RateLimitingQueue


class ModelQueue:
    """Executable spec of client-go's Type (queue.go): queue is ordered,
    dirty is the coalescing set, processing defers redelivery."""

    def __init__(self):
        self.queue = []
        self.dirty = set()
        self.processing = set()

    def add(self, item):
        if item in self.dirty:
            return
        self.dirty.add(item)
        if item in self.processing:
            return
        self.queue.append(item)

    def get(self):
        if not self.queue:
            return None
        item = self.queue.pop(0)
        self.processing.add(item)
        self.dirty.discard(item)
        return item

    def done(self, item):
        self.processing.discard(item)
        if item in self.dirty:
            self.queue.append(item)


def drain(q, got_log):
    """Pop everything deliverable, tracking what a worker would see."""
    while True:
        item, shutdown = q.get(timeout=0)
        if item is None:
            return
        got_log.append(item)
        q.done(item)


OPS = ["add_a", "add_b", "get", "done"]


def run_schedule(schedule):
    """Apply one call order to both implementations; return their visible
    traces (sequence of items handed to the worker) or a mismatch."""
    real = RateLimitingQueue(name="")
    model = ModelQueue()
    real_in_hand = []
    model_in_hand = []
    trace_real, trace_model = [], []
    for op in schedule:
        if op == "add_a":
            real.add("a")
            model.add("a")
        elif op == "add_b":
            real.add("b")
            model.add("b")
        elif op == "get":
            r, shutdown = real.get(timeout=0)
            m = model.get()
            assert r == m, (schedule, "get mismatch", r, m)
            if r is not None:
                # contract: never hand out an item already being processed
                assert r not in real_in_hand, (schedule, "double-processing", r)
                real_in_hand.append(r)
                model_in_hand.append(m)
                trace_real.append(r)
                trace_model.append(m)
        elif op == "done":
            if real_in_hand:
                r = real_in_hand.pop(0)
                m = model_in_hand.pop(0)
                real.done(r)
                model.done(m)
    # drain both to quiescence: every coalesced re-add must deliver
    while True:
        r, _ = real.get(timeout=0)
        m = model.get()
        assert r == m, (schedule, "drain mismatch", r, m)
        if r is None:
            break
        trace_real.append(r)
        trace_model.append(m)
        real.done(r)
        model.done(m)
        # a done may have re-queued from dirty; loop continues
    assert trace_real == trace_model
    return trace_real


def test_all_schedules_up_to_length_8_match_the_model():
    n = 0
    for length in range(1, 9):
        for schedule in itertools.product(OPS, repeat=length):
            run_schedule(schedule)
            n += 1
    assert n == sum(4**k for k in range(1, 9))  # 87,380 schedules


def test_redelivery_after_done_exactly_once():
    """The canonical coalescing case spelled out: add while processing
    redelivers exactly once after done, regardless of how many adds."""
    q = RateLimitingQueue(name="")
    q.add("x")
    item, _ = q.get(timeout=0)
    assert item == "x"
    for _ in range(5):
        q.add("x")  # all coalesce into one pending redelivery
    assert q.get(timeout=0) == (None, False)  # not while processing
    q.done("x")
    assert q.get(timeout=0)[0] == "x"  # exactly one redelivery
    q.done("x")
    assert q.get(timeout=0) == (None, False)
