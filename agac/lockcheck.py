"""Lock-order checking — the Python analog of ``go test -race`` for this
codebase's deadlock class (SURVEY §5 lists race detection as an aux
subsystem; the reference's CI runs without ``-race`` and r1 shipped only a
stress-loop proxy).

CPython's GIL already serializes data access per bytecode, so classic
torn-read data races are not this framework's failure mode — *lock-order
inversions* between its long-held locks are (store._lock, informer
._cache_lock, FakeAWSBackend.lock, controller hint locks).  A cycle in the
acquisition-order graph means two threads can deadlock under the right
interleaving even if no test run ever hangs.

Usage: ``install()`` wraps ``threading.Lock``/``RLock`` so every
acquisition records (held-locks → new-lock) edges in a global order graph;
``check()`` raises ``LockOrderViolation`` listing any cycle with both
acquisition stacks.  ``make test-race`` runs the concurrency suites with
this installed (AGAC_LOCKCHECK=1 in tests/conftest.py).

Re-entrant acquisition of the same RLock is ignored (legal), and edges are
deduplicated by (acquire-site, acquire-site) pair so the graph stays
small.  Overhead is one dict lookup + occasional stack capture per
acquisition — debug-tier, not for production.
"""

from __future__ import annotations

import threading
import traceback
from typing import Dict, List, Set, Tuple

_real_lock = threading.Lock
_real_rlock = threading.RLock

_state = threading.local()
_graph_lock = _real_lock()
# edge: (lock_site_a -> lock_site_b) meaning b was acquired while a held
_edges: Dict[Tuple[str, str], Tuple[str, str]] = {}
_installed = False


class LockOrderViolation(AssertionError):
    pass


def _site() -> str:
    """Identify a lock by WHERE it was created (file:line), which names the
    subsystem — individual lock instances of the same site class together
    (every FakeAWSBackend lock is 'backend.py:NNN').  Returns "" for locks
    created outside this project (stdlib queue/logging/condition internals)
    so those stay completely untracked — zero noise, zero overhead."""
    for frame in reversed(traceback.extract_stack(limit=12)):
        if frame.filename.endswith("agac/lockcheck.py"):
            continue
        if (
            "/agac/" in frame.filename
            or "/tests/" in frame.filename
            or frame.filename.endswith(("bench.py", "soak.py"))
        ):
            return f"{frame.filename.rsplit('/', 1)[-1]}:{frame.lineno}"
        return ""
    return ""


class _CheckedLock:
    __slots__ = ("_inner", "_site", "_reentrant", "_owner", "_count")

    def __init__(self, inner, site: str, reentrant: bool):
        self._inner = inner
        self._site = site
        self._reentrant = reentrant
        self._owner = None
        self._count = 0

    # -- bookkeeping -------------------------------------------------------
    def _held_stack(self) -> List[str]:
        stack = getattr(_state, "held", None)
        if stack is None:
            stack = _state.held = []
        return stack

    def _note_acquire(self):
        me = threading.get_ident()
        if self._reentrant and self._owner == me:
            self._count += 1
            return  # re-entrant: no new edge
        held = self._held_stack()
        if held:
            top = held[-1]
            if top != self._site:
                edge = (top, self._site)
                if edge not in _edges:
                    where = "".join(traceback.format_stack(limit=6)[:-2])
                    with _graph_lock:
                        _edges.setdefault(edge, (where, ""))
        held.append(self._site)
        self._owner = me
        self._count = 1

    def _note_release(self):
        if self._reentrant and self._count > 1:
            self._count -= 1
            return
        held = self._held_stack()
        if held and held[-1] == self._site:
            held.pop()
        elif self._site in held:  # out-of-order release (legal, rare)
            held.remove(self._site)
        self._owner = None
        self._count = 0

    # -- lock API ----------------------------------------------------------
    def acquire(self, *a, **kw):
        got = self._inner.acquire(*a, **kw)
        if got:
            self._note_acquire()
        return got

    def release(self):
        self._note_release()
        self._inner.release()

    def __enter__(self):
        self.acquire()
        return self

    def __exit__(self, *exc):
        self.release()
        return False

    def locked(self):
        return self._inner.locked()

    # threading.Condition integration: delegate the private protocol so a
    # Condition built on a checked lock keeps correct ownership semantics
    # (RLock has these; for a plain Lock emulate Condition's fallbacks)
    def _is_owned(self):
        inner_is_owned = getattr(self._inner, "_is_owned", None)
        if inner_is_owned is not None:
            return inner_is_owned()
        if self._inner.acquire(False):
            self._inner.release()
            return False
        return True

    def _release_save(self):
        self._note_release()
        inner = getattr(self._inner, "_release_save", None)
        if inner is not None:
            return inner()
        self._inner.release()
        return None

    def _acquire_restore(self, state):
        inner = getattr(self._inner, "_acquire_restore", None)
        if inner is not None:
            inner(state)
        else:
            self._inner.acquire()
        self._note_acquire()


def _make_lock():
    site = _site()
    if not site:
        return _real_lock()
    return _CheckedLock(_real_lock(), site, reentrant=False)


def _make_rlock():
    site = _site()
    if not site:
        return _real_rlock()
    return _CheckedLock(_real_rlock(), site, reentrant=True)


def install():
    """Wrap threading.Lock/RLock constructors. Idempotent."""
    global _installed
    if _installed:
        return
    threading.Lock = _make_lock
    threading.RLock = _make_rlock
    _installed = True


def uninstall():
    global _installed
    threading.Lock = _real_lock
    threading.RLock = _real_rlock
    _installed = False


def reset():
    with _graph_lock:
        _edges.clear()


def edges() -> Set[Tuple[str, str]]:
    with _graph_lock:
        return set(_edges)


def check():
    """Raise LockOrderViolation if the acquisition-order graph has a cycle
    (two lock sites each acquired while the other is held → deadlockable)."""
    with _graph_lock:
        graph: Dict[str, Set[str]] = {}
        for a, b in _edges:
            graph.setdefault(a, set()).add(b)

    # iterative DFS cycle detection
    WHITE, GRAY, BLACK = 0, 1, 2
    color = {node: WHITE for node in graph}
    for start in graph:
        if color.get(start, WHITE) != WHITE:
            continue
        stack = [(start, iter(graph.get(start, ())))]
        color[start] = GRAY
        path = [start]
        while stack:
            node, it = stack[-1]
            advanced = False
            for nxt in it:
                c = color.get(nxt, WHITE)
                if c == GRAY:
                    cycle = path[path.index(nxt):] + [nxt]
                    detail = []
                    for a, b in zip(cycle, cycle[1:]):
                        where = _edges.get((a, b), ("", ""))[0]
                        detail.append(f"  {a} -> {b}\n{where}")
                    raise LockOrderViolation(
                        "lock-order cycle (potential deadlock):\n"
                        + "\n".join(detail)
                    )
                if c == WHITE:
                    color[nxt] = GRAY
                    path.append(nxt)
                    stack.append((nxt, iter(graph.get(nxt, ()))))
                    advanced = True
                    break
            if not advanced:
                color[node] = BLACK
                path.pop()
                stack.pop()
