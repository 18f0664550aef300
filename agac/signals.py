"""Signal handling: SIGINT/SIGTERM → stop event; second signal → exit(1)
(reference ``pkg/signals/signals.go:16-30``)."""

from __future__ import annotations

import signal
import sys
import threading

_registered = False


def setup_signal_handler() -> threading.Event:
    """Returns a stop Event set on the first SIGINT/SIGTERM; a second signal
    exits immediately with status 1.  May only be called once per process
    (like the reference's panic-on-second-call onlyOneSignalHandler)."""
    global _registered
    if _registered:
        raise RuntimeError("setup_signal_handler called twice")
    _registered = True

    stop = threading.Event()

    def handler(signum, frame):
        if stop.is_set():
            sys.exit(1)
        stop.set()

    signal.signal(signal.SIGINT, handler)
    signal.signal(signal.SIGTERM, handler)
    return stop
