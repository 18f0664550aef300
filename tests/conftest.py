import os
import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))

_LOCKCHECK = os.environ.get("AGAC_LOCKCHECK") == "1"

if _LOCKCHECK:
    # install BEFORE any agac module creates a lock (race-detection tier;
    # see agac/lockcheck.py and `make test-race`)
    from agac import lockcheck

    lockcheck.install()


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require a GPU box (run with -m gpu)"
    )


def pytest_sessionfinish(session, exitstatus):
    if _LOCKCHECK:
        from agac import lockcheck

        # raises LockOrderViolation (fails the run) on any cycle observed
        # across every test in the session
        lockcheck.check()
        sys.stderr.write(
            f"\n[lockcheck] {len(lockcheck.edges())} lock-order edges observed, "
            "no cycles\n"
        )
