"""Controller health endpoints (/healthz liveness, /readyz caches-synced)
— absent in the reference (only its webhook has /healthz), required for
deployment probes; wired into the chart behind controller.healthPort."""

import threading
import time

import requests

from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.health import HealthServer
from agac.kube.client import InMemoryKubeClient
from agac.manager import ControllerConfig, Manager


def test_healthz_always_ok_readyz_tracks_manager():
    state = {"m": None}
    server = HealthServer(
        0, ready_fn=lambda: state["m"] is not None and state["m"].is_ready(),
        host="127.0.0.1",
    )
    server.start()
    base = f"http://127.0.0.1:{server.port}"
    try:
        assert requests.get(f"{base}/healthz", timeout=5).status_code == 200
        # standby shape: alive but not ready
        assert requests.get(f"{base}/readyz", timeout=5).status_code == 503

        stop = threading.Event()
        manager = Manager()
        manager.run(InMemoryKubeClient(), ControllerConfig(),
                    FakeCloudFactory(FakeAWSBackend()), stop,
                    resync_period=300.0, block=False)
        state["m"] = manager
        try:
            assert manager.wait_until_ready()
            deadline = time.monotonic() + 5
            while requests.get(f"{base}/readyz", timeout=5).status_code != 200:
                assert time.monotonic() < deadline
                time.sleep(0.05)
        finally:
            stop.set()
        assert requests.get(f"{base}/unknown", timeout=5).status_code == 404
    finally:
        server.shutdown()
