"""Cluster-grade wire-client behavior: paginated LIST
(limit/continue loop), watch BOOKMARK handling, and 429/Retry-After
backoff — the client-go reflector features the reference gets for free via
its SharedInformerFactory (pkg/manager/manager.go:52-53).  The hermetic
apiserver (agac.kube.httpapi) serves the same wire features so these paths
are exercised end to end over real HTTP.
"""

import json
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.kube.httpapi import APIServer
from agac.kube.k8s import K8sKubeClient
from agac.kube.kubeconfig import RestConfig
from agac.kube.rest import RestKubeClient
from agac.kube.store import APIStore, GoneError


def mk_service(name, ns="default"):
    return corev1.Service(metadata=ObjectMeta(name=name, namespace=ns))


# ---------------------------------------------------------------------------
# Store-level pagination semantics
# ---------------------------------------------------------------------------
class TestStoreListPage:
    def test_pages_partition_and_share_rv(self):
        store = APIStore()
        for i in range(7):
            store.create(mk_service(f"svc-{i:03d}"))
        seen, rvs, token = [], [], None
        while True:
            items, rv, token = store.list_page("Service", limit=3, continue_token=token)
            seen.extend(o.metadata.name for o in items)
            rvs.append(rv)
            if token is None:
                break
        assert seen == [f"svc-{i:03d}" for i in range(7)]
        assert len(set(rvs)) == 1  # one snapshot rv across all pages
        assert len(rvs) == 3  # 3+3+1

    def test_no_limit_is_single_page(self):
        store = APIStore()
        store.create(mk_service("a"))
        items, rv, token = store.list_page("Service")
        assert len(items) == 1 and token is None

    def test_malformed_token_is_gone(self):
        store = APIStore()
        with pytest.raises(GoneError):
            store.list_page("Service", limit=2, continue_token="not-base64!!")

    def test_expired_token_is_gone(self):
        from agac.kube import store as store_mod

        store = APIStore()
        for i in range(4):
            store.create(mk_service(f"svc-{i}"))
        _, _, token = store.list_page("Service", limit=2)
        # churn enough events to evict the token's rv window from the log
        for _ in range(store_mod._EVENT_LOG_SIZE + 10):
            store.create(mk_service("churn"))
            store.delete("Service", "default", "churn")
        with pytest.raises(GoneError):
            store.list_page("Service", limit=2, continue_token=token)


# ---------------------------------------------------------------------------
# Over HTTP: both wire schemes
# ---------------------------------------------------------------------------
@pytest.fixture
def server():
    store = APIStore()
    srv = APIServer(store, watch_idle_seconds=0.1)
    srv.start()
    yield srv
    srv.shutdown()


def k8s_client(srv, **kw):
    return K8sKubeClient(RestConfig(host=srv.url), **kw)


class TestHTTPListPagination:
    N = 1201

    def seed(self, store, n=None):
        for i in range(n or self.N):
            store.create(mk_service(f"svc-{i:05d}"))

    def test_k8s_client_paginates(self, server):
        self.seed(server.store)
        client = k8s_client(server, page_size=500)
        items, rv = client.list("Service")
        assert len(items) == self.N
        assert sorted(o.metadata.name for o in items) == [
            f"svc-{i:05d}" for i in range(self.N)
        ]
        assert rv > 0

    def test_rest_client_paginates(self, server):
        self.seed(server.store)
        client = RestKubeClient(server.url)
        items, rv = client.list("Service", page_size=500)
        assert len(items) == self.N

    def test_wire_actually_chunks(self, server):
        """The server really sends continue tokens (not one giant page)."""
        self.seed(server.store, 7)
        import requests

        r = requests.get(
            f"{server.url}/api/v1/services", params={"limit": "3"}, timeout=5
        )
        body = r.json()
        assert len(body["items"]) == 3
        cont = body["metadata"]["continue"]
        assert cont
        r2 = requests.get(
            f"{server.url}/api/v1/services",
            params={"limit": "3", "continue": cont},
            timeout=5,
        )
        body2 = r2.json()
        assert len(body2["items"]) == 3
        assert body2["metadata"]["resourceVersion"] == body["metadata"]["resourceVersion"]
        assert body["items"][-1]["metadata"]["name"] < body2["items"][0]["metadata"]["name"]

    def test_expired_continue_is_410_and_client_restarts(self, server):
        """410 on a continuation restarts the whole list (reflector
        contract) instead of surfacing an error or a truncated result."""
        from agac.kube import store as store_mod

        self.seed(server.store, 6)
        import requests

        r = requests.get(
            f"{server.url}/api/v1/services", params={"limit": "2"}, timeout=5
        )
        cont = r.json()["metadata"]["continue"]
        for _ in range(store_mod._EVENT_LOG_SIZE + 10):
            server.store.create(mk_service("churn"))
            server.store.delete("Service", "default", "churn")
        r = requests.get(
            f"{server.url}/api/v1/services",
            params={"limit": "2", "continue": cont},
            timeout=5,
        )
        assert r.status_code == 410
        assert r.json()["reason"] == "Expired"
        # the client hides all of this: full list still converges
        client = k8s_client(server, page_size=2)
        items, _ = client.list("Service")
        assert len(items) == 6


class TestWatchBookmarks:
    def test_k8s_client_receives_bookmarks(self, server):
        server.store.create(mk_service("a"))
        client = k8s_client(server)
        _, rv = client.list("Service")
        watch = client.watch("Service", resource_version=rv)
        try:
            deadline = time.monotonic() + 5
            got = None
            while time.monotonic() < deadline:
                ev = watch.get(timeout=0.5)
                if ev is not None and ev.type == "BOOKMARK":
                    got = ev
                    break
            assert got is not None, "no BOOKMARK within 5s of idle"
            assert got.obj is None
            assert got.resource_version >= rv - 1
        finally:
            watch.stop()

    def test_informer_advances_rv_past_bookmark(self, server):
        """After an idle bookmark, a re-watch resumes at the bookmarked rv:
        churn that happened and was fully replayed is NOT redelivered."""
        client = k8s_client(server)
        from agac.kube.informer import Informer

        stop = threading.Event()
        informer = Informer(client, "Service", resync_period=0)
        events = []
        informer.add_event_handler(
            on_add=lambda o: events.append(("add", o.metadata.name)),
            on_update=lambda o, n: events.append(("upd", n.metadata.name)),
            on_delete=lambda o: events.append(("del", o.metadata.name)),
        )
        informer.run(stop)
        try:
            server.store.create(mk_service("x"))
            deadline = time.monotonic() + 5
            while ("add", "x") not in events:
                assert time.monotonic() < deadline
                time.sleep(0.02)
            # idle long enough for several bookmarks, then check the
            # informer's internal resume point has moved to the store head
            time.sleep(0.5)
            assert informer.has_synced()
            assert events.count(("add", "x")) == 1  # bookmark churned nothing
        finally:
            stop.set()
            informer.stop()

    def test_informer_full_sync_through_pages(self, server):
        """VERDICT r1 item 2 acceptance: a 10k-object store syncs through
        500-item pages into a complete informer cache."""
        for i in range(10_000):
            server.store.create(mk_service(f"svc-{i:05d}"))
        client = k8s_client(server, page_size=500)
        from agac.kube.informer import Informer, wait_for_cache_sync

        stop = threading.Event()
        informer = Informer(client, "Service", resync_period=0)
        informer.run(stop)
        try:
            assert wait_for_cache_sync(stop, informer, timeout=60.0)
            assert len(informer.cache_list()) == 10_000
            # live events still flow after the paged sync
            server.store.create(mk_service("tail"))
            deadline = time.monotonic() + 5
            while informer.cache_get("default", "tail") is None:
                assert time.monotonic() < deadline
                time.sleep(0.02)
        finally:
            stop.set()
            informer.stop()


# ---------------------------------------------------------------------------
# 429 / Retry-After
# ---------------------------------------------------------------------------
class _Throttling429Handler(BaseHTTPRequestHandler):
    """Returns 429 with Retry-After for the first N requests, then an empty
    ServiceList."""

    fail_first = 2
    seen = None  # list shared via class closure

    def log_message(self, fmt, *args):
        pass

    def do_GET(self):
        self.seen.append(time.monotonic())
        if len(self.seen) <= self.fail_first:
            body = json.dumps(
                {
                    "kind": "Status",
                    "status": "Failure",
                    "reason": "TooManyRequests",
                    "code": 429,
                }
            ).encode()
            self.send_response(429)
            self.send_header("Retry-After", "0.05")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)
            return
        body = json.dumps(
            {
                "kind": "ServiceList",
                "apiVersion": "v1",
                "metadata": {"resourceVersion": "7"},
                "items": [],
            }
        ).encode()
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)


class TestRetryAfterBackoff:
    def test_list_retries_through_429(self):
        seen = []
        handler = type("H", (_Throttling429Handler,), {"seen": seen})
        httpd = ThreadingHTTPServer(("127.0.0.1", 0), handler)
        threading.Thread(target=httpd.serve_forever, daemon=True).start()
        try:
            url = f"http://127.0.0.1:{httpd.server_address[1]}"
            client = K8sKubeClient(RestConfig(host=url), max_retries=5)
            items, rv = client.list("Service")
            assert items == [] and rv == 7
            assert len(seen) == 3  # two 429s + the success
        finally:
            httpd.shutdown()
            httpd.server_close()

    def test_retries_exhaust_and_raise(self):
        seen = []
        handler = type(
            "H", (_Throttling429Handler,), {"seen": seen, "fail_first": 99}
        )
        httpd = ThreadingHTTPServer(("127.0.0.1", 0), handler)
        threading.Thread(target=httpd.serve_forever, daemon=True).start()
        try:
            url = f"http://127.0.0.1:{httpd.server_address[1]}"
            client = K8sKubeClient(RestConfig(host=url), max_retries=2)
            from agac.kube.store import APIError

            with pytest.raises(APIError):
                client.list("Service")
            assert len(seen) == 3  # initial + 2 retries
        finally:
            httpd.shutdown()
            httpd.server_close()


class TestWatchReconnect:
    def test_informer_resumes_after_server_closes_watch(self, server):
        """Real apiservers close watch streams periodically; the informer
        must reconnect from its last delivered rv WITHOUT relisting and
        without missing events that happened while disconnected."""
        list_calls = []
        client = k8s_client(server)
        orig_list = client.list

        def counting_list(kind, namespace=None):
            list_calls.append(kind)
            return orig_list(kind, namespace)

        client.list = counting_list
        from agac.kube.informer import Informer, wait_for_cache_sync

        stop = threading.Event()
        informer = Informer(client, "Service", resync_period=0)
        stop_evt_names = []
        informer.add_event_handler(
            on_add=lambda o: stop_evt_names.append(o.metadata.name)
        )
        informer.run(stop)
        try:
            assert wait_for_cache_sync(stop, informer)
            server.store.create(mk_service("before"))
            deadline = time.monotonic() + 5
            while "before" not in stop_evt_names:
                assert time.monotonic() < deadline
                time.sleep(0.02)

            # server closes every active watch subscription
            for w in list(server.store._watches):
                w.stop()
            # events continue while the client is disconnected
            server.store.create(mk_service("during"))

            deadline = time.monotonic() + 10
            while informer.cache_get("default", "during") is None:
                assert time.monotonic() < deadline
                time.sleep(0.02)
            # the replayed event arrived through a RE-WATCH, not a relist
            assert list_calls == ["Service"], list_calls
            assert "during" in stop_evt_names  # delivered exactly as an add
            assert stop_evt_names.count("during") == 1
        finally:
            stop.set()
            informer.stop()


class TestWatchTimeout:
    def test_server_honors_timeout_seconds_and_informer_survives(self, server):
        """?timeoutSeconds= ends the stream like a real apiserver; the
        informer reconnects across expiries without losing events."""
        client = k8s_client(server, watch_timeout_seconds=1)
        from agac.kube.informer import Informer, wait_for_cache_sync

        stop = threading.Event()
        informer = Informer(client, "Service", resync_period=0)
        adds = []
        informer.add_event_handler(on_add=lambda o: adds.append(o.metadata.name))
        informer.run(stop)
        try:
            assert wait_for_cache_sync(stop, informer)
            # run across several 1s watch expiries, creating during each
            for i in range(3):
                server.store.create(mk_service(f"tw-{i}"))
                deadline = time.monotonic() + 10
                while f"tw-{i}" not in adds:
                    assert time.monotonic() < deadline
                    time.sleep(0.02)
                time.sleep(1.2)  # guarantee at least one expiry between
            assert sorted(adds) == ["tw-0", "tw-1", "tw-2"]
        finally:
            stop.set()
            informer.stop()


class TestPagedSyncUnderChurn:
    def test_informer_converges_while_store_churns(self, server):
        """A multi-page initial sync races live creates/deletes; whatever
        mix of continue-tokens, 410 restarts and watch replays happens, the
        informer must end exactly consistent with the store."""
        for i in range(900):
            server.store.create(mk_service(f"base-{i:04d}"))
        churn_stop = threading.Event()

        def churner():
            i = 0
            while not churn_stop.is_set():
                server.store.create(mk_service(f"churn-{i:04d}"))
                if i >= 5:
                    server.store.delete("Service", "default", f"churn-{i-5:04d}")
                i += 1
                time.sleep(0.002)

        t = threading.Thread(target=churner, daemon=True)
        client = k8s_client(server, page_size=50)  # 18+ pages
        from agac.kube.informer import Informer, wait_for_cache_sync

        stop = threading.Event()
        informer = Informer(client, "Service", resync_period=0)
        t.start()
        informer.run(stop)
        try:
            assert wait_for_cache_sync(stop, informer, timeout=60.0)
            churn_stop.set()
            time.sleep(0.3)  # let the tail of the churn drain through watch
            store_names = {
                o.metadata.name for o in server.store.list("Service")[0]
            }
            deadline = time.monotonic() + 10
            while True:
                cache_names = {
                    o.metadata.name for o in informer.cache_list()
                }
                if cache_names == store_names:
                    break
                assert time.monotonic() < deadline, (
                    f"cache != store: missing={sorted(store_names - cache_names)[:5]} "
                    f"extra={sorted(cache_names - store_names)[:5]}"
                )
                time.sleep(0.05)
        finally:
            churn_stop.set()
            stop.set()
            informer.stop()
