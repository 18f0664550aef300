"""Leader election: single-leader invariant, failover with 2 replicas,
release-on-cancel (BASELINE.json config 5)."""

import threading
import time

from agac.kube.client import InMemoryKubeClient
from agac.kube.leaderelection import LeaderElectionConfig, LeaderElector


def fast_config(**overrides):
    defaults = dict(
        lease_duration=0.3, renew_deadline=0.2, retry_period=0.05,
        release_on_cancel=True,
    )
    defaults.update(overrides)
    return LeaderElectionConfig(**defaults)


def make_elector(client, identity, log=None):
    started = threading.Event()

    def on_start(stop_leading):
        started.set()
        stop_leading.wait()

    elector = LeaderElector(
        client,
        name="agac-leader",
        namespace="kube-system",
        identity=identity,
        config=fast_config(),
        on_started_leading=on_start,
        on_new_leader=(lambda ident: log.append(ident)) if log is not None else None,
    )
    return elector, started


def run_in_thread(elector, stop):
    t = threading.Thread(target=elector.run, args=(stop,), daemon=True)
    t.start()
    return t


def test_single_candidate_becomes_leader():
    client = InMemoryKubeClient()
    elector, started = make_elector(client, "a")
    stop = threading.Event()
    t = run_in_thread(elector, stop)
    assert started.wait(timeout=5.0)
    lease = client.get("Lease", "kube-system", "agac-leader")
    assert lease.spec.holder_identity == "a"
    stop.set()
    t.join(timeout=5.0)


def test_only_one_of_two_leads():
    client = InMemoryKubeClient()
    ea, sa = make_elector(client, "a")
    eb, sb = make_elector(client, "b")
    stop = threading.Event()
    ta = run_in_thread(ea, stop)
    tb = run_in_thread(eb, stop)
    time.sleep(0.5)
    assert ea.is_leader.is_set() != eb.is_leader.is_set()
    stop.set()
    ta.join(timeout=5.0)
    tb.join(timeout=5.0)


def test_failover_to_standby():
    client = InMemoryKubeClient()
    observed = []
    ea, sa = make_elector(client, "a")
    eb, sb = make_elector(client, "b", log=observed)
    stop_a = threading.Event()
    stop_b = threading.Event()
    ta = run_in_thread(ea, stop_a)
    assert sa.wait(timeout=5.0)
    tb = run_in_thread(eb, stop_b)
    time.sleep(0.2)
    assert not eb.is_leader.is_set()

    # leader goes away (release_on_cancel frees the lease immediately)
    stop_a.set()
    ta.join(timeout=5.0)
    assert sb.wait(timeout=5.0), "standby should take over"
    lease = client.get("Lease", "kube-system", "agac-leader")
    assert lease.spec.holder_identity == "b"
    assert "a" in observed  # b observed a's leadership first
    stop_b.set()
    tb.join(timeout=5.0)


def test_failover_without_release_waits_for_expiry():
    client = InMemoryKubeClient()
    config = fast_config(release_on_cancel=False)
    ea = LeaderElector(client, "agac-leader", "kube-system", identity="a", config=config)
    stop_a = threading.Event()
    ta = run_in_thread(ea, stop_a)
    deadline = time.monotonic() + 5
    while not ea.is_leader.is_set() and time.monotonic() < deadline:
        time.sleep(0.01)
    assert ea.is_leader.is_set()
    stop_a.set()  # dies without releasing
    ta.join(timeout=5.0)

    eb, sb = make_elector(client, "b")
    stop_b = threading.Event()
    tb = run_in_thread(eb, stop_b)
    # must wait out the 0.3s lease, then acquire
    assert sb.wait(timeout=5.0)
    assert client.get("Lease", "kube-system", "agac-leader").spec.lease_transitions >= 1
    stop_b.set()
    tb.join(timeout=5.0)


def test_lost_lease_stops_leading():
    client = InMemoryKubeClient()
    ea, sa = make_elector(client, "a")
    stop = threading.Event()
    t = run_in_thread(ea, stop)
    assert sa.wait(timeout=5.0)
    # usurp the lease out-of-band (simulates apiserver-side takeover); give
    # the intruder a long duration so "a" cannot reacquire via expiry while
    # its renew loop is still timing out
    lease = client.get("Lease", "kube-system", "agac-leader")
    lease.spec.holder_identity = "intruder"
    lease.spec.lease_duration_seconds = 3600
    from agac.kube.leaderelection import _fmt, _now
    lease.spec.renew_time = _fmt(_now())
    client.update(lease)
    deadline = time.monotonic() + 5
    while ea.is_leader.is_set() and time.monotonic() < deadline:
        time.sleep(0.02)
    assert not ea.is_leader.is_set()
    stop.set()
    t.join(timeout=5.0)
