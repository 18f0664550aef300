"""GPU-box tests.

This repo is a k8s controller (tier mismatch per BASELINE.json) — nothing
computes on the GPU by design.  These tests verify the framework runs
unchanged on the GPU box and that the box itself is live.
"""

import pytest

pytestmark = pytest.mark.gpu


def test_gpu_box_is_live():
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU on this machine (gpu-marked tier runs on the GPU box)")
    x = torch.randn(1024, device="cuda:0")
    assert float(x.abs().sum().item()) > 0


def test_smoke_reconcile_on_gpu_box():
    import __graft_entry__

    __graft_entry__.smoke()


def test_bench_entry_one_step():
    """bench.py's stack builds and converges one step on the box."""
    import bench

    client, backend, services, bindings, stop = bench.build_stack(objects=4, workers=2)
    try:
        bench.run_step(client, backend, services, 0, timeout=60.0)
    finally:
        stop.set()
