#!/usr/bin/env python3
"""Soak: run the full controller stack under continuous churn for
--minutes, tracking convergence failures, reconcile throughput and RSS
(memory-leak detector).  Exit 0 only if every interval converged and RSS
stayed bounded."""

from __future__ import annotations

import argparse
import json
import os
import resource
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import bench  # noqa: E402


def rss_mb() -> float:
    return resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1024.0


def current_rss_mb() -> float:
    with open("/proc/self/status") as f:
        for line in f:
            if line.startswith("VmRSS"):
                return int(line.split()[1]) / 1024.0
    return rss_mb()


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--minutes", type=float, default=10.0)
    parser.add_argument("--objects", type=int, default=64)
    parser.add_argument("--scenario", choices=["ga", "full"], default="full")
    parser.add_argument("--api", choices=["memory", "http"], default="memory")
    parser.add_argument("--report-every", type=float, default=30.0)
    args = parser.parse_args()

    client, backend, services, bindings, stop = bench.build_stack(
        args.objects, workers=1, scenario=args.scenario, api=args.api
    )
    samples = []
    try:
        deadline = time.monotonic() + args.minutes * 60.0
        next_report = time.monotonic() + args.report_every
        step = 0
        interval_steps = 0
        interval_start = time.monotonic()
        base_rss = None
        latencies = []
        while time.monotonic() < deadline:
            t0 = time.monotonic()
            bench.run_step(client, backend, services, step, timeout=120.0,
                           bindings=bindings)
            latencies.append(time.monotonic() - t0)
            step += 1
            interval_steps += 1
            now = time.monotonic()
            if now >= next_report:
                rss = current_rss_mb()
                if base_rss is None:
                    base_rss = rss
                rate = interval_steps * (args.objects + len(bindings)) / (
                    now - interval_start
                )
                samples.append(
                    {"t": round(now - (deadline - args.minutes * 60.0), 1),
                     "steps": step, "obj_per_s": round(rate, 1),
                     "rss_mb": round(rss, 1)}
                )
                print(json.dumps(samples[-1]), flush=True)
                next_report = now + args.report_every
                interval_steps = 0
                interval_start = now
        final_rss = current_rss_mb()
        growth = final_rss - (base_rss or final_rss)
        latencies.sort()

        def pct(p):
            return round(latencies[min(len(latencies) - 1, int(p * len(latencies)))] * 1000, 2)

        verdict = {
            "latency_ms": {"p50": pct(0.50), "p90": pct(0.90),
                           "p99": pct(0.99), "max": round(latencies[-1] * 1000, 2)},
            "api": args.api,
            "soak_minutes": args.minutes,
            "total_steps": step,
            "objects": args.objects + len(bindings),
            "scenario": args.scenario,
            "rss_start_mb": round(base_rss or 0, 1),
            "rss_end_mb": round(final_rss, 1),
            "rss_growth_mb": round(growth, 1),
            "throughput_samples": [s["obj_per_s"] for s in samples],
            "ok": growth < 200.0,
        }
        print(json.dumps(verdict), flush=True)
        sys.exit(0 if verdict["ok"] else 1)
    finally:
        stop.set()


if __name__ == "__main__":
    main()
