#!/usr/bin/env python3
"""Micro-benchmarks mirroring the reference's in-tree benches
(reference: pkg/kmsg/writer/kmsg_test.go:402-412 line building,
pkg/eventstore/database_benchmark_test.go insert/get, plus our own
metrics-scrape and RAS-catalog match). CPU-only; prints one JSON line.

Usage: python scripts/microbench.py [iters]
"""

import json
import sys
import time

sys.path.insert(0, ".")


def timed(fn, iters):
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    return (time.perf_counter() - t0) / iters * 1e6  # µs/op


def main() -> None:
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 5000

    out = {}

    from gpud_amd.pkg.kmsg.writer import build_line

    out["kmsg_build_line_us"] = timed(
        lambda: build_line(2, "amdgpu 0000:0a:00.0: ring gfx_0.0.0 timeout"),
        iters,
    )

    from gpud_amd.pkg.kmsg.parser import parse_line

    line = "3,4619,1234567,-;amdgpu 0000:0a:00.0: amdgpu: ring gfx_0.0.0 timeout"
    out["kmsg_parse_line_us"] = timed(lambda: parse_line(line, 0.0), iters)

    from gpud_amd.pkg.ras_catalog import match

    out["ras_catalog_match_hit_us"] = timed(lambda: match(line), iters)
    out["ras_catalog_match_miss_us"] = timed(
        lambda: match("systemd[1]: Started Session 42 of user root."), iters
    )

    from gpud_amd.apiv1.types import Event, utcnow
    from gpud_amd.pkg.eventstore import Store
    from gpud_amd.pkg.sqlite_util import open_memory_pair

    rw, ro = open_memory_pair()
    store = Store(rw, ro)
    bucket = store.bucket("microbench", disable_purge=True)
    ev = Event(time=utcnow(), name="e", type="Info", message="m")
    n_ins = min(iters, 2000)
    out["eventstore_insert_us"] = timed(lambda: bucket.insert(ev), n_ins)
    since = utcnow().replace(year=2000)
    out["eventstore_get_all_us"] = timed(lambda: bucket.get(since), 50)
    store.close()

    import prometheus_client

    from gpud_amd.components.metrics_util import ComponentGauges

    reg = prometheus_client.CollectorRegistry()
    g = ComponentGauges("microbench", reg)
    out["gauge_set_us"] = timed(
        lambda: g.set("microbench_value", "d", 1.0, uuid="u0"), iters
    )

    from gpud_amd.pkg.metrics.scraper import Scraper

    scraper = Scraper(reg)
    out["metrics_scrape_us"] = timed(lambda: scraper.scrape(), 200)

    print(json.dumps({k: round(v, 3) for k, v in out.items()}))


if __name__ == "__main__":
    main()
