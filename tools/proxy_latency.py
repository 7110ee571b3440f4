"""Proxy-path overhead vs the reference's claimed ~1-2 ms: p50/p99 HTTP
round trip of /agent/{id}/chat against the echo engine (WAL append +
dispatch + ack INCLUDED — the reference's number excluded persistence).
CPU-only; run: python tools/proxy_latency.py
"""
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, "tests")
from test_crash_integration import Server, _free_port  # noqa: E402

import httpx  # noqa: E402

srv = Server("/tmp/proxy-lat", _free_port())
try:
    srv.start()
    st, resp = srv.call("POST", "/agents", {"name": "lat", "model": "echo"})
    aid = resp["data"]["id"]
    srv.call("POST", f"/agents/{aid}/start")
    with httpx.Client(base_url=srv.base, timeout=10.0) as c:
        for _ in range(20):
            c.post(f"/agent/{aid}/chat", json={"message": "warm"})
        lat = []
        for i in range(300):
            t0 = time.perf_counter()
            r = c.post(f"/agent/{aid}/chat", json={"message": f"m{i}"})
            lat.append((time.perf_counter() - t0) * 1000)
            assert r.status_code == 200
    lat.sort()
    print(f"proxy+WAL round trip over HTTP (echo engine, n=300): "
          f"p50 {statistics.median(lat):.2f} ms  "
          f"p99 {lat[int(0.99 * len(lat))]:.2f} ms  min {lat[0]:.2f} ms")
finally:
    srv.terminate()
