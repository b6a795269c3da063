#!/usr/bin/env python3
"""
Cluster load test against a LIVE gordo server (the analog of the
reference's Locust harness, benchmarks/load_test/load_test.py:63-104 —
locust is not installed here, so this is a self-contained threaded
driver with the same request mix).

Usage:
    python benchmarks/load_test/load_test.py \
        --host http://localhost:5555 --project my-project \
        --users 16 --duration 60
Each simulated user loops: pick a random served model, POST a
100-row X/y payload to /anomaly/prediction (falling back to
/prediction for non-anomaly models), wait ~1 s (reference wait_time).
"""
import argparse
import json
import random
import statistics
import threading
import time

import numpy as np


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", required=True)
    ap.add_argument("--project", required=True)
    ap.add_argument("--users", type=int, default=16)
    ap.add_argument("--duration", type=float, default=60.0)
    ap.add_argument("--wait", type=float, default=1.0)
    ap.add_argument("--rows", type=int, default=100)
    args = ap.parse_args()

    import requests

    base = f"{args.host}/gordo/v0/{args.project}"
    models = requests.get(f"{base}/models").json()["models"]
    if not models:
        raise SystemExit("server lists no models")

    # discover tags per model once
    tags = {}
    for name in models:
        meta = requests.get(f"{base}/{name}/metadata").json()["metadata"]
        tags[name] = [
            t["name"] if isinstance(t, dict) else t
            for t in meta["dataset"]["tag_list"]
        ]

    from gordo_amd.server.utils import dataframe_to_dict
    import pandas as pd

    stats_lock = threading.Lock()
    latencies, errors = [], [0]
    deadline = time.time() + args.duration

    def user():
        s = requests.Session()
        while time.time() < deadline:
            name = random.choice(models)
            X = pd.DataFrame(
                np.random.random((args.rows, len(tags[name]))),
                columns=tags[name],
            )
            payload = {"X": dataframe_to_dict(X), "y": dataframe_to_dict(X)}
            t0 = time.perf_counter()
            r = s.post(f"{base}/{name}/anomaly/prediction", json=payload)
            if r.status_code == 422:
                r = s.post(f"{base}/{name}/prediction", json=payload)
            dt = time.perf_counter() - t0
            with stats_lock:
                if r.status_code == 200:
                    latencies.append(dt)
                else:
                    errors[0] += 1
            time.sleep(args.wait)

    threads = [threading.Thread(target=user) for _ in range(args.users)]
    t_start = time.time()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    elapsed = time.time() - t_start
    print(json.dumps({
        "requests_ok": len(latencies),
        "errors": errors[0],
        "requests_per_sec": len(latencies) / elapsed,
        "predictions_per_sec": len(latencies) * args.rows / elapsed,
        "mean_latency_ms": statistics.mean(latencies) * 1000 if latencies else None,
        "p95_latency_ms": (
            statistics.quantiles(latencies, n=20)[-1] * 1000
            if len(latencies) >= 20 else None
        ),
        "users": args.users,
        "duration_sec": elapsed,
    }))


if __name__ == "__main__":
    main()
