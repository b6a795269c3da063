"""
ML-server latency benchmarks, replicating the reference harness shape
(reference benchmarks/test_ml_server.py:21-42: X = (100, n_sensors)
random JSON payloads POSTed to /prediction and /anomaly/prediction
through the in-process Flask test client, 100 rounds x 1 iteration).

Excluded from CI like the reference (run explicitly:
``pytest benchmarks/ -q``); scripts/bench_serving.py is the scripted
variant whose numbers land in BASELINE.md.
"""
import statistics
import time

import numpy as np
import pytest

ROUNDS = 100


def _payload(sensors, rows=100):
    from gordo_amd.server.utils import dataframe_to_dict
    import pandas as pd

    X = pd.DataFrame(
        np.random.random((rows, len(sensors))), columns=sensors
    )
    return {"X": dataframe_to_dict(X), "y": dataframe_to_dict(X)}


def _bench(api_client, url, payload, rounds=ROUNDS):
    # warmup
    resp = api_client.post(url, json=payload)
    assert resp.status_code == 200, resp.data[:500]
    times = []
    for _ in range(rounds):
        t0 = time.perf_counter()
        resp = api_client.post(url, json=payload)
        times.append(time.perf_counter() - t0)
        assert resp.status_code == 200
    return times


@pytest.mark.benchmark
def test_bench_ml_server_post(api_client, base_route, sensors):
    times = _bench(api_client, f"{base_route}/prediction", _payload(sensors))
    print(
        f"\n/prediction: mean {statistics.mean(times)*1000:.2f} ms, "
        f"median {statistics.median(times)*1000:.2f} ms, "
        f"{100/statistics.mean(times):.0f} predictions/s (100-row payload)"
    )


@pytest.mark.benchmark
def test_bench_ml_server_anomaly_post(api_client, base_route, sensors):
    times = _bench(
        api_client, f"{base_route}/anomaly/prediction", _payload(sensors)
    )
    print(
        f"\n/anomaly/prediction: mean {statistics.mean(times)*1000:.2f} ms, "
        f"median {statistics.median(times)*1000:.2f} ms, "
        f"{100/statistics.mean(times):.0f} predictions/s (100-row payload)"
    )
