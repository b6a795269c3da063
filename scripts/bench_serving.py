#!/usr/bin/env python3
"""
Serving benchmark — ml_server predictions/sec (BASELINE.md metric #2,
replicating reference benchmarks/test_ml_server.py's payload shape:
100-sample x n_tags JSON POSTs to /prediction and /anomaly/prediction).

Trains a 50-tag feedforward DiffBased model (BASELINE config #2), dumps
it into a model-collection dir, serves it through the Flask app
in-process and measures sustained request throughput with concurrent
worker threads. On a GPU box the model's forward runs through the HIP
grouped-GEMM path.
"""
import argparse
import json
import os
import statistics
import sys
import tempfile
import threading
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

N_TAGS = 50
ROWS = 100  # overridable: --rows (10k-row requests exercise the fused
            # K9 device scoring path, BASELINE config #5)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=200)
    ap.add_argument("--threads", type=int, default=8)
    ap.add_argument("--endpoint", default="anomaly",
                    choices=["anomaly", "prediction", "both"])
    ap.add_argument("--rows", type=int, default=ROWS)
    ap.add_argument("--format", default="json", choices=["json", "parquet"],
                    help="request/response content type (parquet bypasses "
                         "the JSON codec entirely)")
    ap.add_argument("--n-models", type=int, default=1,
                    help="serve N distinct models round-robin (the "
                         "many-model LRU/HBM-residency story)")
    ap.add_argument("--serve-batch", action="store_true",
                    help="enable the per-model micro-batcher "
                         "(GORDO_SERVE_BATCH=1)")
    ap.add_argument("--direct", action="store_true",
                    help="measure model.anomaly() directly (no HTTP/JSON): "
                         "the batched inference engine path, BASELINE "
                         "config #5 shape (honors --n-models/--threads: "
                         "N models resident, T concurrent streams)")
    ap.add_argument("--hipgraph", action="store_true",
                    help="hipGraph-capture the serving forward per "
                         "request shape (GORDO_SERVE_HIPGRAPH=1)")
    ap.add_argument("--http-workers", type=int, default=0,
                    help="serve over REAL HTTP through the prefork "
                         "werkzeug pool with N worker processes "
                         "(0 = in-process Flask test client)")
    ap.add_argument("--profile-stages", action="store_true",
                    help="single-thread per-stage timing of the anomaly "
                         "endpoint work: JSON decode, model forward, "
                         "score, response encode")
    args = ap.parse_args()
    if args.hipgraph:
        os.environ["GORDO_SERVE_HIPGRAPH"] = "1"

    import pandas as pd
    import torch

    from gordo_amd import serializer
    from gordo_amd.builder import local_build
    from gordo_amd.server.utils import dataframe_to_dict

    sensors = [f"bench-tag-{i}" for i in range(N_TAGS)]
    tag_block = "\n".join(f"        - {s}" for s in sensors)
    machine_block = """
  - dataset: |
      tags:
{tags}
      train_start_date: '2019-01-01T00:00:00+00:00'
      train_end_date: '2019-01-08T00:00:00+00:00'
      type: SineWaveDataset
    model: |
      gordo.machine.model.anomaly.diff.DiffBasedAnomalyDetector:
        require_thresholds: true
        base_estimator:
          sklearn.pipeline.Pipeline:
            steps:
            - sklearn.preprocessing.MinMaxScaler
            - gordo.machine.model.models.KerasAutoEncoder:
                kind: feedforward_hourglass
                epochs: 3
    name: serve-bench{i}
"""
    config = "machines:" + "".join(
        machine_block.format(tags=tag_block, i=("" if i == 0 else f"-{i}"))
        for i in range(args.n_models)
    )
    with tempfile.TemporaryDirectory() as td:
        collection = os.path.join(td, "1577836800000")
        for model, machine in local_build(config):
            d = os.path.join(collection, machine.name)
            serializer.dump(model, d, metadata=json.loads(machine.to_json()),
                            info={})
        os.environ["MODEL_COLLECTION_DIR"] = collection

        if args.direct:
            import pandas as pd

            names = ["serve-bench"] + [
                f"serve-bench-{i}" for i in range(1, args.n_models)
            ]
            models = [serializer.load(os.path.join(collection, n))
                      for n in names]
            X = pd.DataFrame(np.random.random((args.rows, N_TAGS)),
                             columns=sensors)
            for m in models:
                m.anomaly(X, X)  # warmup (captures/caches)
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            counter = {"n": 0}
            lock = threading.Lock()

            def direct_worker():
                while True:
                    with lock:
                        if counter["n"] >= args.rounds:
                            return
                        i = counter["n"]
                        counter["n"] += 1
                    models[i % len(models)].anomaly(X, X)

            t0 = time.time()
            if args.threads > 1:
                ts = [threading.Thread(target=direct_worker)
                      for _ in range(args.threads)]
                for t in ts:
                    t.start()
                for t in ts:
                    t.join()
            else:
                direct_worker()
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            dt = (time.time() - t0) / args.rounds
            print(json.dumps({
                "metric": "batched anomaly() rows/sec (no HTTP)",
                "rows_per_call": args.rows,
                "ms_per_call": dt * 1000,
                "rows_per_sec": args.rows / dt,
                "n_models": args.n_models,
                "threads": args.threads,
                "hipgraph": bool(args.hipgraph),
                "device": "cuda" if torch.cuda.is_available() else "cpu",
            }))
            return

        if args.profile_stages:
            import pandas as pd

            from gordo_amd.server.utils import (
                dataframe_from_dict,
                dataframe_to_dict,
            )

            model = serializer.load(os.path.join(collection, "serve-bench"))
            Xp = pd.DataFrame(np.random.random((args.rows, N_TAGS)),
                              columns=sensors)
            payload = {"X": dataframe_to_dict(Xp), "y": dataframe_to_dict(Xp)}
            raw = json.dumps(payload).encode()
            # one pass to warm everything
            dec = json.loads(raw)
            Xd = dataframe_from_dict(dec["X"])
            yd = dataframe_from_dict(dec["y"])
            frame = model.anomaly(Xd, yd)
            from gordo_amd.server import utils as sutils

            def encode_response(fr):
                # the server's real encode path: C++ fastjson when the
                # frame qualifies, python codec otherwise
                # (server/utils.py frame_json_response)
                fj = sutils._gordo_fastjson
                if fj is not None:
                    index = fr.index.astype(str)
                    return b'{"data": ' + fj.encode_frame(
                        index.tolist(),
                        [str(c[0]) for c in fr.columns],
                        [str(c[1]) for c in fr.columns],
                        fr.values,
                    ) + b"}"
                return json.dumps(
                    {"data": sutils.dataframe_to_dict(fr)}
                ).encode()

            encode_response(frame)
            from gordo_amd.server.utils import _decode_request_fast

            use_fast = _decode_request_fast(raw) is not None
            stages = {k: 0.0 for k in
                      ("json_decode", "df_from_dict", "model_anomaly",
                       "encode_response")}
            n = args.rounds
            for _ in range(n):
                t0 = time.perf_counter()
                if use_fast:
                    # the server's real decode lane (C++ decode_request)
                    dec = _decode_request_fast(raw)
                    t1 = time.perf_counter()
                    Xd, yd = dec["X"], dec["y"]
                else:
                    dec = json.loads(raw)
                    t1 = time.perf_counter()
                    Xd = dataframe_from_dict(dec["X"])
                    yd = dataframe_from_dict(dec["y"])
                t2 = time.perf_counter()
                frame = model.anomaly(Xd, yd)
                if torch.cuda.is_available():
                    torch.cuda.synchronize()
                t3 = time.perf_counter()
                encode_response(frame)
                t4 = time.perf_counter()
                stages["json_decode"] += t1 - t0
                stages["df_from_dict"] += t2 - t1
                stages["model_anomaly"] += t3 - t2
                stages["encode_response"] += t4 - t3
            print(json.dumps({
                "metric": "anomaly endpoint stage budget (ms/request)",
                "rows": args.rows,
                "fast_decode": use_fast,
                "hipgraph": bool(args.hipgraph),
                "device": "cuda" if torch.cuda.is_available() else "cpu",
                "stages_ms": {k: round(v / n * 1000, 3)
                              for k, v in stages.items()},
                "sum_ms": round(sum(stages.values()) / n * 1000, 3),
            }))
            return

        if args.serve_batch:
            os.environ["GORDO_SERVE_BATCH"] = "1"

        if args.http_workers > 0:
            _run_http_bench(args, collection, sensors)
            return

        from gordo_amd.server.server import build_app

        app = build_app()
        app.testing = True

        X = pd.DataFrame(np.random.random((args.rows, N_TAGS)),
                         columns=sensors)
        payload = {"X": dataframe_to_dict(X), "y": dataframe_to_dict(X)}
        parquet_blob = None
        if args.format == "parquet":
            from gordo_amd.server.utils import dataframe_into_parquet_bytes

            parquet_blob = dataframe_into_parquet_bytes(X)
        results = {}
        for endpoint in (
            ["anomaly", "prediction"] if args.endpoint == "both"
            else [args.endpoint]
        ):
            model_names = ["serve-bench"] + [
                f"serve-bench-{i}" for i in range(1, args.n_models)
            ]
            urls = [
                f"/gordo/v0/bench/{name}/"
                + ("anomaly/prediction" if endpoint == "anomaly"
                   else "prediction")
                for name in model_names
            ]
            url = urls[0]
            def post(c, u):
                if args.format == "parquet":
                    import io

                    return c.post(
                        u + "?format=parquet",
                        data={"X": (io.BytesIO(parquet_blob), "X"),
                              "y": (io.BytesIO(parquet_blob), "y")},
                        content_type="multipart/form-data",
                    )
                return c.post(u, json=payload)

            # warmup + correctness
            client = app.test_client()
            resp = post(client, url)
            assert resp.status_code == 200, resp.data[:300]

            latencies = []
            lock = threading.Lock()
            counter = {"n": 0}

            def worker():
                c = app.test_client()
                while True:
                    with lock:
                        if counter["n"] >= args.rounds:
                            return
                        counter["n"] += 1
                        my_url = urls[counter["n"] % len(urls)]
                    t0 = time.perf_counter()
                    r = post(c, my_url)
                    dt = time.perf_counter() - t0
                    assert r.status_code == 200
                    with lock:
                        latencies.append(dt)

            t0 = time.time()
            threads = [
                threading.Thread(target=worker) for _ in range(args.threads)
            ]
            for t in threads:
                t.start()
            for t in threads:
                t.join()
            elapsed = time.time() - t0
            rps = len(latencies) / elapsed
            results[endpoint] = {
                "requests_per_sec": rps,
                "predictions_per_sec": rps * args.rows,
                "mean_latency_ms": statistics.mean(latencies) * 1000,
                "p50_latency_ms": statistics.median(latencies) * 1000,
                "rounds": len(latencies),
                "format": args.format,
                "threads": args.threads,
                "payload_rows": args.rows,
                "n_tags": N_TAGS,
                "n_models": args.n_models,
                "device": "cuda" if torch.cuda.is_available() else "cpu",
            }
        print(json.dumps({"metric": "ml_server predictions/sec",
                          "results": results}))


def _run_http_bench(args, collection, sensors):
    """Real-HTTP benchmark against the prefork multi-process server:
    the number the reference's gunicorn-worker deployment would see."""
    import socket
    import signal
    import statistics
    import subprocess
    import urllib.request

    import pandas as pd

    from gordo_amd.server.utils import dataframe_to_dict

    # GPU boxes may set http_proxy; 127.0.0.1 must never route there
    opener = urllib.request.build_opener(
        urllib.request.ProxyHandler({})
    )

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    env = dict(os.environ)
    env["MODEL_COLLECTION_DIR"] = collection
    proc = subprocess.Popen(
        [sys.executable, "-m", "gordo_amd", "run-server",
         "--host", "127.0.0.1", "--port", str(port),
         "--workers", str(args.http_workers)],
        env=env,
    )
    try:
        X = pd.DataFrame(np.random.random((args.rows, N_TAGS)),
                         columns=sensors)
        payload = json.dumps(
            {"X": dataframe_to_dict(X), "y": dataframe_to_dict(X)}
        ).encode()
        results = {}
        endpoints = (["anomaly", "prediction"] if args.endpoint == "both"
                     else [args.endpoint])
        model_names = ["serve-bench"] + [
            f"serve-bench-{i}" for i in range(1, args.n_models)
        ]
        for endpoint in endpoints:
            urls = [
                f"http://127.0.0.1:{port}/gordo/v0/bench/{name}/"
                + ("anomaly/prediction" if endpoint == "anomaly"
                   else "prediction")
                for name in model_names
            ]

            def post(u):
                req = urllib.request.Request(
                    u, data=payload,
                    headers={"Content-Type": "application/json"},
                )
                with opener.open(req, timeout=60) as r:
                    assert r.status == 200
                    r.read()

            deadline = time.time() + 180
            last_err = None
            while True:  # wait for workers up + model load
                try:
                    post(urls[0])
                    break
                except Exception as e:
                    last_err = e
                    if time.time() > deadline:
                        print(json.dumps({
                            "metric": "ml_server predictions/sec "
                                      "(real HTTP, prefork)",
                            "error": repr(last_err),
                        }))
                        raise
                    time.sleep(0.5)
            for u in urls:
                post(u)  # warm every model
            latencies = []
            lock = threading.Lock()
            counter = {"n": 0}

            def worker():
                while True:
                    with lock:
                        if counter["n"] >= args.rounds:
                            return
                        counter["n"] += 1
                        my_url = urls[counter["n"] % len(urls)]
                    t0 = time.perf_counter()
                    post(my_url)
                    with lock:
                        latencies.append(time.perf_counter() - t0)

            t0 = time.time()
            ts = [threading.Thread(target=worker)
                  for _ in range(args.threads)]
            for t in ts:
                t.start()
            for t in ts:
                t.join()
            elapsed = time.time() - t0
            rps = len(latencies) / elapsed
            results[endpoint] = {
                "requests_per_sec": rps,
                "predictions_per_sec": rps * args.rows,
                "mean_latency_ms": statistics.mean(latencies) * 1000,
                "p50_latency_ms": statistics.median(latencies) * 1000,
                "rounds": len(latencies),
                "http_workers": args.http_workers,
                "threads": args.threads,
                "payload_rows": args.rows,
                "n_models": args.n_models,
            }
        print(json.dumps({
            "metric": "ml_server predictions/sec (real HTTP, prefork)",
            "results": results,
        }))
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=20)
        except subprocess.TimeoutExpired:
            proc.kill()


if __name__ == "__main__":
    main()
