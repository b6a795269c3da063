#!/usr/bin/env python3
"""
Flagship benchmark: machines built/hour on the 1000-machine mixed
feedforward/LSTM fleet config (BASELINE.json config #4), weak-scaled at
125 machines per GPU (N=8 GPUs -> the full 1000-machine config).

One step = one complete build of this rank's fleet shard: synthetic
sine-tag data fetch, TimeSeriesSplit(3) cross-validation (3 fold fits +
scoring + DiffBased thresholds), the final full fit, metadata assembly
and model serialization — i.e. exactly what `gordo build` does per
machine, through the packed grouped-MFMA engine.

Contract (driver): python bench.py --gpus N --steps K --warmup W
  * launched under torch.distributed.run for N>1 (one rank per GPU,
    RCCL); reads RANK/LOCAL_RANK/WORLD_SIZE from env.
  * W untimed warmup steps, then exactly K timed steps bracketed by
    barrier + torch.cuda.synchronize; MAX elapsed over ranks; rank 0
    prints ONE JSON line.
"""
from __future__ import annotations

import argparse
import json
import os
import shutil
import sys
import tempfile
import time

import numpy as np
import torch

REPO_ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO_ROOT)

N_TAGS = 50
ROWS_DAYS = 30          # 30 days @ 10min = 4320 rows
LOOKBACK = 144
EPOCHS = 5
BATCH_DENSE = 256
BATCH_LSTM = 256
MACHINES_PER_GPU = 125  # x8 GPUs = the 1000-machine config


def make_machine_cfg(i: int, kind: str) -> dict:
    tags = [f"sine-{i}-{j}" for j in range(N_TAGS)]
    dataset = {
        "type": "SineWaveDataset",
        "tag_list": tags,
        "train_start_date": "2019-01-01T00:00:00+00:00",
        "train_end_date": f"2019-01-{1 + ROWS_DAYS:02d}T00:00:00+00:00",
    }
    if kind == "lstm":
        model = {
            "gordo_amd.machine.model.anomaly.diff.DiffBasedAnomalyDetector": {
                "require_thresholds": True,
                "base_estimator": {
                    "sklearn.pipeline.Pipeline": {
                        "steps": [
                            "sklearn.preprocessing.MinMaxScaler",
                            {
                                "gordo_amd.machine.model.models.KerasLSTMAutoEncoder": {
                                    "kind": "lstm_hourglass",
                                    "lookback_window": LOOKBACK,
                                    "epochs": EPOCHS,
                                    "batch_size": BATCH_LSTM,
                                }
                            },
                        ]
                    }
                },
            }
        }
    else:
        model = {
            "gordo_amd.machine.model.anomaly.diff.DiffBasedAnomalyDetector": {
                "require_thresholds": True,
                "base_estimator": {
                    "sklearn.pipeline.Pipeline": {
                        "steps": [
                            "sklearn.preprocessing.MinMaxScaler",
                            {
                                "gordo_amd.machine.model.models.KerasAutoEncoder": {
                                    "kind": "feedforward_hourglass",
                                    "epochs": EPOCHS,
                                    "batch_size": BATCH_DENSE,
                                }
                            },
                        ]
                    }
                },
            }
        }
    return {"name": f"bench-{kind}-{i}", "model": model, "dataset": dataset}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--machines-per-gpu", type=int, default=MACHINES_PER_GPU)
    ap.add_argument("--verbose", action="store_true",
                    help="log per-phase build timings to stderr")
    args = ap.parse_args()

    import logging

    logging.basicConfig(
        level=logging.INFO if args.verbose else logging.WARNING,
        stream=sys.stderr,
    )

    from gordo_amd.machine import Machine
    from gordo_amd.parallel import init_distributed, shard_machines
    from gordo_amd.parallel.packed_builder import PackedFleetBuilder

    rank, world = init_distributed()
    n_gpus = max(args.gpus, world)
    on_gpu = torch.cuda.is_available()
    device = f"cuda:{os.environ.get('LOCAL_RANK', 0)}" if on_gpu else "cpu"

    total_machines = args.machines_per_gpu * n_gpus
    n_lstm = total_machines // 2
    cfgs = [
        make_machine_cfg(i, "lstm" if i < n_lstm else "dense")
        for i in range(total_machines)
    ]
    machines_all = [Machine.from_config(c, project_name="bench") for c in cfgs]
    shards = shard_machines(machines_all, world)
    my_machines = [machines_all[i] for i in shards[rank]] if world > 1 else (
        machines_all
    )

    out_dir = tempfile.mkdtemp(prefix=f"gordo-bench-r{rank}-")

    phase_acc: dict = {}

    # at 8 ranks per node, per-rank thread pools must share the host:
    # 16 fetch threads x 8 ranks oversubscribes the CPUs during the
    # data phase (ROADMAP r1 flagged this for the SCALE run)
    workers = max(4, 16 // max(world, 1))

    def one_step(timed: bool = False):
        builder = PackedFleetBuilder(
            my_machines, output_dir=out_dir, model_register_dir=None,
            device=device, save_models=True, data_workers=workers,
        )
        results = builder.build_all()
        failed = [n for n, r in results if isinstance(r, BaseException)]
        if failed:
            errs = {n: repr(r) for n, r in results if isinstance(r, BaseException)}
            raise RuntimeError(f"bench build failures: {errs}")
        if timed:
            for k, v in builder.phase_times.items():
                phase_acc[k] = phase_acc.get(k, 0.0) + v

    import torch.distributed as dist

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        one_step()

    barrier_sync()
    t0 = time.time()
    for _ in range(args.steps):
        one_step(timed=True)
    barrier_sync()
    elapsed = time.time() - t0

    # MAX over ranks
    if world > 1:
        t = torch.tensor(
            [elapsed],
            device=device if on_gpu else "cpu",
            dtype=torch.float64,
        )
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    shutil.rmtree(out_dir, ignore_errors=True)

    if rank == 0 and args.verbose:
        # per-step phase budget (seconds averaged over the timed steps;
        # 'total' is the sequential wall — overlapped keys are informational)
        budget = {k: round(v / args.steps, 3) for k, v in
                  sorted(phase_acc.items())}
        print(json.dumps({"phase_budget_s_per_step": budget}),
              file=sys.stderr)

    if rank == 0:
        ms_per_step = elapsed * 1000.0 / args.steps
        built = total_machines * args.steps
        machines_per_hour = built * 3600.0 / elapsed
        print(json.dumps({
            "metric": "machines built/hour (1000-model config)",
            "value": machines_per_hour,
            "unit": "machines/hour",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if on_gpu else "fp32",
            "data": "synthetic sine-wave tags, random-init weights",
            "config": {
                "model": "mixed KerasLSTMAutoEncoder(lookback=144) + "
                         "feedforward_hourglass AE fleet, "
                         "DiffBasedAnomalyDetector, cv=TimeSeriesSplit(3)",
                "global_batch": total_machines,
                "machines_per_gpu": args.machines_per_gpu,
                "seq_len": LOOKBACK,
                "rows": ROWS_DAYS * 144,
                "n_tags": N_TAGS,
                "epochs": EPOCHS,
                "parallelism": f"fleet-dp{n_gpus} (RCCL shard, grouped "
                               f"MFMA packs per rank)",
            },
        }))


if __name__ == "__main__":
    main()
