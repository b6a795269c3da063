"""
``gordo fleet`` — the MI355X-native single-node fan-out.

No reference analog: this replaces the reference's pod-per-model Argo
workflow with one process per GPU over RCCL/xGMI on a single node
(SURVEY.md §1 "local fan-out" row). ``gordo fleet build`` builds every
Machine in a config, packing same-architecture models into grouped
MFMA GEMM batches per GPU; with --gpus N it relaunches itself under
``torch.distributed.run`` with one rank per GPU.
"""
from __future__ import annotations

import json
import logging
import os
import subprocess
import sys
import time
from typing import Optional

import click

logger = logging.getLogger(__name__)

# default status-file name when the caller gives none; lives inside the
# output dir and is ignored by the server's /models listing (dirs only)
FLEET_STATUS_BASENAME = ".fleet-status.json"


@click.group("fleet")
def fleet_cli():
    """Single-node multi-GPU fleet operations (MI355X-native)."""


@fleet_cli.command("build")
@click.option("--machine-config", type=click.Path(exists=True), required=True,
              envvar="GORDO_FLEET_MACHINE_CONFIG")
@click.option("--project-name", type=str, required=True,
              envvar="GORDO_FLEET_PROJECT_NAME")
@click.option("--output-dir", type=click.Path(), required=True,
              envvar="GORDO_FLEET_OUTPUT_DIR",
              help="Model collection dir; models land at <output-dir>/<name>/")
@click.option("--model-register-dir", type=click.Path(), default=None,
              envvar="GORDO_FLEET_MODEL_REGISTER_DIR")
@click.option("--gpus", type=int, default=0,
              help="Number of GPUs (0 = auto: all visible; runs in-process "
                   "when 0/1)")
@click.option("--replace-cache", is_flag=True)
@click.option("--status-file", type=click.Path(), default=None,
              help="Write per-machine build status JSON here (rank 0)")
@click.option("--requeue/--no-requeue", default=True,
              help="After a distributed run, rebuild failed or missing "
                   "machines in-process (rank-crash recovery)")
def fleet_build(
    machine_config: str,
    project_name: str,
    output_dir: str,
    model_register_dir: Optional[str],
    gpus: int,
    replace_cache: bool,
    status_file: Optional[str],
    requeue: bool = True,
):
    """Build every Machine in the config across the node's GPUs."""
    import torch

    in_torchrun = "RANK" in os.environ
    if not in_torchrun:
        if gpus == 0:
            gpus = torch.cuda.device_count() if torch.cuda.is_available() else 1
        if gpus > 1:
            # relaunch under torch.distributed.run, one rank per GPU
            # standalone rendezvous picks a free port (a fixed port can
            # collide with another torchrun on a shared box)
            cmd = [
                sys.executable, "-m", "torch.distributed.run",
                "--nnodes=1", f"--nproc-per-node={gpus}",
                "--standalone", "--local-addr", "127.0.0.1",
                "-m", "gordo_amd.cli.fleet_worker",
            ]
            env = dict(os.environ)
            env.update(
                GORDO_FLEET_MACHINE_CONFIG=machine_config,
                GORDO_FLEET_PROJECT_NAME=project_name,
                GORDO_FLEET_OUTPUT_DIR=output_dir,
                GORDO_FLEET_REPLACE_CACHE="1" if replace_cache else "0",
            )
            if model_register_dir:
                env["GORDO_FLEET_MODEL_REGISTER_DIR"] = model_register_dir
            if status_file is None:
                status_file = os.path.join(output_dir, FLEET_STATUS_BASENAME)
            env["GORDO_FLEET_STATUS_FILE"] = status_file
            os.makedirs(output_dir, exist_ok=True)
            rc = subprocess.call(cmd, env=env)
            if requeue:
                n_failed = requeue_failed(
                    machine_config, project_name, output_dir,
                    model_register_dir, replace_cache, status_file,
                    distributed_rc=rc,
                )
                raise SystemExit(0 if n_failed == 0 else 1)
            raise SystemExit(rc)

    _run_fleet_build(
        machine_config, project_name, output_dir, model_register_dir,
        replace_cache, status_file,
    )


def requeue_failed(
    machine_config: str,
    project_name: str,
    output_dir: str,
    model_register_dir: Optional[str],
    replace_cache: bool,
    status_file: str,
    distributed_rc: int = 0,
) -> int:
    """Elastic recovery after a distributed fleet run:
    machines that failed, or never produced a model dir because their
    rank died (torchrun tears the whole job down on a rank crash),
    are rebuilt in-process on one device. Returns the number of
    machines still failed afterwards and rewrites ``status_file`` with
    the merged outcome."""
    from ..parallel import PackedFleetBuilder
    from ..workflow import NormalizedConfig
    from ..workflow.workflow_generator import get_dict_from_yaml

    config = get_dict_from_yaml(machine_config)
    all_names = [m.get("name") for m in config.get("machines", [])]

    prior_status: dict = {}
    summary: dict = {}
    if os.path.exists(status_file):
        try:
            with open(status_file) as f:
                summary = json.load(f)
            prior_status = dict(summary.get("status", {}))
        except (ValueError, OSError):
            summary = {}

    def built(name: str) -> bool:
        return os.path.isfile(os.path.join(output_dir, name, "model.pkl"))

    retry = [
        n for n in all_names
        if prior_status.get(n) is not None or not built(n)
    ]
    if not retry:
        return 0
    logger.warning(
        "Requeueing %d machine(s) after distributed run (rc=%d): %s",
        len(retry), distributed_rc, retry,
    )
    sub_config = dict(config)
    sub_config["machines"] = [
        m for m in config.get("machines", []) if m.get("name") in retry
    ]
    norm = NormalizedConfig(sub_config, project_name=project_name)
    builder = PackedFleetBuilder(
        norm.machines,
        output_dir=output_dir,
        model_register_dir=model_register_dir,
        replace_cache=replace_cache,
    )
    for name, res in builder.build_all():
        prior_status[name] = (
            None if not isinstance(res, BaseException) else repr(res)
        )
    n_ok = sum(1 for n in all_names if prior_status.get(n) is None)
    summary.update(
        project=project_name,
        n_machines=len(all_names),
        n_ok=n_ok,
        n_failed=len(all_names) - n_ok,
        requeued=retry,
        status=prior_status,
    )
    with open(status_file, "w") as f:
        json.dump(summary, f, indent=2, default=str)
    return len(all_names) - n_ok


def _run_fleet_build(
    machine_config: str,
    project_name: str,
    output_dir: str,
    model_register_dir: Optional[str],
    replace_cache: bool,
    status_file: Optional[str],
):
    from ..parallel import build_fleet, init_distributed
    from ..workflow import NormalizedConfig
    from ..workflow.workflow_generator import get_dict_from_yaml

    rank, world = init_distributed()
    config = get_dict_from_yaml(machine_config)
    norm = NormalizedConfig(config, project_name=project_name)
    t0 = time.time()
    status = build_fleet(
        norm.machines,
        output_dir=output_dir,
        model_register_dir=model_register_dir,
        replace_cache=replace_cache,
    )
    elapsed = time.time() - t0
    if rank == 0:
        n_ok = sum(1 for _, err in status if err is None)
        summary = {
            "project": project_name,
            "n_machines": len(norm.machines),
            "n_ok": n_ok,
            "n_failed": len(status) - n_ok,
            "elapsed_sec": elapsed,
            "machines_per_hour": len(norm.machines) / elapsed * 3600.0,
            "world_size": world,
            "status": dict(status),
        }
        logger.info(
            "Fleet build: %d/%d machines OK in %.1fs (%.0f machines/hour, %d ranks)",
            n_ok, len(norm.machines), elapsed,
            summary["machines_per_hour"], world,
        )
        if status_file:
            with open(status_file, "w") as f:
                json.dump(summary, f, indent=2, default=str)
        if n_ok < len(status):
            for name, err in status:
                if err is not None:
                    logger.error("Machine %s failed: %s", name, err)
