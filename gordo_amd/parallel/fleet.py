"""
Multi-GPU fleet build — one process per GPU over RCCL/xGMI.

The distributed analog of the reference's Argo DAG:
rank 0 parses the YAML config and scatters serialized Machine specs to
all ranks (``torch.distributed`` backend "nccl" IS RCCL on ROCm; the
control-plane messages are KB-scale so rank-0-rooted scatter/gather is
the right collective shape for xGMI's 7 point-to-point links — SURVEY
§2.5). Each rank builds its shard with the PackedFleetBuilder on its
own GPU and writes the shared model-collection layout; statuses gather
back to rank 0. Machines are sharded round-robin **within each
architecture group** so every rank keeps large grouped-GEMM batches
AND a balanced dense/LSTM mix.

Runs on "gloo" for CPU-only multi-process tests.
"""
from __future__ import annotations

import json
import logging
import os
from typing import Any, Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from .packed_builder import PackedFleetBuilder
from ..machine import Machine

logger = logging.getLogger(__name__)


def init_distributed(backend: Optional[str] = None) -> Tuple[int, int]:
    """Initialize torch.distributed from torchrun env vars; no-op in
    single-process mode. Returns (rank, world_size)."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return 0, 1
    if backend is None:
        backend = os.environ.get("GORDO_DIST_BACKEND")
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
        if local_rank >= torch.cuda.device_count():
            # oversubscribed (more ranks than GPUs, e.g. a CPU-parallel
            # test on a 1-GPU box): RCCL cannot place the rank — fall
            # back to gloo instead of crashing
            backend = "gloo"
        else:
            torch.cuda.set_device(local_rank)
    dist.init_process_group(backend=backend)
    return dist.get_rank(), dist.get_world_size()


def shard_machines(
    machines: List[Machine], world_size: int
) -> List[List[int]]:
    """Round-robin indices within each model-config group so every
    rank's shard keeps big pack batches and a balanced mix."""
    groups: Dict[str, List[int]] = {}
    for i, m in enumerate(machines):
        key = json.dumps(m.model, sort_keys=True, default=str)
        groups.setdefault(key, []).append(i)
    shards: List[List[int]] = [[] for _ in range(world_size)]
    for idxs in groups.values():
        for j, i in enumerate(idxs):
            shards[j % world_size].append(i)
    return shards


def build_fleet(
    machines: List[Machine],
    output_dir: Optional[str] = None,
    model_register_dir: Optional[str] = None,
    device: Optional[str] = None,
    save_models: bool = True,
    replace_cache: bool = False,
) -> List[Tuple[str, Any]]:
    """
    Build all ``machines``, sharded across the ranks of the current
    process group (the whole list when not distributed). Rank 0 scatters
    each rank's shard of serialized machine dicts over the wire (RCCL on
    GPU); every rank builds its shard; per-machine status gathers back.
    Returns the full fleet status list on rank 0 ([] on other ranks).
    """
    rank, world_size = init_distributed()

    if world_size > 1:
        if rank == 0:
            shards = shard_machines(machines, world_size)
            scatter_in = [
                [machines[i].to_dict() for i in shard] for shard in shards
            ]
        else:
            scatter_in = None
        holder: List[Any] = [None]
        dist.scatter_object_list(holder, scatter_in, src=0)
        my_machines = [Machine.from_dict(d) for d in holder[0]]
    else:
        my_machines = machines

    if device is None:
        if torch.cuda.is_available():
            device = f"cuda:{os.environ.get('LOCAL_RANK', 0)}"
        else:
            device = "cpu"

    builder = PackedFleetBuilder(
        my_machines,
        output_dir=output_dir,
        model_register_dir=model_register_dir,
        device=device,
        save_models=save_models,
        replace_cache=replace_cache,
    )
    results = builder.build_all()
    status = [
        (name, None if not isinstance(res, BaseException) else repr(res))
        for name, res in results
    ]

    if world_size > 1:
        gathered: Optional[List[Any]] = [None] * world_size if rank == 0 else None
        dist.gather_object(status, gathered, dst=0)
        dist.barrier()
        if rank == 0:
            flat: List[Tuple[str, Any]] = []
            for shard_status in gathered:
                flat.extend(shard_status)
            return flat
        return []
    return status
