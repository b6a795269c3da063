"""
Packed fleet builder — the MI355X replacement for the reference's
pod-per-model Argo fan-out.

Where the reference trains each Machine in its own Kubernetes pod
(argo-workflow.yml.template:709-776), this builder classifies every
Machine's model definition, groups machines whose device architecture
matches (same layer dims/activations, same row count, same fit args),
and trains each group as ONE ``engine.pack`` — every layer of every
machine in the group is a single grouped MFMA GEMM launch. Cross-
validation (TimeSeriesSplit / KFold), DiffBased threshold calculation,
metadata assembly and the on-disk model layout reproduce
``ModelBuilder`` semantics exactly; machines whose models aren't
packable (arbitrary sklearn pipelines, config #1) fall back to the
per-machine ModelBuilder on CPU.
"""
from __future__ import annotations

import concurrent.futures
import contextlib
import datetime
import hashlib
import json
import logging
import os
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

import numpy as np
import pandas as pd
import torch
from sklearn.base import clone as sk_clone
from sklearn.pipeline import Pipeline

import gordo_amd
from .. import serializer
from ..builder.build_model import ModelBuilder
from ..core.base import GordoBaseDataset
from ..engine.pack import DensePack, LSTMPack
from ..engine.spec import ModelSpec
from ..machine import Machine
from ..machine.metadata import (
    BuildMetadata,
    CrossValidationMetaData,
    DatasetBuildMetadata,
    ModelBuildMetadata,
)
from ..machine.model.anomaly.diff import (
    DiffBasedAnomalyDetector,
    DiffBasedKFCVAnomalyDetector,
)
from ..machine.model.models import _parse_early_stopping, KerasBaseEstimator
from ..machine.model.utils import trail_min_max as _trail_min_max
from ..util import disk_registry

logger = logging.getLogger(__name__)

_nullcontext = contextlib.nullcontext

# deterministic weight-init snapshots keyed by (arch, seeds) — see
# PackedFleetBuilder._build_group
_INIT_CACHE: Dict[Any, torch.Tensor] = {}


# ---------------------------------------------------------------------------
@dataclass
class MachinePlan:
    machine: Machine
    model: Any = None            # instantiated estimator graph
    detector: Optional[DiffBasedAnomalyDetector] = None
    pre_steps: Optional[List[Tuple[str, Any]]] = None
    keras_est: Optional[KerasBaseEstimator] = None
    X: Optional[pd.DataFrame] = None
    y: Optional[pd.DataFrame] = None
    dataset_meta: Dict[str, Any] = field(default_factory=dict)
    query_duration: float = 0.0
    packable: bool = False
    error: Optional[BaseException] = None
    # filled during build
    scores: Dict[str, Any] = field(default_factory=dict)
    splits: Dict[str, Any] = field(default_factory=dict)
    spec: Optional[ModelSpec] = None
    seed: int = 0


def _classify(model) -> Tuple[Optional[DiffBasedAnomalyDetector], Optional[list], Optional[KerasBaseEstimator]]:
    """Recognize the packable shapes:
    [DiffBased*Detector(base_estimator=)] [Pipeline(pre..., )] KerasEst."""
    detector = None
    inner = model
    if isinstance(model, DiffBasedAnomalyDetector):
        detector = model
        inner = model.base_estimator
    pre_steps: List[Tuple[str, Any]] = []
    if isinstance(inner, Pipeline):
        *pre, last = inner.steps
        pre_steps = list(pre)
        inner = last[1]
    if isinstance(inner, KerasBaseEstimator):
        return detector, pre_steps, inner
    return None, None, None


def _machine_seed(name: str, eval_seed: int) -> int:
    h = hashlib.sha256(f"{name}:{eval_seed}".encode()).digest()
    return int.from_bytes(h[:4], "little") & 0x7FFFFFFF


class PackedFleetBuilder:
    """Build many Machines on one device, packing same-architecture
    models into grouped trainers."""

    def __init__(
        self,
        machines: List[Machine],
        output_dir: Optional[str] = None,
        model_register_dir: Optional[str] = None,
        device: Optional[str] = None,
        data_workers: int = 16,
        save_models: bool = True,
        replace_cache: bool = False,
    ):
        self.machines = machines
        self.output_dir = output_dir
        self.model_register_dir = model_register_dir
        self.device = device or (
            "cuda" if torch.cuda.is_available() else "cpu"
        )
        self.data_workers = data_workers
        self.save_models = save_models
        self.replace_cache = replace_cache
        # async model-save pool: serialization of a finished group
        # overlaps the next group's GPU fits (joined in build_all)
        # share the host across ranks: WORLD_SIZE ranks each run a pool
        world = int(os.environ.get("WORLD_SIZE", "1") or 1)
        pool_n = max(4, min(16, (os.cpu_count() or 8) // max(world, 1)))
        self._save_pool = concurrent.futures.ThreadPoolExecutor(pool_n)
        self._save_futures: List[Tuple[MachinePlan, Any]] = []
        # wall-clock phase ledger for the build-step budget table
        # (BASELINE.md): sequential wall segments; overlapped work
        # (threaded final fit) is recorded under its own key.
        self.phase_times: Dict[str, float] = {}

    def _phase(self, key: str, dt: float):
        self.phase_times[key] = self.phase_times.get(key, 0.0) + dt

    # ---- public ----------------------------------------------------------
    def build_all(self) -> List[Tuple[str, Any]]:
        """Build every machine. Returns [(name, Machine-with-metadata |
        exception)]. Models are saved under output_dir/<name>/ when
        save_models is set."""
        t_all0 = time.time()
        plans = [MachinePlan(machine=m) for m in self.machines]
        t0 = time.time()
        self._instantiate_models(plans)
        self._phase("instantiate", time.time() - t0)

        # Fetches run on a thread pool in the BACKGROUND; each
        # prospective group's futures are joined only when that group is
        # about to build, so the data fetch of group g+1 overlaps the
        # GPU fits of group g (pandas resample/join releases the GIL).
        # GORDO_PREFETCH=0 restores the blocking up-front fetch.
        prefetch = os.environ.get("GORDO_PREFETCH", "1") != "0"
        packable = [p for p in plans if p.packable and p.error is None]
        # fallback plans go through ModelBuilder.build, which fetches its
        # own data — no builder-side fetch for them.
        fallback = [p for p in plans if not p.packable and p.error is None]

        fetch_pool = concurrent.futures.ThreadPoolExecutor(self.data_workers)
        try:
            futs = {
                id(p): fetch_pool.submit(self._fetch_one, p)
                for p in packable
            }
            if not prefetch:
                t0 = time.time()
                for f in futs.values():
                    f.result()
                self._phase("fetch", time.time() - t0)

            # NOTE: running independent groups on concurrent threads/streams
            # was measured SLOWER (28.3k vs 33.4k machines/hour): the extra
            # python threads contend on the GIL with the fold-fit threads
            # and starve kernel dispatch. Groups run sequentially; only the
            # CV folds within a group overlap (streams, _fit_folds).
            # Largest pre-groups first: the final group's async
            # adopt+save is the only one the end-of-build save_join
            # cannot hide, so leave the smallest for last. Ties keep
            # config order (sort is stable).
            pre_groups = sorted(
                self._pre_group(packable), key=len, reverse=True
            )
            for pre_group in pre_groups:
                t0 = time.time()
                for p in pre_group:
                    futs[id(p)].result()
                self._phase("fetch", time.time() - t0)
                ready = [p for p in pre_group if p.error is None]
                for group in sorted(self._group(ready), key=len,
                                    reverse=True):
                    try:
                        self._build_group(group)
                    except Exception as e:  # isolate group failures
                        logger.exception("Pack group build failed")
                        for p in group:
                            p.error = e
        finally:
            fetch_pool.shutdown(wait=True)

        t0 = time.time()
        for p in fallback:
            try:
                logger.info("Fallback per-machine build: %s", p.machine.name)
                builder = ModelBuilder(p.machine)
                model, machine_out = builder.build(
                    output_dir=(
                        os.path.join(self.output_dir, p.machine.name)
                        if self.output_dir and self.save_models
                        else None
                    ),
                    model_register_dir=self.model_register_dir,
                    replace_cache=self.replace_cache,
                )
                p.machine = machine_out
                p.model = model
            except Exception as e:
                logger.exception("Fallback build failed: %s", p.machine.name)
                p.error = e

        if fallback:
            self._phase("fallback_builds", time.time() - t0)
        # join outstanding async saves; failures surface per machine
        t0 = time.time()
        if self._save_futures:
            for plans_of_group, f in self._save_futures:
                try:
                    f.result()
                except Exception as e:
                    logger.exception("async adopt/save failed")
                    for p in plans_of_group:
                        p.error = e
            self._save_futures.clear()
        self._phase("save_join", time.time() - t0)
        self._phase("total", time.time() - t_all0)
        if logger.isEnabledFor(logging.INFO):
            budget = ", ".join(
                f"{k}={v:.2f}s" for k, v in sorted(self.phase_times.items())
            )
            logger.info("build-step phase budget: %s", budget)

        results: List[Tuple[str, Any]] = []
        for p in plans:
            results.append(
                (p.machine.name, p.error if p.error is not None else p.machine)
            )
        return results

    # ---- stages ----------------------------------------------------------
    @staticmethod
    def _fetch_one(p: MachinePlan):
        try:
            start = time.time()
            dataset = GordoBaseDataset.from_dict(p.machine.dataset.to_dict())
            p.X, p.y = dataset.get_data()
            p.query_duration = time.time() - start
            p.dataset_meta = dataset.get_metadata()
        except BaseException as e:
            p.error = e

    @staticmethod
    def _pre_group(plans: List[MachinePlan]) -> List[List[MachinePlan]]:
        """Config-level grouping done BEFORE any data is fetched, so each
        prospective group's fetch futures can be joined lazily.

        Key = model definition + evaluation + dataset shape-affecting
        fields (tag COUNTS, date range, resolution, filters — not tag
        names). Machines whose fetched data still differs in row count
        or width are split afterwards by `_group`, which keys on the
        actual arrays."""
        groups: Dict[str, List[MachinePlan]] = {}
        for p in plans:
            d = dict(p.machine.dataset.to_dict())
            for k in ("tags", "tag_list", "target_tag_list"):
                if isinstance(d.get(k), (list, tuple)):
                    d[k] = len(d[k])
            key = json.dumps(
                {
                    "model": p.machine.model,
                    "evaluation": p.machine.evaluation,
                    "dataset": d,
                },
                sort_keys=True,
                default=str,
            )
            groups.setdefault(key, []).append(p)
        return list(groups.values())

    def _instantiate_models(self, plans: List[MachinePlan]):
        for p in plans:
            if p.error is not None:
                continue
            try:
                p.model = serializer.from_definition(p.machine.model)
            except Exception as e:
                p.error = e
                continue
            detector, pre_steps, keras_est = _classify(p.model)
            if keras_est is not None:
                p.detector = detector
                p.pre_steps = pre_steps
                p.keras_est = keras_est
                p.packable = True
            p.seed = _machine_seed(
                p.machine.name, int(p.machine.evaluation.get("seed", 0))
            )

    # pack-size caps: BPTT caches are ~B*T*(11H)*layers bytes per LSTM
    # model per batch (plus hipGraph retention); 128 LSTM models per
    # pack keeps a 288 GB GPU comfortable at lookback 144 while still
    # amortizing kernel launches. Dense models are ~100x lighter.
    MAX_PACK_LSTM = int(os.environ.get("GORDO_MAX_PACK_LSTM", 128))
    MAX_PACK_DENSE = int(os.environ.get("GORDO_MAX_PACK_DENSE", 1024))

    def _group(self, plans: List[MachinePlan]) -> List[List[MachinePlan]]:
        groups: Dict[Any, List[MachinePlan]] = {}
        for p in plans:
            est = p.keras_est
            n_features = p.X.shape[1]
            n_features_out = p.y.shape[1]
            try:
                spec = est.build_pack_spec(n_features, n_features_out)
            except Exception as e:
                p.error = e
                continue
            p.spec = spec
            key = (
                spec.arch_key(),
                len(p.X),
                json.dumps(est.fit_args(), sort_keys=True, default=str),
                json.dumps(p.machine.evaluation, sort_keys=True, default=str),
                tuple(type(s[1]).__name__ for s in p.pre_steps),
                p.detector is not None,
            )
            groups.setdefault(key, []).append(p)
        out: List[List[MachinePlan]] = []
        for key, members in groups.items():
            cap = (
                self.MAX_PACK_LSTM
                if members[0].spec.model_type == "lstm"
                else self.MAX_PACK_DENSE
            )
            for s0 in range(0, len(members), cap):
                out.append(members[s0 : s0 + cap])
        return out

    # ---- the packed group build -----------------------------------------
    def _pack_cls(self, spec: ModelSpec):
        return LSTMPack if spec.model_type == "lstm" else DensePack

    def _make_pack(self, spec: ModelSpec, group: List[MachinePlan],
                   init_p32=None):
        return self._pack_cls(spec)(
            spec,
            G=len(group),
            device=self.device,
            seeds=[p.seed for p in group],
            init_p32=init_p32,
        )

    def _stack(self, arrays: List[np.ndarray], pack) -> torch.Tensor:
        t = torch.from_numpy(np.stack(arrays).astype(np.float32))
        return t.to(pack.device, pack.compute_dtype)

    def _build_group(self, group: List[MachinePlan]):
        spec = group[0].spec
        evaluation = group[0].machine.evaluation
        fit_args = group[0].keras_est.fit_args()
        logger.info(
            "Packed build: %d machines, arch=%s/%s, device=%s",
            len(group), spec.model_type,
            [l.units for l in spec.layers], self.device,
        )

        t_seg = time.time()
        # cache probe: skip machines already registered
        to_build: List[MachinePlan] = []
        for p in group:
            if self.model_register_dir and not self.replace_cache:
                key = ModelBuilder(p.machine).cache_key
                cached = ModelBuilder.check_cache(self.model_register_dir, key)
                if cached:
                    p.machine = Machine.from_dict(serializer.load_metadata(cached))
                    p.model = serializer.load(cached)
                    continue
            to_build.append(p)
        if not to_build:
            return
        group = to_build
        self._phase("cache_probe", time.time() - t_seg)
        t_seg = time.time()

        # pre-transform for the FINAL fit (CPU sklearn scalers etc.,
        # fitted per machine on the full series — reference semantics:
        # Pipeline.fit fits pre-steps on everything it is given). CV
        # folds refit clones of the pre-steps on each fold's train
        # slice (_fold_transform), matching sklearn.cross_validate's
        # per-fold pipeline clone — no test-fold leakage into scores
        # or DiffBased thresholds.
        Xraw_list, y_list = [], []
        for p in group:
            Xraw_list.append(p.X.values.astype(np.float32))
            y_list.append(p.y.values.astype(np.float32))

        def full_pre_transform():
            # runs INSIDE the final-fit closure so the per-machine
            # sklearn fits overlap the concurrent fold fits on GPU
            t0 = time.time()
            out = []
            for p, Xraw in zip(group, Xraw_list):
                Xt = Xraw
                for _, step in p.pre_steps:
                    Xt = step.fit_transform(Xt)
                out.append(np.asarray(Xt, dtype=np.float32))
            self._phase("pre_transform(overlapped)", time.time() - t0)
            return out
        self._phase("data_prep", time.time() - t_seg)

        cv_mode = str(evaluation.get("cv_mode", "full_build")).lower()
        cv_duration = None
        t0_all = time.time()
        fit_state = {}

        def final_fit(stream):
            ctx = (
                torch.cuda.stream(stream) if stream is not None
                else _nullcontext()
            )
            with ctx:
                Xt_list = full_pre_transform()
                fit_state["Xt_list"] = Xt_list
                self._reset_pack(pack, init_snapshot)
                Xd = self._stack(Xt_list, pack)
                Yd = self._stack(y_list, pack)
                t0 = time.time()
                fit_state["history"] = pack.fit(
                    Xd, Yd, **_engine_fit_args(fit_args)
                )
                fit_state["duration"] = time.time() - t0
        # ONE pack serves every fold fit and the final fit: weight init
        # (per-model glorot + orthogonal QR on CPU) costs ~1 s per pack,
        # so fold fits reset to an init snapshot instead of re-creating;
        # the init itself is deterministic in (arch, per-machine seeds)
        # and memoized across builds.
        t_seg = time.time()
        from ..engine.pack import pad_enabled

        # the snapshot layout depends on whether this device pads dims
        # to 8 (GPU) — a CPU (unpadded) snapshot must never be reused
        # for a GPU pack in the same process
        init_key = (
            spec.arch_key(),
            tuple(p.seed for p in group),
            pad_enabled(self.device),
        )
        cached_init = _INIT_CACHE.get(init_key)
        pack = self._make_pack(spec, group, init_p32=cached_init)
        init_snapshot = pack.store.p32.clone()
        self._phase("pack_init", time.time() - t_seg)
        if cached_init is None:
            if len(_INIT_CACHE) >= 32:  # fleet configs can have 9+ groups
                _INIT_CACHE.pop(next(iter(_INIT_CACHE)))
            _INIT_CACHE[init_key] = init_snapshot.cpu()
        if cv_mode in ("cross_val_only", "full_build"):
            # the final fit is independent of the fold fits (fresh init
            # either way): on GPU it overlaps the concurrent fold packs
            # on its own stream. The thread gate must IMPLY _fit_folds'
            # own concurrency gate (group*n_folds <= MAX): if the fold
            # loop were to fall back to its sequential path it would
            # train the SAME shared `pack` the final-fit thread is
            # using, corrupting both. group*(n_folds+1) <= MAX counts
            # every model live at once (folds + final) and implies the
            # fold gate.
            n_folds = self._n_cv_folds(evaluation)
            final_thread = None
            if (
                cv_mode == "full_build"
                and self.device != "cpu"
                and torch.cuda.is_available()
                and len(group) * (n_folds + 1) <= self._fold_cap(spec)
            ):
                import threading

                final_thread = threading.Thread(
                    target=final_fit, args=(torch.cuda.Stream(),)
                )
                final_thread.start()
            t0 = time.time()
            self._cross_validate_group(
                group, Xraw_list, y_list, spec, fit_args, pack, init_snapshot
            )
            cv_duration = time.time() - t0
            self._phase("cv_wall", cv_duration)
            if final_thread is not None:
                t_seg = time.time()
                final_thread.join()
                torch.cuda.synchronize()
                self._phase("final_fit_join", time.time() - t_seg)
            if cv_mode == "cross_val_only":
                for p in group:
                    self._finalize(p, None, cv_duration, final=False)
                return

        # final full fit (already done concurrently when possible)
        if "history" not in fit_state:
            t_seg = time.time()
            final_fit(None)
            self._phase("final_fit_serial", time.time() - t_seg)
        history = fit_state["history"]
        train_duration = fit_state["duration"]
        self._phase("final_fit_gpu(overlapped)", train_duration)

        # per-machine adoption + detector finalization + save — ASYNC:
        # one D2H of the flat parameter buffer then pure CPU work
        # (numpy slicing, metadata assembly, pickle+json dumps), all
        # independent of the NEXT group's GPU fits; runs on the
        # builder-wide pool and is joined at the end of build_all.
        n_rows = len(Xraw_list[0])
        offset = n_rows - pack._n_samples(n_rows)
        n_feat = [x.shape[1] for x in fit_state["Xt_list"]]
        n_feat_out = [y.shape[1] for y in y_list]

        def adopt_and_save(group=group, pack=pack, history=history,
                           train_duration=train_duration,
                           cv_duration=cv_duration, spec=spec):
            t0 = time.time()
            all_states = pack.states_for_all_models()
            for g_idx, p in enumerate(group):
                hist = {
                    k: [float(ep[g_idx]) for ep in v]
                    for k, v in history.items()
                }
                p.keras_est.adopt_pack_result(
                    spec,
                    all_states[g_idx],
                    hist,
                    n_features=n_feat[g_idx],
                    n_features_out=n_feat_out[g_idx],
                )
                if p.detector is not None:
                    p.detector.scaler.fit(p.y)
                self._finalize(p, offset, cv_duration, final=True,
                               train_duration=train_duration / len(group))
                if self.save_models and self.output_dir:
                    out = os.path.join(self.output_dir, p.machine.name)
                    ModelBuilder._save_model(p.model, p.machine, out)
                    if self.model_register_dir:
                        disk_registry.write_key(
                            self.model_register_dir,
                            ModelBuilder(p.machine).cache_key,
                            out,
                        )
            pack.release_graphs()
            self._phase("adopt_save(overlapped)", time.time() - t0)

        # one future per GROUP: plans are only touched by this closure
        # until the join, so no cross-thread aliasing
        self._save_futures.append(
            (list(group), self._save_pool.submit(adopt_and_save))
        )
        logger.info(
            "Packed build of %d machines dispatched in %.2fs",
            len(group), time.time() - t0_all,
        )

    # fold fits are independent — on GPU they run concurrently on
    # separate streams with per-fold packs (the chip is underfilled by
    # one pack's sequence-scan kernels); bounded so concurrent BPTT
    # caches stay well inside HBM.
    # 4 concurrent fits x 128-model packs ~ 80 GB of transient BPTT
    # caches at lookback 144 — comfortable in 288 GB (fold packs run
    # without graph capture, which halves retention)
    MAX_CONCURRENT_FOLD_MODELS = 640       # LSTM packs (BPTT caches)
    MAX_CONCURRENT_FOLD_MODELS_DENSE = 8192  # dense models are ~100x lighter

    def _fold_cap(self, spec) -> int:
        return (
            self.MAX_CONCURRENT_FOLD_MODELS
            if spec.model_type == "lstm"
            else self.MAX_CONCURRENT_FOLD_MODELS_DENSE
        )

    @staticmethod
    def _n_cv_folds(evaluation: Dict[str, Any]) -> int:
        """Fold count of the evaluation's CV splitter (for concurrency
        gating before the splitter actually runs)."""
        split_def = evaluation.get(
            "cv", {"sklearn.model_selection.TimeSeriesSplit": {"n_splits": 3}}
        )
        try:
            return int(serializer.from_definition(split_def).get_n_splits())
        except Exception:
            return 3

    @staticmethod
    def _fold_transform(group, Xraw_list, train_idx, test_idx):
        """Per-fold pre-step fitting (reference semantics: sklearn
        cross_validate clones the pipeline per fold, so scalers see
        only the train slice and transform the test slice). Returns
        (train arrays, test arrays) per machine."""
        Xtr, Xte = [], []
        for p, xr in zip(group, Xraw_list):
            xt, xv = xr[train_idx], xr[test_idx]
            for _, step in p.pre_steps:
                st = sk_clone(step)
                xt = st.fit_transform(xt)
                xv = st.transform(xv)
            Xtr.append(np.asarray(xt, dtype=np.float32))
            Xte.append(np.asarray(xv, dtype=np.float32))
        return Xtr, Xte

    def _device_fold_thresholds(
        self, group, fold_pack, Yd_train, y_list, test_idx, preds_t
    ):
        """Batched DEVICE computation of the DiffBased per-fold
        thresholds (kernels K10/K11 of SURVEY.md §2.3) while the fold
        predictions are still resident: per-machine MinMaxScaler from
        the train y, scaled-MSE/MAE residuals, and
        rolling(w).min().max() over the whole pack in three kernel
        launches — replacing the per-machine pandas/scipy loop of the
        scoring phase. Returns None on CPU or when no plain DiffBased
        detector is in the group (KFCV keeps its reassembled-series
        CPU path)."""
        if fold_pack.device.type != "cuda":
            return None
        wants = [
            g_idx for g_idx, p in enumerate(group)
            if p.detector is not None
            and not isinstance(p.detector, DiffBasedKFCVAnomalyDetector)
        ]
        if not wants:
            return None
        try:
            with torch.no_grad():
                G = len(group)
                n_out = preds_t.shape[1]
                pred = preds_t.float()
                # y stays fp32 on device (the bf16 compute mirror would
                # round the residual statistics ~0.4% vs the CPU path)
                Yte = torch.from_numpy(
                    np.stack([y[test_idx] for y in y_list])
                ).to(fold_pack.device, torch.float32)[:, -n_out:, :]
                ytr = Yd_train.float()
                ymin = ytr.amin(dim=1)          # [G, F]
                yrange = ytr.amax(dim=1) - ymin
                # sklearn MinMaxScaler handle_zeros_in_scale: scale=1
                scale = torch.where(
                    yrange == 0, torch.ones_like(yrange), 1.0 / yrange
                ).unsqueeze(1)                   # [G, 1, F]
                diff_scaled = (pred - Yte) * scale
                scaled_mse = (diff_scaled ** 2).mean(dim=2)   # [G, n]
                mae = (Yte - pred).abs()                      # [G, n, F]
                F = mae.shape[2]
                mae_rows = mae.transpose(1, 2).reshape(G * F, n_out)
                agg = ops.trail_min_max(scaled_mse, 6).cpu().numpy()
                tag = (
                    ops.trail_min_max(mae_rows.contiguous(), 6)
                    .cpu().numpy().reshape(G, F)
                )
                windows = sorted(
                    {
                        int(group[g].detector.window)
                        for g in wants
                        if getattr(group[g].detector, "window", None)
                        is not None
                    }
                )
                smooth = {}
                for w in windows:
                    if n_out < w:
                        smooth[w] = (
                            np.full(G, np.nan),
                            np.full((G, F), np.nan),
                        )
                        continue
                    smooth[w] = (
                        ops.trail_min_max(scaled_mse, w).cpu().numpy(),
                        ops.trail_min_max(mae_rows.contiguous(), w)
                        .cpu().numpy().reshape(G, F),
                    )
            return {"agg": agg, "tag": tag, "smooth": smooth}
        except Exception:
            logger.warning(
                "device threshold path failed; falling back to CPU",
                exc_info=True,
            )
            return None

    def _fit_folds(
        self, folds, group, Xraw_list, y_list, spec, fit_args, pack,
        init_snapshot,
    ):
        import threading

        def run_fold(fold_pack, train_idx, test_idx, out, fold_i, stream):
            try:
                ctx = (
                    torch.cuda.stream(stream)
                    if stream is not None
                    else _nullcontext()
                )
                with ctx:
                    t0 = time.time()
                    Xtr, Xte = self._fold_transform(
                        group, Xraw_list, train_idx, test_idx
                    )
                    Xd = self._stack(Xtr, fold_pack)
                    Yd = self._stack([y[train_idx] for y in y_list], fold_pack)
                    fold_pack.fit(Xd, Yd, **_engine_fit_args(fit_args))
                    t_fit = time.time() - t0
                    t0 = time.time()
                    Xtest = self._stack(Xte, fold_pack)
                    with torch.no_grad():
                        preds_t = fold_pack.predict(Xtest)
                        dev_thr = self._device_fold_thresholds(
                            group, fold_pack, Yd, y_list, test_idx, preds_t
                        )
                        preds = preds_t.float().cpu().numpy()
                    out[fold_i] = (preds, t_fit, time.time() - t0, dev_thr)
            except Exception as e:  # surface via the caller
                out[fold_i] = e

        concurrent = (
            self.device != "cpu"
            and torch.cuda.is_available()
            and len(group) * len(folds) <= self._fold_cap(spec)
        )
        out: Dict[int, Any] = {}
        if concurrent:
            threads = []
            for fold_i, (train_idx, test_idx) in enumerate(folds):
                fold_pack = self._make_pack(spec, group,
                                            init_p32=init_snapshot)
                # no graph capture in concurrent fold packs: capture is
                # process-global and the folds are throughput-bound on
                # real kernels anyway
                fold_pack._graph_enabled = False
                stream = torch.cuda.Stream()
                threads.append(
                    threading.Thread(
                        target=run_fold,
                        args=(fold_pack, train_idx, test_idx, out, fold_i,
                              stream),
                    )
                )
            for t in threads:
                t.start()
            for t in threads:
                t.join()
            torch.cuda.synchronize()
        else:
            for fold_i, (train_idx, test_idx) in enumerate(folds):
                self._reset_pack(pack, init_snapshot)
                run_fold(pack, train_idx, test_idx, out, fold_i, None)
        for fold_i, res in out.items():
            if isinstance(res, Exception):
                raise res
        return out

    @staticmethod
    def _reset_pack(pack, init_snapshot: torch.Tensor):
        pack.store.p32.copy_(init_snapshot)
        pack.store.reset_adam()
        pack.store.sync_lp()

    def _cross_validate_group(
        self,
        group: List[MachinePlan],
        Xraw_list: List[np.ndarray],
        y_list: List[np.ndarray],
        spec: ModelSpec,
        fit_args: Dict[str, Any],
        pack,
        init_snapshot: torch.Tensor,
    ):
        """Packed equivalent of sklearn cross_validate +
        DiffBasedAnomalyDetector.cross_validate: per fold, fresh packs
        fit on the train slice, score the test slice per machine,
        accumulate rolling-min-max thresholds (reference
        diff.py:176-266, build_model.py:244-289)."""
        evaluation = group[0].machine.evaluation
        split_def = evaluation.get(
            "cv", {"sklearn.model_selection.TimeSeriesSplit": {"n_splits": 3}}
        )
        split_obj = serializer.from_definition(split_def)
        metrics_list = ModelBuilder.metrics_from_list(evaluation.get("metrics"))
        scoring_scaler_def = evaluation.get("scoring_scaler")

        N = len(Xraw_list[0])
        X_index = group[0].X.index
        folds = list(split_obj.split(np.zeros((N, 1))))

        # per-machine scoring scalers fitted on full y
        scoring_scalers = []
        for p, y_arr in zip(group, y_list):
            if scoring_scaler_def:
                sd = scoring_scaler_def
                if isinstance(sd, str):
                    sd = {sd: {}}
                sc = serializer.from_definition(sd)
                sc.fit(y_arr)
            else:
                sc = None
            scoring_scalers.append(sc)

        per_machine_scores: List[Dict[str, List[float]]] = [
            {} for _ in group
        ]
        # hoist the per-(metric, tag) score keys: building them per
        # (machine, fold, metric, tag) is ~600k f-strings at the full
        # 1000-machine config
        metric_names = [
            m.__name__.replace("_", "-") for m in metrics_list
        ]
        key_cache: List[Dict[str, List[str]]] = []
        for p in group:
            tags = [t.name for t in p.machine.dataset.target_tag_list]
            key_cache.append({
                mname: [
                    f'{mname}-{tag.replace(" ", "-")}' for tag in tags
                ]
                for mname in metric_names
            })

        t_seg = time.time()
        fold_preds = self._fit_folds(
            folds, group, Xraw_list, y_list, spec, fit_args, pack,
            init_snapshot,
        )
        self._phase("cv_fold_fits", time.time() - t_seg)
        t_seg = time.time()

        # KFCV detectors: thresholds are quantiles of the smoothed
        # validation metric over the FULL reassembled prediction series
        # (reference diff.py:566-635) — accumulate per-machine y_pred
        # across folds here, finalize after the fold loop.
        kfcv_pred = {
            g_idx: np.zeros_like(y_list[g_idx])
            for g_idx, p in enumerate(group)
            if isinstance(p.detector, DiffBasedKFCVAnomalyDetector)
        }
        kfcv_mse = {
            g_idx: np.full(len(y_list[g_idx]), np.nan, dtype=np.float64)
            for g_idx in kfcv_pred
        }

        # batched scoring-scaler transform (K13 vectorization across
        # the pack): all MinMaxScaler-style scalers apply as one affine
        # op on the stacked [G, n, F] arrays; any other scaler type
        # keeps the per-machine path.
        batch_affine = all(
            sc is not None and hasattr(sc, "scale_") and hasattr(sc, "min_")
            for sc in scoring_scalers
        ) and len(group) > 1
        if batch_affine:
            sc_scale = np.stack(
                [np.asarray(sc.scale_) for sc in scoring_scalers]
            )[:, None, :]
            sc_min = np.stack(
                [np.asarray(sc.min_) for sc in scoring_scalers]
            )[:, None, :]

        for fold_i, (train_idx, test_idx) in enumerate(folds):
            t_f0 = time.time()
            preds, t_fit, t_pred, dev_thr = fold_preds[fold_i]
            t_s0 = time.time()

            yt_b = yp_b = None
            if batch_affine:
                n_out = preds.shape[1]
                y_true_b = np.stack(
                    [y[test_idx][-n_out:] for y in y_list]
                )
                yt_b = y_true_b * sc_scale + sc_min
                yp_b = preds * sc_scale + sc_min
                metric_cache = {
                    metric.__name__: _metric_all_tags_batched(
                        metric, yt_b, yp_b
                    )
                    for metric in metrics_list
                }

            for g_idx, p in enumerate(group):
                y_true_full = y_list[g_idx][test_idx]
                y_pred = preds[g_idx]
                y_true = y_true_full[-len(y_pred):]
                sc = scoring_scalers[g_idx]
                if batch_affine:
                    yt, yp = yt_b[g_idx], yp_b[g_idx]
                else:
                    yt = sc.transform(y_true) if sc is not None else y_true
                    yp = sc.transform(y_pred) if sc is not None else y_pred
                for metric, mname in zip(metrics_list, metric_names):
                    if batch_affine:
                        cached = metric_cache.get(metric.__name__)
                        per_tag, agg = (
                            (cached[0][g_idx], cached[1][g_idx])
                            if cached is not None
                            else _metric_all_tags(metric, yt, yp)
                        )
                    else:
                        per_tag, agg = _metric_all_tags(metric, yt, yp)
                    keys = key_cache[g_idx][mname]
                    scores_g = per_machine_scores[g_idx]
                    vals = per_tag.tolist()
                    for col, key in enumerate(keys):
                        scores_g.setdefault(key, []).append(
                            float(vals[col])
                        )
                    scores_g.setdefault(mname, []).append(float(agg))

                # DiffBased thresholds: fold scaler fitted on y_train
                if isinstance(p.detector, DiffBasedKFCVAnomalyDetector):
                    det = p.detector
                    fold_scaler = sk_clone(det.scaler)
                    fold_scaler.fit(y_list[g_idx][train_idx])
                    pred_full = preds[g_idx]
                    # KFold test slices are contiguous index sets; the
                    # model output covers the whole slice (offset 0 for
                    # dense; windows handled below for LSTM)
                    off = len(test_idx) - len(pred_full)
                    rows = test_idx[off:]
                    kfcv_pred[g_idx][rows] = pred_full
                    kfcv_mse[g_idx][rows] = (
                        (fold_scaler.transform(pred_full)
                         - fold_scaler.transform(y_list[g_idx][rows])) ** 2
                    ).mean(axis=1)
                elif p.detector is not None:
                    det = p.detector
                    if dev_thr is not None:
                        # K10 device path: thresholds came back with the
                        # fold predictions (one batched kernel per stat)
                        agg_thr = float(dev_thr["agg"][g_idx])
                        tag_thr = pd.Series(
                            dev_thr["tag"][g_idx], name=f"fold-{fold_i}"
                        )
                    else:
                        fold_scaler = sk_clone(det.scaler)
                        fold_scaler.fit(y_list[g_idx][train_idx])
                        scaled_mse = (
                            (fold_scaler.transform(y_pred)
                             - fold_scaler.transform(y_true)) ** 2
                        ).mean(axis=1)
                        mae = np.abs(np.asarray(y_true) - np.asarray(y_pred))
                        agg_thr = _trail_min_max(scaled_mse, 6)
                        tag_thr = pd.Series(
                            _trail_min_max(mae, 6), name=f"fold-{fold_i}"
                        )
                    if not hasattr(det, "aggregate_thresholds_per_fold_"):
                        det.aggregate_thresholds_per_fold_ = {}
                        det._fold_tag_rows = []
                        det.smooth_aggregate_thresholds_per_fold_ = {}
                        det._fold_smooth_tag_rows = []
                    det.aggregate_thresholds_per_fold_[f"fold-{fold_i}"] = agg_thr
                    det._fold_tag_rows.append(tag_thr)
                    det.aggregate_threshold_ = agg_thr
                    det.feature_thresholds_ = tag_thr
                    # smoothing window is per machine, not per group
                    window = getattr(det, "window", None)
                    if window is not None:
                        if dev_thr is not None:
                            s_agg_v, s_tag_v = dev_thr["smooth"][int(window)]
                            s_agg = float(s_agg_v[g_idx])
                            s_tag = pd.Series(
                                s_tag_v[g_idx], name=f"fold-{fold_i}"
                            )
                        else:
                            s_agg = _trail_min_max(scaled_mse, window)
                            s_tag = pd.Series(
                                _trail_min_max(mae, window),
                                name=f"fold-{fold_i}",
                            )
                        det.smooth_aggregate_thresholds_per_fold_[
                            f"fold-{fold_i}"
                        ] = s_agg
                        det._fold_smooth_tag_rows.append(s_tag)
                        det.smooth_aggregate_threshold_ = s_agg
                        det.smooth_feature_thresholds_ = s_tag
            logger.info(
                "  fold %d: fit %.2fs, predict %.2fs, score+thresholds %.2fs",
                fold_i, t_fit, t_pred, time.time() - t_s0,
            )

        # one concat per machine instead of one per (machine, fold):
        # growing a DataFrame row-by-row re-allocates every time
        for p in group:
            det = p.detector
            rows = getattr(det, "_fold_tag_rows", None)
            if rows is not None:
                det.feature_thresholds_per_fold_ = pd.DataFrame(rows)
                del det._fold_tag_rows
                srows = det._fold_smooth_tag_rows
                det.smooth_feature_thresholds_per_fold_ = (
                    pd.DataFrame(srows) if srows else pd.DataFrame()
                )
                del det._fold_smooth_tag_rows
        self._phase("cv_score_thresholds", time.time() - t_seg)

        # finalize KFCV thresholds over the reassembled series
        for g_idx in kfcv_pred:
            det = group[g_idx].detector
            y_full = y_list[g_idx]
            mse_series = pd.Series(kfcv_mse[g_idx])
            det.aggregate_threshold_ = det._calculate_threshold(mse_series)
            abs_err = pd.DataFrame(np.abs(y_full - kfcv_pred[g_idx]))
            det.feature_thresholds_ = det._calculate_threshold(abs_err)

        # assemble fold stats + split metadata per machine
        for g_idx, p in enumerate(group):
            scores: Dict[str, Any] = {}
            for key, vals in per_machine_scores[g_idx].items():
                arr = np.asarray(vals)
                entry = {
                    "fold-mean": arr.mean(),
                    "fold-std": arr.std(),
                    "fold-max": arr.max(),
                    "fold-min": arr.min(),
                }
                entry.update(
                    {f"fold-{i + 1}": float(v) for i, v in enumerate(vals)}
                )
                scores[key] = entry
            p.scores = scores
            split_metadata: Dict[str, Any] = {}
            for i, (train_ind, test_ind) in enumerate(folds):
                split_metadata.update(
                    {
                        f"fold-{i + 1}-train-start": X_index[train_ind[0]],
                        f"fold-{i + 1}-train-end": X_index[train_ind[-1]],
                        f"fold-{i + 1}-test-start": X_index[test_ind[0]],
                        f"fold-{i + 1}-test-end": X_index[test_ind[-1]],
                        f"fold-{i + 1}-n-train": len(train_ind),
                        f"fold-{i + 1}-n-test": len(test_ind),
                    }
                )
            p.splits = split_metadata

    def _finalize(
        self,
        p: MachinePlan,
        offset: Optional[int],
        cv_duration: Optional[float],
        final: bool,
        train_duration: Optional[float] = None,
    ):
        model_meta = (
            ModelBuilder._extract_metadata_from_model(p.model) if final else {}
        )
        p.machine.metadata.build_metadata = BuildMetadata(
            model=ModelBuildMetadata(
                model_offset=int(offset) if offset is not None else 0,
                model_creation_date=(
                    str(datetime.datetime.now(datetime.timezone.utc).astimezone())
                    if final
                    else None
                ),
                model_builder_version=gordo_amd.__version__,
                model_training_duration_sec=train_duration,
                cross_validation=CrossValidationMetaData(
                    cv_duration_sec=cv_duration,
                    scores=p.scores,
                    splits=p.splits,
                ),
                model_meta=model_meta,
            ),
            dataset=DatasetBuildMetadata(
                query_duration_sec=p.query_duration,
                dataset_meta=p.dataset_meta,
            ),
        )


def _metric_all_tags_batched(metric, yt: np.ndarray, yp: np.ndarray):
    """Batched [G, n, F] version of _metric_all_tags for the four
    default metrics; returns (per_tag [G, F], agg [G]) or None for
    unknown metrics (callers fall back per machine)."""
    name = getattr(metric, "__name__", "")
    diff = yp - yt
    if name == "mean_squared_error":
        per = (diff ** 2).mean(axis=1)
        return per, per.mean(axis=1)
    if name == "mean_absolute_error":
        per = np.abs(diff).mean(axis=1)
        return per, per.mean(axis=1)
    if name == "r2_score":
        ss_res = (diff ** 2).sum(axis=1)
        ss_tot = ((yt - yt.mean(axis=1, keepdims=True)) ** 2).sum(axis=1)
        with np.errstate(divide="ignore", invalid="ignore"):
            per = np.where(
                ss_tot > 0,
                1.0 - ss_res / ss_tot,
                np.where(ss_res == 0, 1.0, 0.0),
            )
        return per, per.mean(axis=1)
    if name == "explained_variance_score":
        var_res = diff.var(axis=1)
        var_y = yt.var(axis=1)
        with np.errstate(divide="ignore", invalid="ignore"):
            per = np.where(
                var_y > 0,
                1.0 - var_res / var_y,
                np.where(var_res == 0, 1.0, 0.0),
            )
        return per, per.mean(axis=1)
    return None


def _metric_all_tags(metric, yt: np.ndarray, yp: np.ndarray):
    """Vectorized per-tag + aggregate computation of the four default
    sklearn metrics (a per-tag python-call loop costs ~60 ms per
    (machine, fold) — 20+ s per 125-machine build step). The aggregate
    equals sklearn's multioutput='uniform_average'. Unknown metrics
    fall back to per-column calls."""
    name = getattr(metric, "__name__", "")
    diff = yp - yt
    if name == "mean_squared_error":
        per = (diff ** 2).mean(axis=0)
        return per, per.mean()
    if name == "mean_absolute_error":
        per = np.abs(diff).mean(axis=0)
        return per, per.mean()
    if name == "r2_score":
        ss_res = (diff ** 2).sum(axis=0)
        ss_tot = ((yt - yt.mean(axis=0)) ** 2).sum(axis=0)
        with np.errstate(divide="ignore", invalid="ignore"):
            # sklearn convention for a zero-variance target column:
            # perfect prediction scores 1.0, anything else 0.0
            per = np.where(
                ss_tot > 0,
                1.0 - ss_res / ss_tot,
                np.where(ss_res == 0, 1.0, 0.0),
            )
        return per, per.mean()
    if name == "explained_variance_score":
        var_res = diff.var(axis=0)
        var_y = yt.var(axis=0)
        with np.errstate(divide="ignore", invalid="ignore"):
            per = np.where(
                var_y > 0,
                1.0 - var_res / var_y,
                np.where(var_res == 0, 1.0, 0.0),
            )
        return per, per.mean()
    per = np.array([metric(yt[:, c], yp[:, c]) for c in range(yt.shape[1])])
    return per, metric(yt, yp)


def _engine_fit_args(fit_args: Dict[str, Any]) -> Dict[str, Any]:
    out: Dict[str, Any] = {}
    for k in ("epochs", "batch_size", "shuffle", "verbose"):
        if k in fit_args:
            out[k] = fit_args[k]
    if fit_args.get("validation_split"):
        out["validation_split"] = float(fit_args["validation_split"])
    # identical callbacks are part of the group key (fit_args), so the
    # lockstep all-models-stalled semantics of pack.fit applies per group
    es = _parse_early_stopping(fit_args.get("callbacks"))
    if es is not None:
        out["early_stopping"] = es
    return out
