"""
Packed many-model trainers.

The MI355X-native answer to the reference's pod-per-model fan-out
(SURVEY.md §2.4): G same-architecture models train in lockstep as ONE
set of grouped tensors — weights ``[G, in, out]``, activations
``[G, B, F]`` — so every layer of every model in the pack is a single
grouped MFMA GEMM launch instead of thousands of tiny ones. Per-model
data, per-model losses, per-model Adam state; one flat fp32 parameter
buffer (+ bf16 compute mirror on GPU) per pack so the optimizer is one
fused kernel over the whole fleet shard.

Training math matches Keras defaults (glorot-uniform kernels,
orthogonal LSTM recurrent kernels, unit forget-gate bias, Adam with
bias correction, MSE loss) — reference factories:
gordo/machine/model/factories/feedforward_autoencoder.py,
lstm_autoencoder.py.
"""
from __future__ import annotations

import logging
import math
import os
from typing import Any, Dict, List, Optional, Tuple

import numpy as np
import torch

from .. import ops
from .spec import ModelSpec

logger = logging.getLogger(__name__)


def _compute_dtype(device: torch.device) -> torch.dtype:
    return torch.bfloat16 if device.type == "cuda" else torch.float32


def _pad8(n: int) -> int:
    return (int(n) + 7) & ~7


def pad_enabled(device) -> bool:
    """Feature/hidden dims are zero-padded to multiples of 8 on GPU
    (GORDO_PAD8=0 disables): bf16x8 = 16 bytes unlocks vectorized
    global staging loads in every grouped GEMM (guide G13 — scalar
    bf16 staging is a 2-2.5x tax), and padded hidden units are
    PROVABLY inert: zero weights + zero bias give g=tanh(0)=0 so
    c=f*0+i*0=0 and h=o*tanh(0)=0 forever; all their gradients are
    exactly 0 (dc=dh=0 chains), so Adam keeps their weights at 0.
    Only the MSE mean needs the REAL element count (ops.mse_bwd
    real_n). CPU packs stay unpadded (they are the fp32 oracle)."""
    return (
        torch.device(device).type == "cuda"
        and os.environ.get("GORDO_PAD8", "1") != "0"
    )


def _glorot_uniform(shape, gen):
    fan_in, fan_out = shape[-2], shape[-1]
    limit = math.sqrt(6.0 / (fan_in + fan_out))
    return (torch.rand(shape, generator=gen, dtype=torch.float32) * 2 - 1) * limit


def _orthogonal(shape, gen):
    rows, cols = shape[-2], shape[-1]
    a = torch.randn((max(rows, cols), min(rows, cols)), generator=gen)
    q, r = torch.linalg.qr(a)
    q = q * torch.sign(torch.diagonal(r))
    if rows < cols:
        q = q.T
    return q[:rows, :cols].contiguous()


class _ParamStore:
    """One flat fp32 master buffer + grads + Adam state (+ bf16 compute
    mirror on GPU), carved into named [G, ...] views."""

    def __init__(self, device: torch.device):
        self.device = device
        self.compute_dtype = _compute_dtype(device)
        self._shapes: List[Tuple[str, Tuple[int, ...]]] = []
        self.logical: Dict[str, Tuple[int, ...]] = {}
        self._offsets: Dict[str, Tuple[int, int]] = {}
        self._total = 0
        self.views: Dict[str, torch.Tensor] = {}
        self.cviews: Dict[str, torch.Tensor] = {}
        self.gviews: Dict[str, torch.Tensor] = {}
        self.step_count = 0

    def declare(self, name: str, shape: Tuple[int, ...],
                logical: Optional[Tuple[int, ...]] = None):
        n = int(np.prod(shape))
        self._offsets[name] = (self._total, n)
        self._shapes.append((name, shape))
        self.logical[name] = tuple(logical) if logical else tuple(shape)
        self._total += n

    def allocate(self):
        dev = self.device
        self.p32 = torch.zeros(self._total, dtype=torch.float32, device=dev)
        self.g32 = torch.zeros(self._total, dtype=torch.float32, device=dev)
        self.m = torch.zeros(self._total, dtype=torch.float32, device=dev)
        self.v = torch.zeros(self._total, dtype=torch.float32, device=dev)
        self.plp = (
            torch.zeros(self._total, dtype=self.compute_dtype, device=dev)
            if self.compute_dtype != torch.float32
            else None
        )
        # device-resident Adam step counter (hipGraph-capturable)
        self.step_buf = (
            torch.zeros(1, dtype=torch.int32, device=dev)
            if dev.type == "cuda"
            else None
        )
        for name, shape in self._shapes:
            off, n = self._offsets[name]
            self.views[name] = self.p32[off : off + n].view(*shape)
            self.gviews[name] = self.g32[off : off + n].view(*shape)
            self.cviews[name] = (
                self.plp[off : off + n].view(*shape)
                if self.plp is not None
                else self.views[name]
            )

    def sync_lp(self):
        if self.plp is not None:
            self.plp.copy_(self.p32)

    def zero_grad(self):
        self.g32.zero_()

    def adam_step(self, lr, beta1, beta2, eps):
        self.step_count += 1
        ops.adam_step(
            self.p32, self.g32, self.m, self.v,
            lr, beta1, beta2, eps, self.step_count, self.plp,
            step_buf=self.step_buf,
        )

    def reset_adam(self):
        self.m.zero_()
        self.v.zero_()
        self.step_count = 0
        if self.step_buf is not None:
            self.step_buf.zero_()


class BasePack:
    def __init__(self, spec: ModelSpec, G: int, device=None, seeds=None,
                 init_p32: Optional[torch.Tensor] = None):
        self.spec = spec
        self.G = G
        self.device = torch.device(
            device if device is not None else
            ("cuda" if torch.cuda.is_available() else "cpu")
        )
        self.store = _ParamStore(self.device)
        self.pad8 = pad_enabled(self.device)
        self._pad = _pad8 if self.pad8 else (lambda n: int(n))
        # per-param custom logical<->physical converters (LSTM gate
        # interleave); default is plain dim slicing by store.logical
        self._extractors: Dict[str, Any] = {}
        self.seeds = list(seeds) if seeds is not None else list(range(G))
        assert len(self.seeds) == G
        self._declare_params()
        self.store.allocate()
        if init_p32 is not None:
            # clone-init: copy a sibling pack's flat parameter snapshot
            # (per-model glorot + orthogonal QR costs ~1 s per pack)
            self.store.p32.copy_(init_p32)
        else:
            self._init_weights()
        self.store.sync_lp()

    # -- interface -------------------------------------------------------
    def _declare_params(self):
        raise NotImplementedError

    def _init_weights(self):
        raise NotImplementedError

    def predict(self, X: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def train_batch(self, Xb, Tb) -> torch.Tensor:
        raise NotImplementedError

    def eval_batch(self, Xb, Tb) -> torch.Tensor:
        """Forward-only per-model loss for a gathered batch (the
        validation pass of fit)."""
        raise NotImplementedError

    # -- common ----------------------------------------------------------
    @property
    def compute_dtype(self):
        return self.store.compute_dtype

    def _to_compute(self, t: torch.Tensor) -> torch.Tensor:
        return t.to(device=self.device, dtype=self.compute_dtype)

    def _pad_io(self, Xb, Tb):
        """Defensive padding for DIRECT train_batch/eval_batch calls
        (fit() pads the whole series up front; callers like profiling
        harnesses or graph capture may pass logical-width tensors)."""
        if self.pad8:
            if Xb.shape[-1] == self._logical_in:
                Xb = self._pad_features(Xb, self._logical_in)
            if Tb is not None and Tb.shape[-1] == self._logical_out:
                Tb = self._pad_features(Tb, self._logical_out)
        return Xb, Tb

    def _pad_features(self, t: torch.Tensor, logical_f: int) -> torch.Tensor:
        """Zero-pad the last (feature) dim to the physical width."""
        pf = self._pad(logical_f)
        if t.shape[-1] == pf:
            return t
        assert t.shape[-1] == logical_f, (t.shape, logical_f)
        return torch.nn.functional.pad(t, (0, pf - logical_f))

    def param_names(self) -> List[str]:
        return [name for name, _ in self.store._shapes]

    def _extract_logical(self, name: str, arr: np.ndarray) -> np.ndarray:
        """Physical [per-model] array -> logical (serialized) layout."""
        ex = self._extractors.get(name)
        if ex is not None:
            return ex[0](arr)
        logical = self.store.logical[name][1:]
        if arr.shape == tuple(logical):
            return arr.copy()
        return arr[tuple(slice(0, d) for d in logical)].copy()

    def _insert_logical(self, name: str, view: torch.Tensor,
                        arr: np.ndarray):
        """Write a logical array into the physical [per-model] view
        (pad regions stay zero)."""
        ex = self._extractors.get(name)
        t = torch.from_numpy(np.ascontiguousarray(arr))
        if ex is not None:
            ex[1](view, t)
            return
        logical = self.store.logical[name][1:]
        if tuple(view.shape) == tuple(logical):
            view.copy_(t)
            return
        view.zero_()
        view[tuple(slice(0, d) for d in logical)].copy_(t)

    def state_for_model(self, g: int) -> Dict[str, np.ndarray]:
        return {
            name: self._extract_logical(
                name, self.store.views[name][g].detach().cpu().numpy()
            )
            for name in self.param_names()
        }

    def states_for_all_models(self) -> List[Dict[str, np.ndarray]]:
        """Per-model weight dicts via ONE device-to-host transfer of the
        flat parameter buffer (per-model-per-param .cpu() slices cost
        ~1 s per 62-model pack)."""
        flat = self.store.p32.detach().cpu().numpy()
        out: List[Dict[str, np.ndarray]] = [dict() for _ in range(self.G)]
        for name, shape in self.store._shapes:
            off, n = self.store._offsets[name]
            arr = flat[off : off + n].reshape(shape)
            for g in range(self.G):
                out[g][name] = self._extract_logical(name, arr[g])
        return out

    def load_model_state(self, g: int, state: Dict[str, np.ndarray]):
        for name, arr in state.items():
            self._insert_logical(
                name, self.store.views[name][g], np.asarray(arr)
            )
        self.store.sync_lp()

    def optimizer_state_for_model(self, g: int) -> Dict[str, np.ndarray]:
        out = {}
        for name in self.param_names():
            off, n = self.store._offsets[name]
            shape = self.store.views[name].shape
            out["m:" + name] = (
                self.store.m[off : off + n].view(*shape)[g].cpu().numpy().copy()
            )
            out["v:" + name] = (
                self.store.v[off : off + n].view(*shape)[g].cpu().numpy().copy()
            )
        return out

    # ---- the epoch loop ------------------------------------------------
    # hipGraph capture of the steady-state train_batch: the whole
    # fwd+bwd+Adam kernel sequence replays as ONE graph launch per
    # full-size batch (shapes static; the shuffled gather runs eagerly
    # into static buffers; the ragged last batch stays eager).
    # OPT-IN (GORDO_HIPGRAPH=1): measured perf-neutral on this workload
    # (it is GPU-bound, not launch-bound), and torch-ROCm's CUDAGraph
    # destructor intermittently aborts the process with "The graph
    # should be registered to the state" (HIPGeneratorImpl.cpp:158)
    # when graphs are garbage-collected in long multi-pack processes.
    _graph_enabled = True  # env re-checked per step; instance sets
    # this False permanently after a capture failure
    # hip stream-capture mode is process-global: captures from two
    # threads at once crash the process. One capture at a time.
    _graph_capture_lock = __import__("threading").Lock()

    def _graph_train_step(self, Xb: torch.Tensor, Tb: torch.Tensor):
        """Replay (capturing on first use) the train_batch graph for
        this batch shape. Returns the live loss tensor, or None when
        capture is unavailable for this configuration."""
        if not (
            self._graph_enabled
            and os.environ.get("GORDO_HIPGRAPH", "0") == "1"
            and self.device.type == "cuda"
        ):
            # _graph_enabled False (set per-instance after a capture
            # failure) wins; otherwise the env is re-read so flipping
            # GORDO_HIPGRAPH at runtime takes effect
            return None
        key = (tuple(Xb.shape), tuple(Tb.shape))
        cached = getattr(self, "_graph_cache", None)
        if cached is None:
            cached = self._graph_cache = {}
        entry = cached.get(key)
        if entry is None:
            if len(cached) >= 4:  # bound distinct shapes per pack
                return None
            try:
                # flush pending GC NOW: a garbage pack's CUDAGraph
                # destructor running inside the upcoming capture (or
                # inside another thread's capture) hits HIP's
                # generator-state unregister check and terminates the
                # process
                import gc

                gc.collect()
                sx = torch.empty_like(Xb)
                st = torch.empty_like(Tb)
                sx.copy_(Xb)
                st.copy_(Tb)
                # warmup + capture run real optimizer steps — snapshot
                # the training state and restore it afterwards so
                # capture does not perturb the training trajectory.
                store = self.store
                snap = (
                    store.p32.clone(), store.m.clone(), store.v.clone(),
                    store.step_buf.clone(), store.step_count,
                )
                with BasePack._graph_capture_lock:
                    s = torch.cuda.Stream()
                    s.wait_stream(torch.cuda.current_stream())
                    with torch.cuda.stream(s):
                        self.train_batch(sx, st)
                    torch.cuda.current_stream().wait_stream(s)
                    graph = torch.cuda.CUDAGraph()
                    # thread_local: other builder threads keep launching
                    # on their own streams while this one captures
                    with torch.cuda.graph(
                        graph, capture_error_mode="thread_local"
                    ):
                        loss = self.train_batch(sx, st)
                store.p32.copy_(snap[0])
                store.m.copy_(snap[1])
                store.v.copy_(snap[2])
                store.step_buf.copy_(snap[3])
                store.step_count = snap[4]
                store.sync_lp()
                entry = cached[key] = (graph, sx, st, loss)
            except Exception:
                logger.warning(
                    "hipGraph capture failed; falling back to eager",
                    exc_info=True,
                )
                self._graph_enabled = False
                return None
        graph, sx, st, loss = entry
        sx.copy_(Xb)
        st.copy_(Tb)
        graph.replay()
        self.store.step_count += 1  # device step_buf is the truth
        return loss

    def release_graphs(self):
        """Drop captured graphs deterministically (outside any capture)
        instead of at arbitrary GC time."""
        for attr in ("_graph_cache", "_pred_graph_cache"):
            cached = getattr(self, attr, None)
            if cached:
                with BasePack._graph_capture_lock:
                    cached.clear()

    def predict_captured(self, X: torch.Tensor) -> torch.Tensor:
        """hipGraph-replayed inference forward for static request shapes
        (the serving hot path: north-star "batched anomaly inference on
        hipGraph-captured steps"). Each distinct input shape captures
        once and replays thereafter — one graph launch instead of one
        host launch per layer. Opt-in via ``GORDO_SERVE_HIPGRAPH=1``
        (the torch-ROCm graph-destructor abort makes capture opt-in
        everywhere — ROADMAP); anything unexpected falls back to the
        eager predict. Thread-safe: replay + readout under a per-pack
        lock (the static buffers are shared)."""
        if not (
            os.environ.get("GORDO_SERVE_HIPGRAPH", "0") == "1"
            and self.device.type == "cuda"
            and self._graph_enabled
        ):
            return self.predict(X)
        Xc = self._to_compute(X)
        key = tuple(Xc.shape)
        lock = getattr(self, "_pred_graph_lock", None)
        if lock is None:
            lock = self._pred_graph_lock = __import__("threading").Lock()
        with lock:
            cached = getattr(self, "_pred_graph_cache", None)
            if cached is None:
                cached = self._pred_graph_cache = {}
            entry = cached.get(key)
            if entry is None:
                if len(cached) >= 8:  # bound distinct request shapes
                    return self.predict(Xc)
                try:
                    import gc

                    gc.collect()  # same destructor hazard as train capture
                    with BasePack._graph_capture_lock:
                        sx = Xc.clone()
                        s = torch.cuda.Stream()
                        s.wait_stream(torch.cuda.current_stream())
                        with torch.cuda.stream(s), torch.no_grad():
                            self.predict(sx)
                        torch.cuda.current_stream().wait_stream(s)
                        graph = torch.cuda.CUDAGraph()
                        with torch.cuda.graph(
                            graph, capture_error_mode="thread_local"
                        ), torch.no_grad():
                            out = self.predict(sx)
                    entry = cached[key] = (graph, sx, out)
                except Exception:
                    logger.warning(
                        "serving hipGraph capture failed; staying eager",
                        exc_info=True,
                    )
                    self._graph_enabled = False
                    return self.predict(Xc)
            graph, sx, out = entry
            sx.copy_(Xc)
            graph.replay()
            # clone before releasing the lock: the static out buffer is
            # overwritten by the next replay
            return out.clone()

    def fit(
        self,
        X: torch.Tensor,
        Y: torch.Tensor,
        epochs: int = 1,
        batch_size: int = 256,
        shuffle: bool = True,
        verbose: int = 0,
        early_stopping: Optional[Dict[str, Any]] = None,
        validation_split: float = 0.0,
        **_,
    ) -> Dict[str, list]:
        """
        Train the whole pack. X: [G, N, F_in] (already on device),
        Y: [G, N, F_out]. Returns a Keras-style history dict with
        per-model per-epoch values: {"loss": [[...G...] per epoch],
        "accuracy": ...}.
        """
        if self.pad8:
            X = self._pad_features(X, self._logical_in)
            Y = self._pad_features(Y, self._logical_out)
        G, N = X.shape[0], X.shape[1]
        adam = self.spec.adam_params
        history: Dict[str, list] = {"loss": [], "accuracy": []}
        n_all = self._n_samples(N)
        # keras validation_split semantics: the LAST fraction of samples
        # (in input order, before shuffling) is held out; history gains
        # val_loss/val_accuracy and EarlyStopping can monitor val_loss
        n_train = n_all
        if validation_split and validation_split > 0.0:
            n_train = int(n_all * (1.0 - float(validation_split)))
            n_train = max(1, min(n_train, n_all))
            if n_train < n_all:
                history["val_loss"] = []
                history["val_accuracy"] = []
        n_val = n_all - n_train
        # EarlyStopping (keras semantics, restore_best_weights=False):
        # training ends when EVERY model in the pack has gone `patience`
        # epochs without improving its train loss by `min_delta` (models
        # train in lockstep, so the group stops when all have stalled;
        # for G=1 this is exactly the keras behavior on `loss`).
        es_patience = es_min_delta = None
        if early_stopping is not None:
            es_patience = int(early_stopping.get("patience", 0))
            es_min_delta = float(early_stopping.get("min_delta", 0.0))
            es_best = np.full(G, np.inf)
            es_wait = np.zeros(G, dtype=np.int64)
        n_batches = max(1, math.ceil(n_train / batch_size))
        gens = [torch.Generator().manual_seed(int(s) & 0x7FFFFFFF) for s in self.seeds]
        for epoch in range(epochs):
            if shuffle:
                perm = torch.stack(
                    [torch.randperm(n_train, generator=g) for g in gens]
                ).to(self.device)
            else:
                perm = (
                    torch.arange(n_train, device=self.device)
                    .unsqueeze(0)
                    .expand(G, -1)
                )
            epoch_loss = torch.zeros(G, dtype=torch.float32, device=self.device)
            samples_seen = 0
            for b in range(n_batches):
                idx = perm[:, b * batch_size : (b + 1) * batch_size]
                if idx.shape[1] == 0:
                    continue
                Xb, Tb = self._gather_batch(X, Y, idx)
                loss = self._graph_train_step(Xb, Tb)
                if loss is None:
                    loss = self.train_batch(Xb, Tb)
                epoch_loss += loss * idx.shape[1]
                samples_seen += idx.shape[1]
            epoch_loss = (epoch_loss / max(samples_seen, 1)).cpu().tolist()
            history["loss"].append(epoch_loss)
            history["accuracy"].append([0.0] * G)
            if n_val > 0:
                val_loss = torch.zeros(G, dtype=torch.float32, device=self.device)
                vseen = 0
                for vs_ in range(n_train, n_all, batch_size):
                    vidx = (
                        torch.arange(
                            vs_, min(vs_ + batch_size, n_all), device=self.device
                        ).unsqueeze(0).expand(G, -1)
                    )
                    Xvb, Tvb = self._gather_batch(X, Y, vidx)
                    val_loss += self.eval_batch(Xvb, Tvb) * vidx.shape[1]
                    vseen += vidx.shape[1]
                val_losses = (val_loss / max(vseen, 1)).cpu().tolist()
                history["val_loss"].append(val_losses)
                history["val_accuracy"].append([0.0] * G)
            if verbose:
                logger.info(
                    "epoch %d/%d mean-loss=%.6g", epoch + 1, epochs,
                    float(np.mean(epoch_loss)),
                )
            if es_patience is not None:
                monitor_val = (
                    str((early_stopping or {}).get("monitor", "loss"))
                    .startswith("val") and n_val > 0
                )
                losses = np.asarray(
                    history["val_loss"][-1] if monitor_val else epoch_loss
                )
                improved = losses < (es_best - es_min_delta)
                es_best = np.minimum(es_best, losses)
                es_wait = np.where(improved, 0, es_wait + 1)
                if epoch > 0 and (es_wait >= es_patience).all():
                    if verbose:
                        logger.info(
                            "early stopping at epoch %d/%d", epoch + 1, epochs
                        )
                    break
        return history

    def _n_samples(self, N: int) -> int:
        return N

    def _gather_batch(self, X, Y, idx):
        G, F = X.shape[0], X.shape[2]
        Fo = Y.shape[2]
        Xb = X.gather(1, idx.unsqueeze(-1).expand(G, idx.shape[1], F))
        Tb = Y.gather(1, idx.unsqueeze(-1).expand(G, idx.shape[1], Fo))
        return Xb, Tb


class DensePack(BasePack):
    """Grouped feedforward autoencoder trainer (kernels K1-K4 of
    SURVEY.md §2.3)."""

    def _declare_params(self):
        dims = self.spec.dense_dims()
        self.layer_meta_logical = dims
        p = self._pad
        # physical layer dims padded to 8 (pad units provably inert —
        # see pad_enabled); activations between layers inherit the
        # padded widths so every grouped GEMM stages 16-byte vectors
        self.layer_meta = [
            (p(fin), p(fout), act, l1) for fin, fout, act, l1 in dims
        ]
        self._logical_in = dims[0][0]
        self._logical_out = dims[-1][1]
        for i, (fin, fout, _act, _l1) in enumerate(self.layer_meta):
            lfin, lfout = dims[i][0], dims[i][1]
            self.store.declare(f"W{i}", (self.G, fin, fout),
                               logical=(self.G, lfin, lfout))
            self.store.declare(f"b{i}", (self.G, fout),
                               logical=(self.G, lfout))

    def _init_weights(self):
        for g in range(self.G):
            gen = torch.Generator().manual_seed(int(self.seeds[g]) & 0x7FFFFFFF)
            for i, (lfin, lfout, _act, _l1) in enumerate(
                self.layer_meta_logical
            ):
                self.store.views[f"W{i}"][g][:lfin, :lfout].copy_(
                    _glorot_uniform((lfin, lfout), gen)
                )
                # biases stay zero

    def forward(self, X: torch.Tensor) -> List[torch.Tensor]:
        acts = [X]
        a = X
        for i, (_fin, _fout, act, _l1) in enumerate(self.layer_meta):
            a = ops.grouped_linear_fwd(
                a, self.store.cviews[f"W{i}"],
                self.store.views[f"b{i}"], act,
            )
            acts.append(a)
        return acts

    def predict(self, X: torch.Tensor) -> torch.Tensor:
        Xc = self._to_compute(X)
        if self.pad8:
            Xc = self._pad_features(Xc, self._logical_in)
        out = self.forward(Xc)[-1]
        return out[..., : self._logical_out]

    def eval_batch(self, Xb, Tb) -> torch.Tensor:
        # pad diffs are exactly 0; divide by the REAL element count
        Xb, Tb = self._pad_io(Xb, Tb)
        out = self.forward(Xb)[-1].float()
        n = Xb.shape[1] * self._logical_out
        return ((out - Tb.float()) ** 2).sum(dim=(1, 2)) / n

    def train_batch(self, Xb, Tb) -> torch.Tensor:
        Xb, Tb = self._pad_io(Xb, Tb)
        acts = self.forward(Xb)
        loss, dA = ops.mse_bwd(
            acts[-1], Tb.to(acts[-1].dtype),
            Xb.shape[1] * self._logical_out,
        )
        for i in range(len(self.layer_meta) - 1, -1, -1):
            _fin, _fout, act, l1 = self.layer_meta[i]
            dZ = ops.act_l1_bwd(dA, acts[i + 1], act, l1)
            dW, db = ops.grouped_linear_wgrad(acts[i], dZ)
            self.store.gviews[f"W{i}"].copy_(dW)
            self.store.gviews[f"b{i}"].copy_(db)
            if i > 0:
                dA = ops.grouped_linear_bwd_data(dZ, self.store.cviews[f"W{i}"])
        a = self.spec.adam_params
        self.store.adam_step(a["lr"], a["beta1"], a["beta2"], a["eps"])
        return loss


class LSTMPack(BasePack):
    """Grouped stacked-LSTM autoencoder/forecast trainer (kernels K5-K7).

    The sliding-window featurizer is zero-copy: windows are gathered
    straight out of the resident [G, N, F] series tensor (the device
    analog of create_keras_timeseriesgenerator, reference
    models.py:713-793)."""

    @staticmethod
    def _gate_converters(H, pH, row_l):
        """Logical<->physical converters for gate-interleaved LSTM
        params: physical gate blocks are pH wide (padded), logical H —
        so slicing alone can't extract them."""
        def extract(arr):
            if arr.ndim == 1:
                return np.concatenate(
                    [arr[gb * pH : gb * pH + H] for gb in range(4)]
                ).copy()
            r = arr[:row_l] if row_l else arr
            return np.concatenate(
                [r[:, gb * pH : gb * pH + H] for gb in range(4)], axis=1
            ).copy()

        def insert(view, t):
            view.zero_()
            if view.dim() == 1:
                for gb in range(4):
                    view[gb * pH : gb * pH + H].copy_(
                        t[gb * H : (gb + 1) * H]
                    )
            else:
                rl = row_l if row_l else view.shape[0]
                for gb in range(4):
                    view[:rl, gb * pH : gb * pH + H].copy_(
                        t[:, gb * H : (gb + 1) * H]
                    )

        return extract, insert

    def _declare_params(self):
        spec = self.spec
        p = self._pad
        self.lstm_meta = []          # PHYSICAL (fin, H, return_sequences)
        self.lstm_meta_logical = []  # logical dims for serialization
        fin_l = spec.n_features
        lstm_layers = [l for l in spec.layers if l.kind == "lstm"]
        dense_layers = [l for l in spec.layers if l.kind == "dense"]
        assert len(dense_layers) == 1, "LSTM spec needs exactly one output dense layer"
        self._logical_in = fin_l
        for i, layer in enumerate(lstm_layers):
            H_l = layer.units
            fin, H = p(fin_l), p(H_l)
            self.store.declare(f"Wx{i}", (self.G, fin, 4 * H),
                               logical=(self.G, fin_l, 4 * H_l))
            self.store.declare(f"Wh{i}", (self.G, H, 4 * H),
                               logical=(self.G, H_l, 4 * H_l))
            self.store.declare(f"bl{i}", (self.G, 4 * H),
                               logical=(self.G, 4 * H_l))
            if H != H_l or fin != fin_l:
                self._extractors[f"Wx{i}"] = self._gate_converters(
                    H_l, H, fin_l
                )
                self._extractors[f"Wh{i}"] = self._gate_converters(
                    H_l, H, H_l
                )
                self._extractors[f"bl{i}"] = self._gate_converters(
                    H_l, H, None
                )
            self.lstm_meta.append((fin, H, layer.return_sequences))
            self.lstm_meta_logical.append((fin_l, H_l, layer.return_sequences))
            fin_l = H_l
        out_layer = dense_layers[0]
        self._logical_out = out_layer.units
        self.dense_meta = (p(fin_l), p(out_layer.units), out_layer.activation)
        self.dense_meta_logical = (fin_l, out_layer.units, out_layer.activation)
        self.store.declare("Wd", (self.G, p(fin_l), p(out_layer.units)),
                           logical=(self.G, fin_l, out_layer.units))
        self.store.declare("bd", (self.G, p(out_layer.units)),
                           logical=(self.G, out_layer.units))

    def _init_weights(self):
        # draw the LOGICAL matrices in the same RNG order as the
        # unpadded CPU pack (seed determinism across devices), then
        # scatter into the padded gate blocks
        for g in range(self.G):
            gen = torch.Generator().manual_seed(int(self.seeds[g]) & 0x7FFFFFFF)
            for i, (fin_l, H_l, _rs) in enumerate(self.lstm_meta_logical):
                _pfin, pH, _ = self.lstm_meta[i]
                wx = _glorot_uniform((fin_l, 4 * H_l), gen)
                wh = torch.cat(
                    [_orthogonal((H_l, H_l), gen) for _ in range(4)], dim=1
                )
                b = torch.zeros(4 * H_l)
                b[H_l : 2 * H_l] = 1.0  # unit forget-gate bias (Keras)
                self._insert_logical(
                    f"Wx{i}", self.store.views[f"Wx{i}"][g], wx.numpy()
                )
                self._insert_logical(
                    f"Wh{i}", self.store.views[f"Wh{i}"][g], wh.numpy()
                )
                self._insert_logical(
                    f"bl{i}", self.store.views[f"bl{i}"][g], b.numpy()
                )
            fin_l, fout_l, _act = self.dense_meta_logical
            wd = _glorot_uniform((fin_l, fout_l), gen)
            self.store.views["Wd"][g][:fin_l, :fout_l].copy_(wd)

    # ---- sequence forward/backward ------------------------------------
    def _use_fused(self, B: Optional[int] = None) -> bool:
        """The fused on-device sequence-scan kernels (lstm_seq.hip) run
        the whole T-step recurrence in one launch per layer (Wh in LDS
        for H<=64, streamed from L2 for 64<H<=256). Used on GPU when
        every layer has a fused kernel AND — for big-H stacks — the
        scan grid fills the chip: the round-2 GPU A/B at dims
        256/128/64 measured fused 14% SLOWER at G=16 (128 workgroups on
        256 CUs) and 7% faster at G=64 (512 WGs), so stacks with H>64
        take the fused path only when G*ceil(B/32) >= 256 (the
        per-timestep grouped-GEMM fallback splits the gate columns over
        many more workgroups and wins in the underfilled regime)."""
        if self.device.type != "cuda":
            return False
        if not all(
            ops.lstm_seq_available(H) for _fin, H, _rs in self.lstm_meta
        ):
            return False
        if any(H > 64 for _fin, H, _rs in self.lstm_meta):
            rows = B if B is not None else 256
            if self.G * ((rows + 31) // 32) < 256:
                return False
        return True

    def _forward_seq(self, Xw: torch.Tensor, keep: bool):
        """Xw: [G, B, T, F]. Returns (y, cache)."""
        G, B, T, _ = Xw.shape
        if self._use_fused(B):
            return self._forward_seq_fused(Xw, keep)
        seq = Xw
        cache = []
        for li, (fin, H, rs) in enumerate(self.lstm_meta):
            Wx = self.store.cviews[f"Wx{li}"]
            Wh = self.store.cviews[f"Wh{li}"]
            b = self.store.views[f"bl{li}"]
            flat_in = seq.reshape(G, B * T, fin)
            gates_all = ops.grouped_linear_fwd(flat_in, Wx, b, "linear").view(
                G, B, T, 4 * H
            )
            h = torch.zeros(G, B, H, dtype=self.compute_dtype, device=self.device)
            c = torch.zeros(G, B, H, dtype=torch.float32, device=self.device)
            hs = torch.empty(
                G, B, T, H, dtype=self.compute_dtype, device=self.device
            )
            cs = (
                torch.empty(G, B, T, H, dtype=torch.float32, device=self.device)
                if keep
                else None
            )
            gacts = (
                torch.empty(
                    G, B, T, 4 * H, dtype=self.compute_dtype, device=self.device
                )
                if keep
                else None
            )
            for t in range(T):
                gates = gates_all[:, :, t].contiguous()
                ops.grouped_gemm_acc(h, Wh, gates)
                h, c, gact = ops.lstm_pointwise_fwd(gates, c)
                hs[:, :, t] = h
                if keep:
                    cs[:, :, t] = c
                    gacts[:, :, t] = gact
            cache.append(
                dict(seq_in=seq if keep else None, hs=hs, cs=cs, gacts=gacts)
            )
            seq = hs
        h_last = seq[:, :, -1].contiguous()
        fin, fout, act = self.dense_meta
        y = ops.grouped_linear_fwd(
            h_last, self.store.cviews["Wd"], self.store.views["bd"], act
        )
        if keep:
            cache.append(dict(h_last=h_last, y=y))
        return y, cache

    def _forward_seq_fused(self, Xw: torch.Tensor, keep: bool):
        """GPU path: one fused scan per layer. Layers fitting the v4
        geometry compute the x-side gate GEMM INSIDE the scan (no xW
        HBM round trip — the fleet's scans are bandwidth-bound) and
        skip the cs/gacts stores on inference passes; other layers use
        the two-step x-GEMM + scan."""
        import os as _os

        G, B, T, _ = Xw.shape
        v4_on = _os.environ.get("GORDO_LSTM_V4", "1") != "0"
        seq = Xw
        cache = []
        for li, (fin, H, rs) in enumerate(self.lstm_meta):
            Wx = self.store.cviews[f"Wx{li}"]
            Wh = self.store.cviews[f"Wh{li}"]
            b = self.store.views[f"bl{li}"]
            if v4_on and ops.lstm_v4_available(H, fin):
                out = ops.lstm_seq_fwd_fused(seq, Wx, Wh, b,
                                             store_aux=keep)
                hs = out[0]
                cs = out[1] if keep else None
                gacts = out[2] if keep else None
            else:
                xW = ops.grouped_linear_fwd(
                    seq.reshape(G, B * T, fin), Wx, b, "linear"
                ).view(G, B, T, 4 * H)
                hs, cs, gacts = ops.lstm_seq_fwd(xW, Wh)
                if not keep:
                    cs = gacts = None
            cache.append(
                dict(
                    seq_in=seq if keep else None,
                    hs=hs,
                    cs=cs,
                    gacts=gacts,
                )
            )
            seq = hs
        h_last = seq[:, :, -1].contiguous()
        fin, fout, act = self.dense_meta
        y = ops.grouped_linear_fwd(
            h_last, self.store.cviews["Wd"], self.store.views["bd"], act
        )
        if keep:
            cache.append(dict(h_last=h_last, y=y))
        return y, cache

    def predict_windows(self, Xw: torch.Tensor) -> torch.Tensor:
        Xc = self._to_compute(Xw)
        if self.pad8 and Xc.shape[-1] == self._logical_in:
            Xc = self._pad_features(Xc, self._logical_in)
        y, _ = self._forward_seq(Xc, keep=False)
        return y[..., : self._logical_out]

    def eval_batch(self, Xw, Tb) -> torch.Tensor:
        Xw, Tb = self._pad_io(Xw, Tb)
        y = self.predict_windows(Xw).float()
        return (
            (y - Tb.float()[..., : self._logical_out]) ** 2
        ).mean(dim=(1, 2))

    def _train_batch_fused(self, Xw, Tb) -> torch.Tensor:
        G, B, T, _ = Xw.shape
        y, cache = self._forward_seq_fused(Xw, keep=True)
        loss, dY = ops.mse_bwd(
            y, Tb.to(y.dtype), B * self._logical_out
        )

        fin, fout, act = self.dense_meta
        head = cache[-1]
        dZ = ops.act_l1_bwd(dY, head["y"], act, 0.0)
        dWd, dbd = ops.grouped_linear_wgrad(head["h_last"], dZ)
        self.store.gviews["Wd"].copy_(dWd)
        self.store.gviews["bd"].copy_(dbd)
        dh_last = ops.grouped_linear_bwd_data(dZ, self.store.cviews["Wd"])

        import os as _os

        v5_on = _os.environ.get("GORDO_LSTM_V4", "1") != "0"
        dSeq = None
        for li in range(len(self.lstm_meta) - 1, -1, -1):
            fin, H, rs = self.lstm_meta[li]
            lc = cache[li]
            Wh = self.store.cviews[f"Wh{li}"]
            Wx = self.store.cviews[f"Wx{li}"]
            last_only = dSeq is None
            if li > 0 and v5_on and ops.lstm_v4_available(H, fin):
                # v5: the dSeq GEMM rides inside the reverse scan
                dG_flat, next_dSeq = ops.lstm_seq_bwd_fused(
                    dh_last if last_only else dSeq,
                    lc["gacts"], lc["cs"], Wh, Wx, last_only,
                )
                dG_flat = dG_flat.view(G, B * T, 4 * H)
            else:
                next_dSeq = None
                dG_flat = ops.lstm_seq_bwd(
                    dh_last if last_only else dSeq,
                    lc["gacts"], lc["cs"], Wh, last_only,
                ).view(G, B * T, 4 * H)
            hs = lc["hs"]
            # combined pass: dZ (dG) staged once for both weight grads;
            # dWh reads h_{t-1} via in-kernel shifted addressing
            dWx, dWh, dbl = ops.grouped_wgrad_xh(
                lc["seq_in"].reshape(G, B * T, fin), hs, dG_flat, T
            )
            self.store.gviews[f"Wx{li}"].copy_(dWx)
            self.store.gviews[f"Wh{li}"].copy_(dWh)
            self.store.gviews[f"bl{li}"].copy_(dbl)
            # free this layer's BPTT cache before descending: the
            # allocator reuses it for the next layer's dSeq/transients
            # (whole-batch caches are the memory ceiling at G~100+).
            cache[li] = None
            del lc, hs
            prev_dSeq = dSeq
            if li > 0:
                dSeq = (
                    next_dSeq
                    if next_dSeq is not None
                    else ops.grouped_linear_bwd_data(dG_flat, Wx).view(
                        G, B, T, fin
                    )
                )
            del dG_flat, prev_dSeq

        a = self.spec.adam_params
        self.store.adam_step(a["lr"], a["beta1"], a["beta2"], a["eps"])
        return loss

    def train_batch(self, Xw, Tb) -> torch.Tensor:
        Xw, Tb = self._pad_io(Xw, Tb)
        if self._use_fused(Xw.shape[1]):
            return self._train_batch_fused(Xw, Tb)
        G, B, T, _ = Xw.shape
        y, cache = self._forward_seq(Xw, keep=True)
        loss, dY = ops.mse_bwd(
            y, Tb.to(y.dtype), B * self._logical_out
        )

        # output dense layer backward
        fin, fout, act = self.dense_meta
        head = cache[-1]
        dZ = ops.act_l1_bwd(dY, head["y"], act, 0.0)
        dWd, dbd = ops.grouped_linear_wgrad(head["h_last"], dZ)
        self.store.gviews["Wd"].copy_(dWd)
        self.store.gviews["bd"].copy_(dbd)
        dh_last = ops.grouped_linear_bwd_data(dZ, self.store.cviews["Wd"])

        # BPTT through the LSTM stack
        dSeq: Optional[torch.Tensor] = None  # grad on layer output sequence
        for li in range(len(self.lstm_meta) - 1, -1, -1):
            fin, H, rs = self.lstm_meta[li]
            lc = cache[li]
            hs, cs, gacts = lc["hs"], lc["cs"], lc["gacts"]
            Wh = self.store.cviews[f"Wh{li}"]
            Wx = self.store.cviews[f"Wx{li}"]
            dG = torch.empty(
                G, B, T, 4 * H, dtype=self.compute_dtype, device=self.device
            )
            dh_carry = torch.zeros(
                G, B, H, dtype=self.compute_dtype, device=self.device
            )
            dc_carry = torch.zeros(G, B, H, dtype=torch.float32, device=self.device)
            for t in range(T - 1, -1, -1):
                dh_t = dh_carry
                if dSeq is not None:
                    dh_t = dh_t + dSeq[:, :, t]
                elif li == len(self.lstm_meta) - 1 and t == T - 1:
                    dh_t = dh_t + dh_last
                c_prev = (
                    cs[:, :, t - 1]
                    if t > 0
                    else torch.zeros_like(dc_carry)
                )
                dgates, dc_carry = ops.lstm_pointwise_bwd(
                    dh_t, dc_carry, gacts[:, :, t], cs[:, :, t], c_prev
                )
                dG[:, :, t] = dgates
                dh_carry = ops.grouped_linear_bwd_data(dgates, Wh)
            # batched weight grads over all (B, T) rows
            dG_flat = dG.view(G, B * T, 4 * H)
            dWx, dWh, dbl = ops.grouped_wgrad_xh(
                lc["seq_in"].reshape(G, B * T, fin), hs, dG_flat, T
            )
            self.store.gviews[f"Wx{li}"].copy_(dWx)
            self.store.gviews[f"Wh{li}"].copy_(dWh)
            self.store.gviews[f"bl{li}"].copy_(dbl)
            if li > 0:
                dSeq = ops.grouped_linear_bwd_data(dG_flat, Wx).view(G, B, T, fin)

        a = self.spec.adam_params
        self.store.adam_step(a["lr"], a["beta1"], a["beta2"], a["eps"])
        return loss

    # ---- windowed fit/predict over a series ---------------------------
    @property
    def lookback(self) -> int:
        return self.spec.lookback_window

    @property
    def lookahead(self) -> int:
        return self.spec.lookahead

    def n_windows(self, N: int) -> int:
        return N - self.lookback + 1 - self.lookahead

    _n_samples = n_windows

    def _gather_batch(self, X, Y, idx):
        """idx: [G, B] window start indices. Builds [G,B,T,F] window
        tensors (K7 HIP gather kernel on GPU) and [G,B,Fo] targets
        straight from the series."""
        G, N, F = X.shape
        Fo = Y.shape[2]
        B = idx.shape[1]
        T = self.lookback
        Xw = ops.window_gather(X, idx, T)
        trow = idx + (T - 1 + self.lookahead)
        Tb = Y.gather(1, trow.unsqueeze(-1).expand(G, B, Fo))
        return Xw, Tb

    def predict(self, X: torch.Tensor, batch_size: int = 4096) -> torch.Tensor:
        """X: [G, N, F] series → [G, n_windows, Fo] predictions
        (offset-aligned like the reference: output starts at row
        lookback-1+lookahead)."""
        X = self._to_compute(X)
        if self.pad8:
            X = self._pad_features(X, self._logical_in)
        G, N, F = X.shape
        nw = self.n_windows(N)
        outs = []
        dummyY = X.new_zeros(G, N, self.dense_meta[1])
        for s in range(0, nw, batch_size):
            idx = (
                torch.arange(s, min(s + batch_size, nw), device=self.device)
                .unsqueeze(0)
                .expand(G, -1)
            )
            Xw, _ = self._gather_batch(X, dummyY, idx)
            outs.append(self.predict_windows(Xw))
        return torch.cat(outs, dim=1)
