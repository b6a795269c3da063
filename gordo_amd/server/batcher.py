"""
Serving micro-batcher — coalesce concurrent predictions for one model.

SURVEY §7 phase-4 item ("request batcher"): with the JSON codec off the
critical path, high-concurrency GPU serving becomes forward-bound on
many tiny (≈100-row) kernel launches. The batcher lets concurrent
requests for the SAME model ride one packed forward: the first arrival
becomes the leader, waits a short window for followers, stacks their
rows, runs one predict, and splits the output back.

Safety: batching is only valid for row-aligned models (len(output) ==
len(input)); windowed models (LSTM lookback) would mix windows across
request boundaries, so the batcher self-detects misalignment on the
first call and bypasses that model permanently.

Opt-in: ``GORDO_SERVE_BATCH=1`` (window via
``GORDO_SERVE_BATCH_WINDOW_MS``, default 2 ms). Off by default pending
GPU measurement (ROADMAP).
"""
from __future__ import annotations

import logging
import os
import threading
from typing import Any, Callable, Dict, List, Optional

import numpy as np

logger = logging.getLogger(__name__)


class MicroBatcher:
    """Coalesces concurrent ``predict(X)`` calls into stacked calls."""

    def __init__(
        self,
        predict_fn: Callable[[np.ndarray], np.ndarray],
        window_ms: float = 2.0,
        max_rows: int = 16384,
    ):
        self._predict = predict_fn
        self._window_s = window_ms / 1000.0
        self._max_rows = max_rows
        self._lock = threading.Lock()
        self._have_leader = False
        self._pending: List[dict] = []
        self._aligned: Optional[bool] = None  # unknown until first call

    def predict(self, X: np.ndarray) -> np.ndarray:
        X = np.asarray(X)
        if self._aligned is False or X.ndim != 2:
            return self._predict(X)
        if self._aligned is None:
            # probe call: establish whether the model is row-aligned
            out = self._predict(X)
            with self._lock:
                self._aligned = len(out) == len(X)
            return out

        slot = {"X": X, "event": threading.Event(), "out": None, "err": None}
        with self._lock:
            if self._have_leader:
                self._pending.append(slot)
                follower = True
            else:
                self._have_leader = True
                follower = False

        if follower:
            slot["event"].wait()
            if slot["err"] is not None:
                raise slot["err"]
            if slot["out"] is None:
                # leader couldn't take us (row budget): run solo
                return self._predict(X)
            return slot["out"]

        # leader: collect followers for one window, then execute
        if self._window_s > 0:
            threading.Event().wait(self._window_s)
        with self._lock:
            batch = []
            rows = len(X)
            rest = []
            for s in self._pending:
                if (
                    rows + len(s["X"]) <= self._max_rows
                    and s["X"].shape[1:] == X.shape[1:]
                ):
                    batch.append(s)
                    rows += len(s["X"])
                else:
                    rest.append(s)
            # from here on, new arrivals elect their own leader; nothing
            # stays parked: over-budget/mismatched slots run solo NOW
            self._pending = []
            self._have_leader = False
        for s in rest:
            s["event"].set()  # out stays None -> the follower runs solo
        if not batch:
            return self._predict(X)
        stacked = np.concatenate([X] + [s["X"] for s in batch], axis=0)
        try:
            out = self._predict(stacked)
            if len(out) != len(stacked):  # misaligned after all
                with self._lock:
                    self._aligned = False
                raise ValueError("model output is not row-aligned")
        except Exception as exc:
            for s in batch:
                s["err"] = exc
                s["event"].set()
            raise
        ofs = len(X)
        mine = out[:ofs]
        for s in batch:
            n = len(s["X"])
            s["out"] = out[ofs:ofs + n]
            ofs += n
            s["event"].set()
        return mine


import weakref

# Weakly-keyed registry: a batcher dies with its model, so models
# evicted from the serving LRU (and their HBM residency) are freed
# normally — an id()-keyed dict would pin every model's predict
# closure forever. Not stored as a model attribute: the batcher holds
# a threading.Lock, which would break model pickling (download-model).
_batchers: "weakref.WeakKeyDictionary[Any, MicroBatcher]" = (
    weakref.WeakKeyDictionary()
)
_batchers_fallback: Dict[int, MicroBatcher] = {}  # un-weakref-able models
_batchers_lock = threading.Lock()


def enabled() -> bool:
    return os.environ.get("GORDO_SERVE_BATCH") == "1"


def batched_predict(model: Any, X: np.ndarray) -> np.ndarray:
    """Route ``model.predict`` through the model's micro-batcher."""
    with _batchers_lock:
        try:
            b = _batchers.get(model)
        except TypeError:  # unhashable/un-weakref-able model
            b = _batchers_fallback.get(id(model))
        if b is None:
            window = float(os.environ.get("GORDO_SERVE_BATCH_WINDOW_MS", 2.0))
            b = MicroBatcher(model.predict, window_ms=window)
            try:
                _batchers[model] = b
            except TypeError:
                _batchers_fallback[id(model)] = b
    return b.predict(X)
