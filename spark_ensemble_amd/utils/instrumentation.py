"""Structured training instrumentation.

The reference wraps every ``train`` in Spark's ``Instrumentation``
(``instrumented { instr => ... }`` — e.g. reference
BaggingRegressor.scala:117-131) and logs the pipeline stage, params and
per-iteration counters (``instr.logNamedValue("iteration", i)``,
BoostingClassifier.scala:182).  This module is the MI355X rebuild of that
subsystem: a structured per-round log — round index,
train/validation loss, stage weight, wall ms — emitted on rank 0 through
the standard :mod:`logging` machinery, plus lightweight named wall-clock
timers for per-phase attribution (histogram build, split search, line
search ...), synchronized with the HIP stream so GPU phases are measured,
not queued.

Usage::

    with instrumented(self, dataset) as instr:
        for i in range(k):
            with instr.timed("round"):
                ...
            instr.log_round(i, loss=err, weight=w)

Enable console output with ``SEA_LOG=1`` (INFO) or ``SEA_LOG=debug``; the
records are also appended to ``Instrumentation.history`` on the instance so
tests and callers can assert on them without parsing log text.
"""

from __future__ import annotations

import logging
import os
import time
from contextlib import contextmanager
from typing import Dict, List, Optional

import torch

logger = logging.getLogger("spark_ensemble_amd")

_env = os.environ.get("SEA_LOG", "")
if _env and not logger.handlers:
    _h = logging.StreamHandler()
    _h.setFormatter(logging.Formatter("[sea %(levelname).1s] %(message)s"))
    logger.addHandler(_h)
    logger.setLevel(logging.DEBUG if _env.lower() == "debug" else logging.INFO)


def _rank() -> int:
    from ..parallel import get_comm

    try:
        return get_comm().rank
    except Exception:  # noqa: BLE001
        return 0


class Instrumentation:
    """Per-fit structured logger + timer registry (rank-0 emission)."""

    def __init__(self, estimator, dataset=None):
        self.stage = type(estimator).__name__
        self.uid = getattr(estimator, "uid", "?")
        try:
            estimator._instr = self  # expose the live record on the estimator
        except Exception:  # noqa: BLE001
            pass
        self.history: List[Dict] = []
        self.timers: Dict[str, float] = {}
        self._t0 = time.perf_counter()
        self._emit = _rank() == 0
        self._device: Optional[torch.device] = None
        if dataset is not None:
            try:
                feats = dataset["features"]
                self._device = feats.device
                if self._emit:
                    logger.info(
                        "%s(%s) fit: rows=%d features=%d device=%s",
                        self.stage, self.uid, feats.shape[0], feats.shape[1],
                        feats.device,
                    )
            except Exception:  # noqa: BLE001
                pass

    # -- params ------------------------------------------------------------
    def log_params(self, estimator, *names):
        if not self._emit or not logger.isEnabledFor(logging.INFO):
            return
        vals = {}
        for nm in names or list(getattr(estimator, "_params", {})):
            try:
                vals[nm] = estimator.getOrDefault(nm)
            except Exception:  # noqa: BLE001
                pass
        logger.info("%s params: %s", self.stage, vals)

    # -- named values / per-round records ----------------------------------
    def log_named_value(self, name: str, value):
        if self._emit:
            logger.info("%s %s=%s", self.stage, name, value)

    def log_round(self, round_idx: int, **metrics):
        rec = {"round": round_idx, "ms": self.elapsed_ms(), **metrics}
        self.history.append(rec)
        if self._emit and logger.isEnabledFor(logging.INFO):
            body = " ".join(
                f"{k}={v:.6g}" if isinstance(v, float) else f"{k}={v}"
                for k, v in metrics.items()
            )
            logger.info("%s round %d: %s (t=%.1fms)", self.stage, round_idx,
                        body, rec["ms"])

    def elapsed_ms(self) -> float:
        return (time.perf_counter() - self._t0) * 1000.0

    # -- timers --------------------------------------------------------------
    @contextmanager
    def timed(self, name: str, sync: bool = True):
        """Accumulate wall ms under ``name``; syncs the HIP stream around the
        region when a GPU device is in play so the measurement is real."""
        if sync and self._device is not None and self._device.type == "cuda":
            torch.cuda.synchronize(self._device)
        t0 = time.perf_counter()
        try:
            yield
        finally:
            if sync and self._device is not None and self._device.type == "cuda":
                torch.cuda.synchronize(self._device)
            self.timers[name] = self.timers.get(name, 0.0) + (
                time.perf_counter() - t0
            ) * 1000.0

    def finish(self):
        if self._emit and logger.isEnabledFor(logging.INFO):
            summary = " ".join(f"{k}={v:.1f}ms" for k, v in self.timers.items())
            logger.info("%s done in %.1fms %s", self.stage, self.elapsed_ms(),
                        summary)


@contextmanager
def instrumented(estimator, dataset=None):
    instr = Instrumentation(estimator, dataset)
    # expose the live instrumentation on the estimator so helpers deep in the
    # call tree (line search, tree grower) can attribute time to it
    estimator._instr = instr
    try:
        yield instr
    finally:
        instr.finish()
        estimator._instr = None
