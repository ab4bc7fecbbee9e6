"""Round-state checkpoint / resume for sequential meta-estimators.

The reference's ``PeriodicRDDCheckpointer`` (used at e.g. reference
BoostingClassifier.scala:169-173,267 and GBMRegressor.scala:314-318,442)
is lineage management only — ``fit`` there cannot resume a half-trained
ensemble.  SURVEY.md §5.4 calls for the MI355X rebuild to turn the
``checkpointInterval`` param into real, durable round-state dumps: an
ensemble is trivially resumable from {models so far, stage weights, round
index}, since margins can be recomputed by replaying the saved stages.

Layout under ``<dir>/``::

    state.json                # {"round": i, "weights": [...], "extra": {...}}
    model-0/ model-1/ ...     # nested model saves (persistence layout)
    model-3-0/ model-3-1/ ... # GBM classifier: per-class nesting

Writes are rank-0 only (split decisions are identical on every rank, so
the models are too) and atomic per round: state.json is written last, to a
temp name, then renamed.
"""

from __future__ import annotations

import json
import os
import shutil
from typing import Any, Dict, List, Optional, Sequence, Tuple

from .. import persistence


def _is_rank0() -> bool:
    from ..parallel import get_comm

    try:
        return get_comm().rank == 0
    except Exception:  # noqa: BLE001
        return True


def save_round_state(
    ckpt_dir: str,
    round_idx: int,
    models: Sequence,  # flat models, or per-round lists of per-class models
    weights: Sequence,  # floats, or per-round lists of floats
    extra: Optional[Dict[str, Any]] = None,
) -> None:
    """Dump the ensemble state after ``round_idx`` rounds (rank 0 only)."""
    if not _is_rank0():
        return
    os.makedirs(ckpt_dir, exist_ok=True)
    nested = bool(models) and isinstance(models[0], (list, tuple))
    for i, m in enumerate(models):
        if nested:
            for k, mk in enumerate(m):
                p = os.path.join(ckpt_dir, f"model-{i}-{k}")
                if not os.path.exists(p):
                    mk.save(p, overwrite=True)
        else:
            p = os.path.join(ckpt_dir, f"model-{i}")
            if not os.path.exists(p):
                m.save(p, overwrite=True)
    if nested:
        weights_json = [[float(x) for x in w] for w in weights]
    else:
        weights_json = [float(w) for w in weights]
    state = {
        "round": int(round_idx),
        "nested": nested,
        "weights": weights_json,
        "extra": extra or {},
    }
    tmp = os.path.join(ckpt_dir, ".state.json.tmp")
    with open(tmp, "w") as f:
        json.dump(state, f)
    os.replace(tmp, os.path.join(ckpt_dir, "state.json"))


def load_round_state(
    ckpt_dir: Optional[str],
) -> Optional[Tuple[int, List, List, Dict[str, Any]]]:
    """Return (round, models, weights, extra) or None if no usable state."""
    if not ckpt_dir:
        return None
    sp = os.path.join(ckpt_dir, "state.json")
    if not os.path.exists(sp):
        return None
    with open(sp) as f:
        state = json.load(f)
    r = int(state["round"])
    nested = bool(state.get("nested"))
    models: List = []
    for i in range(r):
        if nested:
            ms = []
            k = 0
            while os.path.isdir(os.path.join(ckpt_dir, f"model-{i}-{k}")):
                ms.append(
                    persistence.load_instance(os.path.join(ckpt_dir, f"model-{i}-{k}"))
                )
                k += 1
            if not ms:
                return None  # partial dump: ignore
            models.append(ms)
        else:
            p = os.path.join(ckpt_dir, f"model-{i}")
            if not os.path.isdir(p):
                return None
            models.append(persistence.load_instance(p))
    return r, models, state["weights"], state.get("extra", {})


def clear(ckpt_dir: Optional[str]) -> None:
    if ckpt_dir and os.path.isdir(ckpt_dir) and _is_rank0():
        shutil.rmtree(ckpt_dir, ignore_errors=True)
