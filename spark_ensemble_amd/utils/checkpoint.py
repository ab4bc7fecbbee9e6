"""Round-state checkpoint / resume for sequential meta-estimators.

The reference's ``PeriodicRDDCheckpointer`` (used at e.g. reference
BoostingClassifier.scala:169-173,267 and GBMRegressor.scala:314-318,442)
is lineage management only — ``fit`` there cannot resume a half-trained
ensemble.  SURVEY.md §5.4 calls for the MI355X rebuild to turn the
``checkpointInterval`` param into real, durable round-state dumps: an
ensemble is trivially resumable from {models so far, stage weights, round
index}, since margins can be recomputed by replaying the saved stages.

Validity contract: a checkpoint is only resumed when its **fingerprint**
(estimator class + full param map, minus execution-only knobs, + dataset
shape and a cheap content hash) matches the current fit — so a re-run
after changing params or data, or a CrossValidator fitting per fold,
never silently adopts stale state (the reference's ``checkpointInterval``
never changes the fitted result).  A successful ``fit`` clears its own
checkpoint directory; the dump only survives a crash.

Layout under ``<dir>/``::

    state.json                # {"round": i, "weights": [...], "extra": {...},
                              #  "fingerprint": "...", "model_dirs": [...]}
    model-0/ model-1/ ...     # nested model saves (persistence layout)
    model-3-0/ model-3-1/ ... # GBM classifier: per-class nesting

Writes are rank-0 only (split decisions are identical on every rank, so
the models are too) and atomic: each model is written to a temp directory
and renamed into place; state.json is written last, to a temp name, then
renamed — and lists exactly the model dirs it trusts, so a partial dir
from a crash mid-save is never adopted.
"""

from __future__ import annotations

import hashlib
import json
import os
import shutil
from typing import Any, Dict, List, Optional, Sequence, Tuple

from .. import persistence

# Params that change execution, not the fitted result — excluded from the
# fingerprint so e.g. resuming with a different checkpointInterval works.
_EXEC_ONLY_PARAMS = ("checkpointDir", "checkpointInterval", "parallelism")


def _is_rank0() -> bool:
    from ..parallel import get_comm

    try:
        return get_comm().rank == 0
    except Exception:  # noqa: BLE001
        return True


def _describe_estimator(est) -> Any:
    """Flat, JSON-able description of an estimator (class + params,
    nested estimators described recursively)."""
    if est is None:
        return None
    if isinstance(est, (list, tuple)):
        return [_describe_estimator(e) for e in est]
    d: Dict[str, Any] = {"__class__": type(est).__name__}
    for p in est.params:
        name = p.name
        if name in _EXEC_ONLY_PARAMS:
            continue
        if name in est._NESTED_PARAM_NAMES:
            d[name] = _describe_estimator(est.getOrNone(name))
            continue
        try:
            d[name] = est.getOrDefault(name)
        except KeyError:
            d[name] = None
    return d


def fingerprint(est, n_rows: int, n_features: int, y=None, w=None) -> str:
    """Hash of everything that determines the fitted result: estimator
    class + param map (recursing into nested base learners), dataset
    shape, and cheap content sums of label/weight."""
    payload: Dict[str, Any] = {
        "est": _describe_estimator(est),
        "n_rows": int(n_rows),
        "n_features": int(n_features),
    }
    if y is not None:
        payload["y_sum"] = float(y.double().sum().item())
    if w is not None:
        payload["w_sum"] = float(w.double().sum().item())
    blob = json.dumps(payload, sort_keys=True, default=str).encode()
    return hashlib.sha256(blob).hexdigest()[:24]


def _atomic_model_save(model, path: str) -> None:
    """Write ``model`` to ``path`` via tmp-dir + rename so a crash mid-save
    never leaves a partial directory under the final name."""
    tmp = path + ".tmp"
    if os.path.isdir(tmp):
        shutil.rmtree(tmp, ignore_errors=True)
    model.save(tmp, overwrite=True)
    if os.path.isdir(path):  # pre-existing (unlisted/stale): replace it
        shutil.rmtree(path, ignore_errors=True)
    os.rename(tmp, path)


def save_round_state(
    ckpt_dir: str,
    round_idx: int,
    models: Sequence,  # flat models, or per-round lists of per-class models
    weights: Sequence,  # floats, or per-round lists of floats
    extra: Optional[Dict[str, Any]] = None,
    fingerprint: Optional[str] = None,
    _saved_dirs: Optional[set] = None,
) -> None:
    """Dump the ensemble state after ``round_idx`` rounds (rank 0 only).

    ``_saved_dirs`` (optional, caller-held set) skips re-saving model dirs
    this fit already wrote — dirs NOT in it (stale from another run) are
    rewritten, never adopted.
    """
    if not _is_rank0():
        return
    os.makedirs(ckpt_dir, exist_ok=True)
    nested = bool(models) and isinstance(models[0], (list, tuple))
    dirs: List[str] = []
    for i, m in enumerate(models):
        if nested:
            for k, mk in enumerate(m):
                name = f"model-{i}-{k}"
                dirs.append(name)
                if _saved_dirs is None or name not in _saved_dirs:
                    _atomic_model_save(mk, os.path.join(ckpt_dir, name))
                    if _saved_dirs is not None:
                        _saved_dirs.add(name)
        else:
            name = f"model-{i}"
            dirs.append(name)
            if _saved_dirs is None or name not in _saved_dirs:
                _atomic_model_save(m, os.path.join(ckpt_dir, name))
                if _saved_dirs is not None:
                    _saved_dirs.add(name)
    if nested:
        weights_json = [[float(x) for x in w] for w in weights]
    else:
        weights_json = [float(w) for w in weights]
    state = {
        "round": int(round_idx),
        "nested": nested,
        "weights": weights_json,
        "extra": extra or {},
        "fingerprint": fingerprint,
        "model_dirs": dirs,
    }
    tmp = os.path.join(ckpt_dir, ".state.json.tmp")
    with open(tmp, "w") as f:
        json.dump(state, f)
    os.replace(tmp, os.path.join(ckpt_dir, "state.json"))


def load_round_state(
    ckpt_dir: Optional[str],
    fingerprint: Optional[str] = None,
) -> Optional[Tuple[int, List, List, Dict[str, Any]]]:
    """Return (round, models, weights, extra) or None if no usable state.

    When both the stored and requested fingerprints are present they must
    match — otherwise the checkpoint belongs to a different fit and is
    ignored (stale-resume returned a wrong model)."""
    if not ckpt_dir:
        return None
    sp = os.path.join(ckpt_dir, "state.json")
    if not os.path.exists(sp):
        return None
    with open(sp) as f:
        state = json.load(f)
    stored_fp = state.get("fingerprint")
    if fingerprint is not None and stored_fp is not None and stored_fp != fingerprint:
        return None
    r = int(state["round"])
    nested = bool(state.get("nested"))
    listed = state.get("model_dirs")
    trusted = set(listed) if listed is not None else None

    def ok(name: str) -> bool:
        if trusted is not None and name not in trusted:
            return False
        return os.path.isdir(os.path.join(ckpt_dir, name))

    models: List = []
    for i in range(r):
        if nested:
            ms = []
            k = 0
            while ok(f"model-{i}-{k}"):
                ms.append(
                    persistence.load_instance(os.path.join(ckpt_dir, f"model-{i}-{k}"))
                )
                k += 1
            if not ms:
                return None  # partial dump: ignore
            models.append(ms)
        else:
            if not ok(f"model-{i}"):
                return None
            models.append(persistence.load_instance(os.path.join(ckpt_dir, f"model-{i}")))
    return r, models, state["weights"], state.get("extra", {})


def clear(ckpt_dir: Optional[str]) -> None:
    if ckpt_dir and os.path.isdir(ckpt_dir) and _is_rank0():
        shutil.rmtree(ckpt_dir, ignore_errors=True)
