"""MLlib-layout-compatible persistence.

Reproduces the directory contract the reference inherits from Spark ML
(reference ensembleParams.scala:87-103,107-146,148-194 and e.g.
GBMRegressor.scala:563-605):

    <path>/metadata/part-00000      one JSON line: class, uid, params, ...
    <path>/learner/                 nested base-learner estimator save
    <path>/stacker/                 nested stacker estimator save
    <path>/learner-<i>/             nested estimators (HasBaseLearners)
    <path>/model-<i>/               nested fitted models
    <path>/model-<i>-<k>/           GBM classifier two-level nesting
    <path>/data-<i>/part-00000      per-learner JSON rows (weight, subspace)
    <path>/data/                    tensor payloads (torch.save)
    <path>/init/                    GBM init model

Loading is reflective like Spark's ``DefaultParamsReader.loadParamsInstance``:
the metadata's class name is imported and instantiated.
"""

from __future__ import annotations

import importlib
import json
import os
import time
from typing import Any, Dict, Optional

import torch

FRAMEWORK_VERSION = "0.1.0"


def _metadata_path(path: str) -> str:
    return os.path.join(path, "metadata", "part-00000")


def save_metadata(instance, path: str, extra: Optional[Dict[str, Any]] = None):
    """Write ``<path>/metadata/part-00000`` (one JSON object per line, like
    a Spark text dataset with a single partition)."""
    os.makedirs(os.path.join(path, "metadata"), exist_ok=True)
    cls = type(instance)
    meta = {
        "class": f"{cls.__module__}.{cls.__qualname__}",
        "timestamp": int(time.time() * 1000),
        "frameworkVersion": FRAMEWORK_VERSION,
        "uid": instance.uid,
        "paramMap": instance._jsonParams(),
        "defaultParamMap": {
            k: v
            for k, v in instance._defaultParamMap.items()
            if k not in instance._NESTED_PARAM_NAMES
            and _json_safe(v)
        },
    }
    if extra:
        meta.update(extra)
    with open(_metadata_path(path), "w") as f:
        f.write(json.dumps(meta) + "\n")
    # Spark writes an empty _SUCCESS marker per dataset; keep it for layout
    # parity with MLlib tooling that checks for it.
    open(os.path.join(path, "metadata", "_SUCCESS"), "w").close()


def _json_safe(v) -> bool:
    try:
        json.dumps(v)
        return True
    except (TypeError, ValueError):
        return False


def load_metadata(path: str) -> Dict[str, Any]:
    with open(_metadata_path(path)) as f:
        return json.loads(f.readline())


def load_class(qualname: str):
    module, _, name = qualname.rpartition(".")
    mod = importlib.import_module(module)
    obj = mod
    for part in name.split("."):
        obj = getattr(obj, part)
    return obj


def load_instance(path: str):
    """Reflectively load any saved Params instance (estimator or model)."""
    meta = load_metadata(path)
    cls = load_class(meta["class"])
    return cls._load_from(path, meta)


def save_tensors(path: str, tensors: Dict[str, torch.Tensor]):
    os.makedirs(path, exist_ok=True)
    cpu = {k: v.detach().cpu() for k, v in tensors.items()}
    torch.save(cpu, os.path.join(path, "tensors.pt"))


def load_tensors(path: str, device=None) -> Dict[str, torch.Tensor]:
    out = torch.load(os.path.join(path, "tensors.pt"), map_location="cpu",
                     weights_only=True)
    if device is not None:
        out = {k: v.to(device) for k, v in out.items()}
    return out


def save_json_rows(path: str, rows):
    """``data-<i>`` style dataset: one JSON object per line."""
    os.makedirs(path, exist_ok=True)
    with open(os.path.join(path, "part-00000"), "w") as f:
        for r in rows:
            f.write(json.dumps(r) + "\n")
    open(os.path.join(path, "_SUCCESS"), "w").close()


def load_json_rows(path: str):
    with open(os.path.join(path, "part-00000")) as f:
        return [json.loads(line) for line in f if line.strip()]
