"""Loader for the in-tree gfx950 HIP extension.

The extension is built IN-TREE (spark_ensemble_amd/_hip_ops.<abi>.so) by
``python setup.py build_ext --inplace`` (driven by torch.utils.cpp_extension
with PYTORCH_ROCM_ARCH=gfx950) so the .so travels to GPU boxes with the
repo snapshot.  No JIT cache under $HOME is involved.
"""

from __future__ import annotations

import importlib


def load():
    return importlib.import_module("spark_ensemble_amd._hip_ops")
