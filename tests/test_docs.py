"""Documentation snippets must execute (the reference mdoc-checks its
docs in the sbt `docs` project — build.sbt).

All ```python blocks of a file run SEQUENTIALLY in one namespace (later
blocks continue from earlier ones, as in the rendered docs), on CPU,
with row counts shrunk and /tmp save paths redirected.  Multi-process
blocks (torchrun / RCCL) are skipped — tests/test_dist* covers them.
"""

import os
import re

import pytest
import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

FILES = [
    "README.md",
    "docs/overview.md",
    "docs/example.md",
    "docs/bagging.md",
    "docs/boosting.md",
    "docs/gbm.md",
    "docs/stacking.md",
    "docs/tuning.md",
]

SKIP_MARKERS = ("torch.distributed", "torchrun", "init_from_env",
                "local_rank")


def _shrink(code: str, tmp: str) -> str:
    code = re.sub(r"\b(\d{1,3})_(\d{3})_(\d{3})\b", "3000", code)
    code = re.sub(r"\b(\d{3})_(\d{3})\b", "3000", code)
    # CPU-check scale: many-learner ensembles -> few learners
    code = re.sub(r"setNumBaseLearners\((\d+)\)",
                  lambda m: f"setNumBaseLearners({min(int(m.group(1)), 4)})",
                  code)
    code = re.sub(r"setNumFolds\((\d+)\)", "setNumFolds(2)", code)
    code = re.sub(r"numFolds=(\d+)", "numFolds=2", code)
    code = re.sub(r'addGrid\("numBaseLearners", \[[^\]]*\]\)',
                  'addGrid("numBaseLearners", [2, 3])', code)
    code = re.sub(r"setMaxIter\((\d+)\)",
                  lambda m: f"setMaxIter({min(int(m.group(1)), 10)})", code)
    code = code.replace('device="cuda:0"', 'device="cpu"')
    code = code.replace('"/tmp/', f'"{tmp}/')
    return code


PRELUDE = '''
import torch
import spark_ensemble_amd as sea
from spark_ensemble_amd.frame import TensorFrame
from spark_ensemble_amd.utils.io import (
    synthetic_classification, synthetic_regression,
)
df = synthetic_classification(3000, 16, k=3, seed=0)
train = synthetic_classification(3000, 16, k=3, seed=1)
test = synthetic_classification(1000, 16, k=3, seed=2)
dfr = synthetic_regression(3000, 16, seed=0)
'''


@pytest.mark.parametrize("rel", [f for f in FILES
                                 if os.path.exists(os.path.join(ROOT, f))])
def test_doc_snippets_run(rel, tmp_path, monkeypatch):
    text = open(os.path.join(ROOT, rel)).read()
    blocks = re.findall(r"```python\n(.*?)```", text, re.S)
    assert blocks, f"{rel} has no python blocks"
    env = {"__name__": "__doc_snippet__", "torch": torch}
    # the per-algorithm guides are fragments continuing an ambient
    # session (the reference's mdoc does the same with a shared prelude)
    exec(compile(PRELUDE, "prelude", "exec"), env)
    monkeypatch.chdir(tmp_path)  # snippets may save to relative paths
    ran = 0
    for i, code in enumerate(blocks):
        if any(s in code for s in SKIP_MARKERS):
            continue
        exec(compile(_shrink(code, str(tmp_path)), f"{rel}#{i}", "exec"),
             env)
        ran += 1
    assert ran > 0
