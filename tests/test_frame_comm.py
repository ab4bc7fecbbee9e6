"""TensorFrame transforms + single-process Comm short-circuit coverage."""

import torch

from spark_ensemble_amd.frame import TensorFrame
from spark_ensemble_amd.parallel import Comm


def _df():
    return TensorFrame(
        features=torch.arange(12.0).reshape(6, 2),
        label=torch.arange(6.0),
    )


def test_frame_transforms():
    df = _df()
    assert df.columns == ["features", "label"]
    assert len(df) == 6 and df.num_features() == 2
    d2 = df.withColumn("w", torch.ones(6))
    assert "w" in d2 and "w" not in df
    d3 = d2.drop("w")
    assert "w" not in d3
    d4 = df.withRenamed("label", "y")
    assert "y" in d4 and "label" not in d4
    sel = df.select("label")
    assert sel.columns == ["label"]
    filt = df.filter(df["label"] >= 3)
    assert len(filt) == 3
    filt2 = df.filter(torch.tensor([0, 2]))
    assert torch.equal(filt2["label"], torch.tensor([0.0, 2.0]))


def test_frame_row_mismatch_raises():
    import pytest

    with pytest.raises(ValueError):
        TensorFrame(features=torch.zeros(3, 2), label=torch.zeros(4))


def test_frame_cache_keyed_by_source_identity():
    df = _df()
    x = df["features"]
    df.cache_put("bins", x, 32, "payload")
    assert df.cache_get("bins", x, 32) == "payload"
    assert df.cache_get("bins", x, 64) is None
    assert df.cache_get("bins", x.clone(), 32) is None
    # column transforms share the cache
    assert df.withColumn("w", torch.ones(6)).cache_get("bins", x, 32) == "payload"


def test_comm_single_process_short_circuits():
    c = Comm()
    assert not c.is_distributed
    t = torch.ones(3)
    assert c.all_reduce_(t) is t
    assert c.all_reduce_scalar(2.5) == 2.5
    assert c.all_reduce_scalar(2.5, "max") == 2.5
    assert c.all_gather_object({"a": 1}) == [{"a": 1}]
    assert c.broadcast_(t) is t
    c.barrier()  # no-op
    assert "world=1" in repr(c)
