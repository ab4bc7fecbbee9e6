"""Gradient/hessian checking of every GBM loss against finite differences
(the analog of reference GBMLossSuite.scala:84-125, which runs Breeze
GradientTester through the distributed aggregator at tolerance 1e-5)."""

import pytest
import torch

from spark_ensemble_amd.boosting.losses import (
    AbsoluteLoss,
    BernoulliLoss,
    ExponentialLoss,
    HuberLoss,
    LogCoshLoss,
    LogLoss,
    QuantileLoss,
    ScaledLogCoshLoss,
    SquaredLoss,
)

ALL_LOSSES = [
    SquaredLoss(),
    AbsoluteLoss(),
    LogCoshLoss(),
    ScaledLogCoshLoss(0.3),
    HuberLoss(0.8),
    QuantileLoss(0.7),
    LogLoss(4),
    ExponentialLoss(),
    BernoulliLoss(),
]


def _label_pred(loss, n=64, seed=0):
    g = torch.Generator().manual_seed(seed)
    if loss.name == "logloss":
        y = torch.randint(0, loss.num_classes, (n,), generator=g).double()
    elif loss.name in ("exponential", "bernoulli"):
        y = torch.randint(0, 2, (n,), generator=g).double()
    else:
        y = torch.randn(n, generator=g).double()
    label = loss.encode_label(y)
    pred = torch.randn(n, loss.dim, generator=g).double()
    return label, pred


@pytest.mark.parametrize("loss", ALL_LOSSES, ids=lambda l: l.name)
def test_gradient_finite_difference(loss):
    label, pred = _label_pred(loss)
    grad = loss.gradient(label, pred)
    eps = 1e-6
    for j in range(loss.dim):
        p_plus = pred.clone()
        p_plus[:, j] += eps
        p_minus = pred.clone()
        p_minus[:, j] -= eps
        fd = (loss.loss(label, p_plus) - loss.loss(label, p_minus)) / (2 * eps)
        # non-smooth losses (absolute/huber/quantile) can disagree exactly at
        # kinks; random continuous preds avoid them w.p. 1
        assert torch.allclose(grad[:, j], fd, atol=1e-4), loss.name


@pytest.mark.parametrize(
    "loss",
    [l for l in ALL_LOSSES if l.has_hessian],
    ids=lambda l: l.name,
)
def test_hessian_is_gradient_of_gradient(loss):
    # reference wraps the hessian as the gradient of the gradient
    # (GBMLossSuite.scala:96-105)
    label, pred = _label_pred(loss, seed=1)
    hess = loss.hessian(label, pred)
    eps = 1e-6
    for j in range(loss.dim):
        p_plus = pred.clone()
        p_plus[:, j] += eps
        p_minus = pred.clone()
        p_minus[:, j] -= eps
        fd = (
            loss.gradient(label, p_plus)[:, j] - loss.gradient(label, p_minus)[:, j]
        ) / (2 * eps)
        assert torch.allclose(hess[:, j], fd, atol=1e-4), loss.name


def test_encode_labels():
    assert LogLoss(3).encode_label(torch.tensor([0.0, 2.0])).tolist() == [
        [1.0, 0.0, 0.0],
        [0.0, 0.0, 1.0],
    ]
    assert ExponentialLoss().encode_label(torch.tensor([0.0, 1.0])).tolist() == [
        [-1.0],
        [1.0],
    ]


def test_raw2probability_shapes_and_quirks():
    raw = torch.tensor([[0.7], [-0.3]])
    pe = ExponentialLoss().raw2probability(raw)
    assert pe.shape == (2, 2)
    assert torch.allclose(pe.sum(dim=1), torch.ones(2))
    pb = BernoulliLoss().raw2probability(raw)
    # reference sign quirk: p1 uses exp(+raw) (GBMLoss.scala:311-316)
    assert torch.allclose(pb[:, 1], 1.0 / (1.0 + torch.exp(raw[:, 0]))), pb
    pl = LogLoss(3).raw2probability(torch.randn(5, 3))
    assert torch.allclose(pl.sum(dim=1), torch.ones(5), atol=1e-6)


def test_negative_gradient():
    loss = SquaredLoss()
    label, pred = _label_pred(loss, seed=2)
    assert torch.allclose(
        loss.negative_gradient(label, pred), -(loss.gradient(label, pred))
    )


def test_newton_line_search_matches_brent():
    """The safeguarded-Newton stage-weight search (smooth losses) must land
    on the same minimizer as Brent within tolerance."""
    import torch
    from spark_ensemble_amd.boosting.line_search import (
        _newton_1d, optimize_weight_1d,
    )
    from spark_ensemble_amd.boosting.losses import (
        BernoulliLoss, LogCoshLoss, SquaredLoss,
    )
    from scipy.optimize import minimize_scalar

    g = torch.Generator().manual_seed(31)
    n = 5000
    for loss in (SquaredLoss(), LogCoshLoss(), BernoulliLoss()):
        if loss.name == "bernoulli":
            y = loss.encode_label(torch.randint(0, 2, (n,), generator=g).float())
        else:
            y = torch.randn(n, generator=g)
        y = y.reshape(n, 1)
        pred = torch.randn(n, 1, generator=g) * 0.3
        direction = (y - pred) * 0.5 + torch.randn(n, 1, generator=g) * 0.1
        w = torch.rand(n, generator=g) + 0.5

        a_newton = _newton_1d(loss, y, pred, direction, w, None, 100, 1e-8,
                              0.0, 100.0)
        assert a_newton is not None

        def phi(a):
            p = pred + direction * float(a)
            return float((loss.loss(y, p) * w).sum())

        res = minimize_scalar(phi, bounds=(0.0, 100.0), method="bounded",
                              options={"xatol": 1e-10})
        assert abs(a_newton - float(res.x)) < 1e-3, (
            loss.name, a_newton, float(res.x))
        # and the public entry point routes smooth losses through Newton
        a_pub = optimize_weight_1d(loss, y, pred, direction, w, None)
        assert abs(a_pub - float(res.x)) < 1e-3
