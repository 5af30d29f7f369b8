"""Objective gradient correctness — analytic gradients checked against
torch autograd of the corresponding loss (reference analog:
tests/cpp/objective/*.cc)."""
import numpy as np
import pytest
import torch

from xgboost_amd.data import MetaInfo
from xgboost_amd.objectives import create_objective


def _info(y, n, weights=None, **kw):
    info = MetaInfo(num_row=n, num_col=1, labels=np.asarray(y, np.float32))
    if weights is not None:
        info.weights = np.asarray(weights, np.float32)
    for k, v in kw.items():
        setattr(info, k, v)
    return info


def _check_autograd(obj, loss_fn, y, preds=None, atol=1e-4, h_atol=None):
    n = len(y)
    info = _info(y, n)
    p = torch.tensor(preds if preds is not None
                     else np.random.RandomState(0).randn(n),
                     dtype=torch.float64).view(n, 1).requires_grad_(True)
    loss = loss_fn(p.view(-1), torch.tensor(np.asarray(y, np.float64)))
    g_auto = torch.autograd.grad(loss.sum(), p)[0].view(-1)
    g, h = obj.get_gradient(p.detach().float(), info, 0)
    assert torch.allclose(g.view(-1).double(), g_auto, atol=atol), \
        f"grad mismatch: {g.view(-1)[:4]} vs {g_auto[:4]}"
    assert (h > 0).all()


def test_squarederror_gradient():
    obj = create_objective("reg:squarederror")
    _check_autograd(obj, lambda p, y: 0.5 * (p - y) ** 2,
                    np.random.RandomState(1).randn(50))


def test_logistic_gradient():
    obj = create_objective("binary:logistic")
    y = (np.random.RandomState(1).rand(50) > 0.5).astype(np.float64)
    _check_autograd(
        obj, lambda p, t: torch.nn.functional.binary_cross_entropy_with_logits(
            p, t, reduction="none"), y)


def test_poisson_gradient():
    obj = create_objective("count:poisson")
    y = np.random.RandomState(1).poisson(3.0, 50).astype(np.float64)
    _check_autograd(obj, lambda p, t: torch.exp(p) - t * p, y)


def test_gamma_gradient():
    obj = create_objective("reg:gamma")
    y = np.random.RandomState(1).gamma(2.0, 1.0, 50) + 0.1
    _check_autograd(obj, lambda p, t: p + t * torch.exp(-p), y)


def test_tweedie_gradient():
    rho = 1.3
    obj = create_objective("reg:tweedie", {"tweedie_variance_power": rho})
    y = np.random.RandomState(1).gamma(2.0, 1.0, 50)
    _check_autograd(
        obj, lambda p, t: -t * torch.exp((1 - rho) * p) / (1 - rho)
        + torch.exp((2 - rho) * p) / (2 - rho), y)


def test_pseudohuber_gradient():
    obj = create_objective("reg:pseudohubererror", {"huber_slope": 2.0})
    y = np.random.RandomState(1).randn(50)
    s = 2.0
    _check_autograd(
        obj, lambda p, t: s * s * (torch.sqrt(1 + ((p - t) / s) ** 2) - 1), y)


def test_squaredlogerror_gradient():
    obj = create_objective("reg:squaredlogerror")
    y = np.abs(np.random.RandomState(1).randn(50)) + 0.5
    preds = np.abs(np.random.RandomState(2).randn(50)) + 0.5
    _check_autograd(
        obj, lambda p, t: 0.5 * (torch.log1p(p) - torch.log1p(t)) ** 2,
        y, preds=preds)


def test_quantile_gradient_signs():
    obj = create_objective("reg:quantileerror", {"quantile_alpha": 0.9})
    y = np.zeros(4, np.float32)
    p = torch.tensor([[-1.0], [1.0], [-2.0], [2.0]])
    info = _info(y, 4)
    g, h = obj.get_gradient(p, info, 0)
    # below target: gradient -alpha; above: 1-alpha
    assert np.allclose(g.view(-1).numpy(), [-0.9, 0.1, -0.9, 0.1], atol=1e-6)


def test_hinge():
    obj = create_objective("binary:hinge")
    y = np.array([0, 0, 1, 1], np.float32)
    p = torch.tensor([[-2.0], [0.5], [0.5], [2.0]])
    g, h = obj.get_gradient(p, _info(y, 4), 0)
    assert g[0, 0] == 0.0    # correct w/ margin
    assert g[1, 0] == 1.0    # inside margin, label -1
    assert g[2, 0] == -1.0
    assert g[3, 0] == 0.0


def test_softmax_gradient():
    obj = create_objective("multi:softprob", {"num_class": 3})
    y = np.array([0, 1, 2, 1], np.float32)
    info = _info(y, 4)
    p = torch.randn(4, 3, dtype=torch.float64).requires_grad_(True)
    loss = torch.nn.functional.cross_entropy(
        p, torch.tensor([0, 1, 2, 1]), reduction="sum")
    g_auto = torch.autograd.grad(loss, p)[0]
    g, h = obj.get_gradient(p.detach().float(), info, 0)
    assert torch.allclose(g.double(), g_auto, atol=1e-4)


def test_absoluteerror_gradient_direction():
    obj = create_objective("reg:absoluteerror")
    y = np.array([0.0, 0.0], np.float32)
    p = torch.tensor([[1.0], [-1.0]])
    g, h = obj.get_gradient(p, _info(y, 2), 0)
    assert g[0, 0] > 0 and g[1, 0] < 0
    assert (h > 0).all()


def test_weights_scale_gradients():
    obj = create_objective("reg:squarederror")
    y = np.ones(3, np.float32)
    p = torch.zeros(3, 1)
    g1, h1 = obj.get_gradient(p, _info(y, 3), 0)
    g2, h2 = obj.get_gradient(p, _info(y, 3, weights=[2.0, 2.0, 2.0]), 0)
    assert torch.allclose(g2, 2 * g1)
    assert torch.allclose(h2, 2 * h1)


def test_init_estimation():
    y = np.array([1.0, 2.0, 3.0, 4.0], np.float32)
    obj = create_objective("reg:squarederror")
    assert abs(obj.init_estimation(_info(y, 4)) - 2.5) < 1e-5
    objl = create_objective("binary:logistic")
    yb = np.array([0, 0, 0, 1], np.float32)
    bs = objl.init_estimation(_info(yb, 4))
    assert 0.0 < bs < 0.5


def test_scale_pos_weight():
    obj = create_objective("binary:logistic", {"scale_pos_weight": 4.0})
    y = np.array([0.0, 1.0], np.float32)
    p = torch.zeros(2, 1)
    g, h = obj.get_gradient(p, _info(y, 2), 0)
    assert abs(g[1, 0]) == pytest.approx(4 * abs(g[0, 0]), rel=1e-5)


def test_aft_uncensored():
    obj = create_objective("survival:aft")
    y = np.array([1.0, 2.0, 5.0], np.float32)
    info = _info(y, 3)
    p = torch.zeros(3, 1)
    g, h = obj.get_gradient(p, info, 0)
    assert (h > 0).all()
    # under-prediction (time > exp(0)=1) -> negative gradient (push up)
    assert g[2, 0] < 0


def test_unknown_objective_raises():
    with pytest.raises(ValueError):
        create_objective("not:a:loss")


def test_cox_gradient_matches_autograd():
    """survival:cox gradient vs torch autograd of the Breslow partial
    likelihood (negative labels = censored, reference convention)."""
    rng = np.random.RandomState(0)
    n = 60
    t = rng.exponential(2, n).astype(np.float32)
    event = rng.rand(n) > 0.3
    y = np.where(event, t, -t).astype(np.float32)
    info = MetaInfo()
    info.labels = y
    info.num_row = n
    obj = create_objective("survival:cox")
    margin = torch.tensor(rng.randn(n, 1).astype(np.float32))
    g, _ = obj.get_gradient(margin, info, 0)
    m = margin.double().clone().requires_grad_(True)
    times = torch.tensor(np.abs(y).astype(np.float64))
    exp_m = torch.exp(m.view(-1))
    nll = torch.zeros((), dtype=torch.float64)
    for i in range(n):
        if not event[i]:
            continue
        risk = times >= times[i]
        nll = nll - (m.view(-1)[i] - torch.log(exp_m[risk].sum()))
    nll.backward()
    assert np.abs(g.view(-1).numpy()
                  - m.grad.view(-1).numpy()).max() < 1e-5


@pytest.mark.parametrize("dist", ["normal", "logistic", "extreme"])
def test_aft_interval_gradient_matches_autograd(dist):
    """survival:aft censored-interval gradients vs torch autograd of
    the interval likelihood CDF(zu) - CDF(zl) (probability_distribution
    .h formulas).  Rows whose interval likelihood underflows float32
    are excluded (cancellation noise, same as the reference's fp32
    evaluation)."""
    rng = np.random.RandomState(0)
    n = 50
    lb = rng.exponential(2, n).astype(np.float32) + 0.1
    ub = lb + rng.exponential(1, n).astype(np.float32)
    ub[::5] = np.inf
    lb2 = lb.copy()
    lb2[::7] = 0.0
    info = MetaInfo()
    info.num_row = n
    info.labels = lb2
    info.label_lower_bound = lb2
    info.label_upper_bound = ub
    obj = create_objective("survival:aft", {
        "aft_loss_distribution": dist,
        "aft_loss_distribution_scale": 1.1})
    margin = torch.tensor(rng.randn(n, 1).astype(np.float32))
    g, _ = obj.get_gradient(margin, info, 0)
    m = margin.double().clone().requires_grad_(True)
    s = 1.1
    yl = torch.tensor(lb2.astype(np.float64)).clamp(min=1e-12)
    yu = torch.tensor(ub.astype(np.float64))
    zl = (torch.log(yl) - m.view(-1)) / s
    zu = (torch.log(yu.clamp(max=1e12)) - m.view(-1)) / s
    if dist == "normal":
        def cdf(z):
            return 0.5 * (1 + torch.erf(z / np.sqrt(2)))
    elif dist == "logistic":
        cdf = torch.sigmoid
    else:
        def cdf(z):
            return 1 - torch.exp(-torch.exp(z))
    like = torch.where(torch.tensor(np.isfinite(ub)),
                       cdf(zu) - cdf(zl), 1 - cdf(zl))
    nll = -torch.log(like.clamp(min=1e-300)).sum()
    nll.backward()
    ok = like.detach().numpy() > 1e-5
    diff = np.abs(g.view(-1).numpy() - m.grad.view(-1).numpy())[ok]
    assert diff.max() < 5e-3, diff.max()
