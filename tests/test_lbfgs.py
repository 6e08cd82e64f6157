"""LBFGS baseline + history hook (reference LBFGS.scala:45-46 /
Optimizer.scala:39-40 getAllWeights)."""

import numpy as np
import torch

from asyncframework_amd.algos import LBFGS
from asyncframework_amd.data.synthetic import synthetic_dense


def _data(objective, n=2000, d=30, seed=0):
    return synthetic_dense(n, d, seed=seed, dtype=torch.float32,
                           device=torch.device("cpu"), objective=objective)


def test_lbfgs_solves_lsq_to_least_squares_solution():
    X, y = _data("lsq")
    w = LBFGS(max_iter=200, printer_freq=10).optimize(X, y)
    # compare against the closed-form least-squares solution
    w_star = torch.linalg.lstsq(X, y.unsqueeze(1)).solution.squeeze(1)
    r = float(((X @ w - y) ** 2).mean())
    r_star = float(((X @ w_star - y) ** 2).mean())
    assert r <= r_star * 1.01 + 1e-8


def test_lbfgs_logistic_matches_scipy():
    X, y = _data("logistic")
    opt = LBFGS(max_iter=150, printer_freq=25, objective="logistic")
    w = opt.optimize(X, y)

    from scipy.optimize import minimize
    Xn, yn = X.numpy().astype(np.float64), y.numpy().astype(np.float64)

    def f(v):
        z = Xn @ v
        zy = z * (2 * yn - 1)
        return float(np.mean(np.logaddexp(0.0, -zy)))

    res = minimize(f, np.zeros(X.shape[1]), method="L-BFGS-B",
                   options={"maxiter": 300})
    ours = f(w.numpy().astype(np.float64))
    assert ours <= res.fun * 1.02 + 1e-6


def test_lbfgs_history_hook():
    X, y = _data("lsq")
    opt = LBFGS(max_iter=60, printer_freq=10)
    w = opt.optimize(X, y)
    hist = opt.get_all_weights()
    assert len(hist) >= 3           # initial + periodic + final
    assert hist[0][0] == 0
    assert torch.all(hist[0][1] == 0)
    assert torch.allclose(hist[-1][1], w.cpu())
    ts = [t for t, _ in hist]
    assert ts == sorted(ts)
    # the recorded curve is monotone-improving at the recorded points
    objs = [float(((X @ wi - y) ** 2).mean()) for _, wi in hist]
    assert objs[-1] < objs[0]
