"""L-BFGS baseline with the reference's weight-history hook.

The reference adds ``getAllWeights`` to BOTH of MLlib's optimizers — the
SGD path (GradientDescent.scala:154-157, the ``Warray`` every-100-iters
history) AND LBFGS (mllib/.../optimization/LBFGS.scala:45-46) — so
loss-vs-time curves can be plotted for either baseline. The MI355X
framework's SGD history lives in ``Server.opt_vars``; this module is the
LBFGS analog: a torch two-loop-recursion L-BFGS over the full-batch LSQ /
logistic objective, recording ``(ms_since_start, w)`` every
``printer_freq`` iterations and exposing ``get_all_weights()``.

Runs on CPU or GPU tensors (full-batch gradient = one GEMV per iteration —
library territory, not a custom kernel)."""

from __future__ import annotations

import time
from typing import List, Optional, Tuple

import torch


def _loss_grad(X: torch.Tensor, y: torch.Tensor, w: torch.Tensor,
               objective: str) -> Tuple[float, torch.Tensor]:
    Xf = X if X.dtype == torch.float32 else X.float()
    z = Xf @ w
    if objective == "logistic":
        zy = z * (2.0 * y - 1.0)
        loss = torch.nn.functional.softplus(-zy).mean()
        p = torch.sigmoid(z)
        g = Xf.t() @ (p - y) / X.shape[0]
    else:
        r = z - y
        loss = (r * r).mean()
        g = 2.0 * (Xf.t() @ r) / X.shape[0]
    return float(loss), g


class LBFGS:
    """Two-loop-recursion L-BFGS with backtracking Armijo line search.

    API mirrors the reference hook surface: ``optimize()`` returns the
    final weights; ``get_all_weights()`` returns the recorded
    ``[(ms, w_cpu), ...]`` history (reference ``Optimizer.getAllWeights``,
    optimization/Optimizer.scala:39-40)."""

    def __init__(self, memory: int = 10, max_iter: int = 100,
                 tol: float = 1e-8, printer_freq: int = 100,
                 objective: str = "lsq"):
        self.m = memory
        self.max_iter = max_iter
        self.tol = tol
        self.printer_freq = max(1, printer_freq)
        self.objective = objective
        self._hist: List[Tuple[int, torch.Tensor]] = []

    def get_all_weights(self) -> List[Tuple[int, torch.Tensor]]:
        return list(self._hist)

    def optimize(self, X: torch.Tensor, y: torch.Tensor,
                 w0: Optional[torch.Tensor] = None) -> torch.Tensor:
        t0 = time.perf_counter()
        d = X.shape[1]
        w = (w0.clone().float() if w0 is not None
             else torch.zeros(d, dtype=torch.float32, device=X.device))
        self._hist = [(0, w.detach().cpu().clone())]
        S: List[torch.Tensor] = []
        Y: List[torch.Tensor] = []
        loss, g = _loss_grad(X, y, w, self.objective)
        for it in range(self.max_iter):
            if float(g.norm()) < self.tol:
                break
            # two-loop recursion
            q = g.clone()
            alphas = []
            for s, yv in zip(reversed(S), reversed(Y)):
                rho = 1.0 / float(yv.dot(s))
                a = rho * float(s.dot(q))
                alphas.append((a, rho, s, yv))
                q -= a * yv
            if S:
                gamma = float(S[-1].dot(Y[-1])) / float(Y[-1].dot(Y[-1]))
                q *= gamma
            for a, rho, s, yv in reversed(alphas):
                b = rho * float(yv.dot(q))
                q += (a - b) * s
            p = -q
            # Armijo backtracking
            step, c1 = 1.0, 1e-4
            gTp = float(g.dot(p))
            if gTp >= 0:  # non-descent (numerical): reset memory
                S.clear()
                Y.clear()
                p = -g
                gTp = -float(g.dot(g))
            while step > 1e-12:
                new_loss, new_g = _loss_grad(X, y, w + step * p,
                                             self.objective)
                if new_loss <= loss + c1 * step * gTp:
                    break
                step *= 0.5
            w_new = w + step * p
            s = w_new - w
            yv = new_g - g
            if float(yv.dot(s)) > 1e-12:  # curvature condition
                S.append(s)
                Y.append(yv)
                if len(S) > self.m:
                    S.pop(0)
                    Y.pop(0)
            w, loss, g = w_new, new_loss, new_g
            if (it + 1) % self.printer_freq == 0:
                ms = int((time.perf_counter() - t0) * 1000)
                self._hist.append((ms, w.detach().cpu().clone()))
        ms = int((time.perf_counter() - t0) * 1000)
        self._hist.append((ms, w.detach().cpu().clone()))
        return w
