"""L-BFGS and OWL-QN in plain torch, SPMD-deterministic.

Replaces the reference's in-cuML L-BFGS/OWL-QN solvers (reference
classification.py:1046-1052 passes linesearch_max_iter=20, lbfgs_memory=10).
The closure returns (loss, grad) that are ALREADY all-reduced, so every rank
runs bit-identical optimizer steps — the only communication per iteration is
the one fused gradient+loss all-reduce inside the closure.
"""

from __future__ import annotations

from typing import Callable, Optional, Tuple

import torch


def lbfgs(
    w0: torch.Tensor,
    closure: Callable[[torch.Tensor], Tuple[float, torch.Tensor]],
    max_iter: int = 100,
    tol: float = 1e-6,
    history: int = 10,
    l1_strength: Optional[torch.Tensor] = None,
    linesearch_max_iter: int = 20,
) -> Tuple[torch.Tensor, float, int, list]:
    """Minimize f(w) (+ sum(l1_strength*|w|) when given -> OWL-QN).

    closure: w -> (smooth loss, smooth grad).
    Returns (w, final_loss_including_l1, n_iters, objective_history) —
    the history mirrors Spark's objectiveHistory (reference
    tests_large/test_large_logistic_regression.py compares it).
    Convergence: relative objective decrease < tol (Spark's criterion family).
    """
    w = w0.clone().to(torch.float64)
    use_l1 = l1_strength is not None and bool((l1_strength > 0).any())
    if use_l1:
        l1 = l1_strength.to(torch.float64)

    def full_obj(loss: float, wv: torch.Tensor) -> float:
        if use_l1:
            return loss + float((l1 * wv.abs()).sum().item())
        return loss

    loss, grad = closure(w)
    grad = grad.to(torch.float64)
    obj = full_obj(loss, w)
    obj_history = [obj]

    s_hist: list = []
    y_hist: list = []
    rho_hist: list = []
    n_iter = 0

    for it in range(max_iter):
        n_iter = it + 1
        if use_l1:
            # pseudo-gradient (OWL-QN)
            pg = grad.clone()
            nz = w != 0
            pg[nz] += l1[nz] * torch.sign(w[nz])
            zero = ~nz
            gp = grad + l1
            gm = grad - l1
            pg[zero] = torch.where(
                gm[zero] > 0, gm[zero], torch.where(gp[zero] < 0, gp[zero], torch.zeros_like(gp[zero]))
            )
            g_eff = pg
        else:
            g_eff = grad

        gnorm = float(g_eff.norm().item())
        if gnorm < 1e-14:
            break

        # two-loop recursion
        q = g_eff.clone()
        alphas = []
        for s, y, rho in zip(reversed(s_hist), reversed(y_hist), reversed(rho_hist)):
            a = rho * float((s @ q).item())
            alphas.append(a)
            q -= a * y
        if y_hist:
            ys = float((s_hist[-1] @ y_hist[-1]).item())
            yy = float((y_hist[-1] @ y_hist[-1]).item())
            q *= ys / max(yy, 1e-300)
        for (s, y, rho), a in zip(zip(s_hist, y_hist, rho_hist), reversed(alphas)):
            b = rho * float((y @ q).item())
            q += (a - b) * s
        d = -q

        if use_l1:
            # project direction onto the pseudo-gradient descent orthant
            d = torch.where(d * (-g_eff) > 0, d, torch.zeros_like(d))
            orthant = torch.where(w != 0, torch.sign(w), torch.sign(-g_eff))

        # backtracking line search (Armijo on the full objective)
        dg = float((g_eff @ d).item())
        if dg >= 0:  # not a descent direction: reset
            d = -g_eff
            dg = -gnorm * gnorm
            s_hist.clear()
            y_hist.clear()
            rho_hist.clear()
        step = 1.0 if y_hist else min(1.0, 1.0 / max(gnorm, 1e-12))
        c1 = 1e-4
        ok = False
        for _ in range(linesearch_max_iter):
            w_new = w + step * d
            if use_l1:
                w_new = torch.where(
                    torch.sign(w_new) == orthant, w_new, torch.zeros_like(w_new)
                )
            loss_new, grad_new = closure(w_new)
            obj_new = full_obj(loss_new, w_new)
            if obj_new <= obj + c1 * step * dg or obj_new < obj:
                ok = True
                break
            step *= 0.5
        if not ok:
            break

        grad_new = grad_new.to(torch.float64)
        s_vec = w_new - w
        y_vec = grad_new - grad
        sy = float((s_vec @ y_vec).item())
        if sy > 1e-12:
            s_hist.append(s_vec)
            y_hist.append(y_vec)
            rho_hist.append(1.0 / sy)
            if len(s_hist) > history:
                s_hist.pop(0)
                y_hist.pop(0)
                rho_hist.pop(0)

        rel = abs(obj - obj_new) / max(abs(obj), abs(obj_new), 1.0)
        w, grad, loss, obj = w_new, grad_new, loss_new, obj_new
        obj_history.append(obj)
        if rel < tol:
            break

    return w, obj, n_iter, obj_history
