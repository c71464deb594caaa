"""L-BFGS and OWL-QN batch optimizers.

The reference stack drives training through breeze's LBFGS/OWLQN on the JVM
driver (SURVEY.md §2.6 "breeze LBFGS/OWLQN"): LogisticRegression and the MLP
are quasi-Newton full-batch fits. Re-implemented here in torch so the history
vectors live wherever the gradient does (GPU for the real configs) and so the
loop is replicated deterministically on every rank — the objective closure
all-reduces the gradient, after which every rank takes the identical step
(SURVEY.md §1 control-flow invariant).

``lbfgs``: two-loop recursion + strong-Wolfe line search.
``owlqn``:  LBFGS on the smooth part + pseudo-gradient / orthant projection
for the L1 term (only indices >= ``l1_start`` are penalized — Spark never
regularizes intercepts).
"""

from __future__ import annotations

from typing import Callable, List, Tuple

import torch

FnGrad = Callable[[torch.Tensor], Tuple[float, torch.Tensor]]


def _dot(a: torch.Tensor, b: torch.Tensor) -> float:
    return float(torch.dot(a, b))


def _strong_wolfe(
    f_grad: FnGrad,
    x: torch.Tensor,
    f0: float,
    g0: torch.Tensor,
    d: torch.Tensor,
    t0: float = 1.0,
    c1: float = 1e-4,
    c2: float = 0.9,
    max_ls: int = 25,
) -> Tuple[float, float, torch.Tensor, int]:
    """Strong Wolfe line search. Returns (t, f_new, g_new, n_evals)."""
    dg0 = _dot(g0, d)
    if dg0 >= 0:
        # not a descent direction; fall back to steepest descent step
        return 0.0, f0, g0, 0

    t = t0
    t_prev, f_prev, dg_prev = 0.0, f0, dg0
    n_evals = 0
    t_lo = t_hi = None
    f_lo = g_lo = dg_lo = None
    f_hi = dg_hi = None
    g_t = g0
    f_t = f0

    # bracket phase
    for i in range(max_ls):
        f_t, g_t = f_grad(x + t * d)
        n_evals += 1
        dg_t = _dot(g_t, d)
        if f_t > f0 + c1 * t * dg0 or (i > 0 and f_t >= f_prev):
            t_lo, f_lo, dg_lo = t_prev, f_prev, dg_prev
            t_hi, f_hi, dg_hi = t, f_t, dg_t
            break
        if abs(dg_t) <= -c2 * dg0:
            return t, f_t, g_t, n_evals
        if dg_t >= 0:
            t_lo, f_lo, dg_lo = t, f_t, dg_t
            t_hi, f_hi, dg_hi = t_prev, f_prev, dg_prev
            break
        t_prev, f_prev, dg_prev = t, f_t, dg_t
        t = 2.0 * t
    else:
        return t, f_t, g_t, n_evals

    # zoom phase
    for _ in range(max_ls):
        t = 0.5 * (t_lo + t_hi)
        f_t, g_t = f_grad(x + t * d)
        n_evals += 1
        dg_t = _dot(g_t, d)
        if f_t > f0 + c1 * t * dg0 or f_t >= f_lo:
            t_hi, f_hi, dg_hi = t, f_t, dg_t
        else:
            if abs(dg_t) <= -c2 * dg0:
                return t, f_t, g_t, n_evals
            if dg_t * (t_hi - t_lo) >= 0:
                t_hi, f_hi, dg_hi = t_lo, f_lo, dg_lo
            t_lo, f_lo, dg_lo = t, f_t, dg_t
        if abs(t_hi - t_lo) < 1e-14:
            break
    return t, f_t, g_t, n_evals


def lbfgs(
    f_grad: FnGrad,
    x0: torch.Tensor,
    max_iter: int = 100,
    tol: float = 1e-6,
    history_size: int = 10,
) -> Tuple[torch.Tensor, List[float]]:
    """Minimize f via L-BFGS. Returns (x, objective_history).

    ``objective_history`` records f at x0 and after each accepted step —
    it becomes the Spark-compatible ``summary.objectiveHistory``.
    """
    x = x0.clone()
    f, g = f_grad(x)
    history = [f]
    s_list: List[torch.Tensor] = []
    y_list: List[torch.Tensor] = []
    rho: List[float] = []

    for it in range(max_iter):
        gnorm = float(g.norm())
        if gnorm == 0.0:
            break
        d = _two_loop(g, s_list, y_list, rho)
        t0 = 1.0 if s_list else min(1.0, 1.0 / max(gnorm, 1e-12))
        t, f_new, g_new, _ = _strong_wolfe(f_grad, x, f, g, d, t0=t0)
        if t == 0.0:
            # line search failed along quasi-Newton direction: reset memory
            if s_list:
                s_list.clear(); y_list.clear(); rho.clear()
                continue
            break
        x_new = x + t * d
        s = x_new - x
        yv = g_new - g
        sy = _dot(s, yv)
        if sy > 1e-10:
            if len(s_list) == history_size:
                s_list.pop(0); y_list.pop(0); rho.pop(0)
            s_list.append(s); y_list.append(yv); rho.append(1.0 / sy)
        # breeze-style relative improvement convergence
        improv = abs(f - f_new) / max(abs(f), abs(f_new), 1e-12)
        x, f, g = x_new, f_new, g_new
        history.append(f)
        if improv < tol:
            break
    return x, history


def _two_loop(
    g: torch.Tensor,
    s_list: List[torch.Tensor],
    y_list: List[torch.Tensor],
    rho: List[float],
) -> torch.Tensor:
    q = g.clone()
    alphas = []
    for s, y, r in zip(reversed(s_list), reversed(y_list), reversed(rho)):
        a = r * _dot(s, q)
        alphas.append(a)
        q -= a * y
    if s_list:
        gamma = _dot(s_list[-1], y_list[-1]) / max(_dot(y_list[-1], y_list[-1]), 1e-12)
        q *= gamma
    for (s, y, r), a in zip(zip(s_list, y_list, rho), reversed(alphas)):
        b = r * _dot(y, q)
        q += (a - b) * s
    return -q


# ---------------------------------------------------------------------------
# OWL-QN (L1)
# ---------------------------------------------------------------------------


def owlqn(
    f_grad: FnGrad,
    x0: torch.Tensor,
    l1_weight: torch.Tensor,
    max_iter: int = 100,
    tol: float = 1e-6,
    history_size: int = 10,
) -> Tuple[torch.Tensor, List[float]]:
    """Minimize f(x) + sum(l1_weight * |x|) via OWL-QN.

    ``l1_weight`` is a per-coordinate nonnegative tensor (0 ⇒ unpenalized,
    e.g. intercepts — Spark's LogisticRegression never regularizes them).
    """
    x = x0.clone()
    f, g = f_grad(x)

    def full_obj(xv: torch.Tensor, fv: float) -> float:
        return fv + float((l1_weight * xv.abs()).sum())

    F = full_obj(x, f)
    history = [F]
    s_list: List[torch.Tensor] = []
    y_list: List[torch.Tensor] = []
    rho: List[float] = []

    for it in range(max_iter):
        pg = _pseudo_gradient(x, g, l1_weight)
        if float(pg.norm()) == 0.0:
            break
        d = _two_loop(pg, s_list, y_list, rho)
        # align: zero components where d disagrees with -pg
        d = torch.where(d * pg < 0, d, torch.zeros_like(d))
        if float(d.norm()) == 0.0:
            break
        # choose orthant
        xi = torch.where(x != 0, torch.sign(x), -torch.sign(pg))
        dgdir = _dot(pg, d)
        t = 1.0 if s_list else min(1.0, 1.0 / max(float(pg.norm()), 1e-12))
        accepted = False
        for _ in range(30):
            x_new = x + t * d
            # orthant projection
            x_new = torch.where(x_new * xi > 0, x_new, torch.zeros_like(x_new))
            f_new, g_new = f_grad(x_new)
            F_new = full_obj(x_new, f_new)
            if F_new <= F + 1e-4 * dgdir * t:
                accepted = True
                break
            t *= 0.5
        if not accepted:
            if s_list:
                s_list.clear(); y_list.clear(); rho.clear()
                continue
            break
        s = x_new - x
        yv = g_new - g
        sy = _dot(s, yv)
        if sy > 1e-10:
            if len(s_list) == history_size:
                s_list.pop(0); y_list.pop(0); rho.pop(0)
            s_list.append(s); y_list.append(yv); rho.append(1.0 / sy)
        improv = abs(F - F_new) / max(abs(F), abs(F_new), 1e-12)
        x, f, g, F = x_new, f_new, g_new, F_new
        history.append(F)
        if improv < tol:
            break
    return x, history


def _pseudo_gradient(x: torch.Tensor, g: torch.Tensor, l1: torch.Tensor) -> torch.Tensor:
    pg = torch.where(
        x != 0,
        g + l1 * torch.sign(x),
        torch.where(
            g + l1 < 0,
            g + l1,
            torch.where(g - l1 > 0, g - l1, torch.zeros_like(g)),
        ),
    )
    return pg
