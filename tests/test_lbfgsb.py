"""Full L-BFGS-B (Byrd-Nocedal, solvers/lbfgs.lbfgsb_fit — the reference's
lbfgsb.c) validated against scipy.optimize's reference implementation."""
import numpy as np
import pytest
import torch
from scipy.optimize import minimize

from sagecal_amd.solvers.lbfgs import lbfgsb_fit


def _fg_np(fun):
    def fg(p):
        f, g = fun(p.numpy())
        return torch.tensor(f, dtype=p.dtype), \
            torch.tensor(g, dtype=p.dtype)
    return fg


def _quad(A, b):
    def fun(x):
        return 0.5 * x @ A @ x - b @ x, A @ x - b
    return fun


def test_bounded_quadratic_matches_scipy():
    rng = np.random.default_rng(3)
    n = 12
    Q = rng.standard_normal((n, n))
    A = Q @ Q.T / n + 0.5 * np.eye(n)
    b = rng.standard_normal(n)
    lb, ub = -0.3 * np.ones(n), 0.4 * np.ones(n)
    fun = _quad(A, b)
    ref = minimize(fun, np.zeros(n), jac=True, method='L-BFGS-B',
                   bounds=list(zip(lb, ub)))
    p, info = lbfgsb_fit(_fg_np(fun), torch.zeros(n, dtype=torch.float64),
                         torch.tensor(lb), torch.tensor(ub), maxiter=200)
    assert info['f1'] <= ref.fun + 1e-8 * (1 + abs(ref.fun))
    assert np.allclose(p.numpy(), ref.x, atol=1e-5)


def test_bounded_rosenbrock_matches_scipy():
    def fun(x):
        f = 100.0 * (x[1] - x[0] ** 2) ** 2 + (1 - x[0]) ** 2
        g = np.array([-400.0 * x[0] * (x[1] - x[0] ** 2) - 2 * (1 - x[0]),
                      200.0 * (x[1] - x[0] ** 2)])
        return f, g
    # bounds exclude the unconstrained optimum (1,1)
    lb, ub = np.array([-2.0, -2.0]), np.array([0.8, 2.0])
    ref = minimize(fun, np.array([-1.2, 1.0]), jac=True, method='L-BFGS-B',
                   bounds=list(zip(lb, ub)))
    p, info = lbfgsb_fit(_fg_np(fun),
                         torch.tensor([-1.2, 1.0], dtype=torch.float64),
                         torch.tensor(lb), torch.tensor(ub), maxiter=300)
    assert info['f1'] <= ref.fun + 1e-6 * (1 + abs(ref.fun))
    # constrained optimum pins x0 at its upper bound
    assert abs(float(p[0]) - 0.8) < 1e-6


def test_active_set_identification():
    # minimizer of ||x - t||^2 with t outside the box on some coords
    t = np.array([2.0, -3.0, 0.1, 0.0, 5.0])

    def fun(x):
        return float(np.sum((x - t) ** 2)), 2.0 * (x - t)
    lb, ub = -np.ones(5), np.ones(5)
    p, info = lbfgsb_fit(_fg_np(fun), torch.zeros(5, dtype=torch.float64),
                         torch.tensor(lb), torch.tensor(ub))
    want = np.clip(t, lb, ub)
    assert np.allclose(p.numpy(), want, atol=1e-7)


def test_unbounded_reduces_to_lbfgs():
    rng = np.random.default_rng(5)
    n = 8
    Q = rng.standard_normal((n, n))
    A = Q @ Q.T / n + np.eye(n)
    b = rng.standard_normal(n)
    fun = _quad(A, b)
    big = 1e8
    p, info = lbfgsb_fit(_fg_np(fun), torch.zeros(n, dtype=torch.float64),
                         torch.full((n,), -big), torch.full((n,), big),
                         maxiter=200)
    xstar = np.linalg.solve(A, b)
    assert np.allclose(p.numpy(), xstar, atol=1e-6)
