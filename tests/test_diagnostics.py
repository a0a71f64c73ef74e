"""Influence-function, FISTA spatial model, MDL, LBFGS-B, taper tests."""
import numpy as np
import pytest
import torch

from sagecal_amd.ops import reference as R


def test_leverage_properties():
    """Leverage in [0, 8]; sums to ~#params (trace of the hat matrix)."""
    from sagecal_amd.solvers import diagnostics as diag
    rng = np.random.default_rng(0)
    N, T = 6, 4
    pairs = [(p, q) for p in range(N) for q in range(p + 1, N)]
    B = len(pairs) * T
    bb = torch.tensor(pairs * T)
    coh = torch.tensor(rng.standard_normal((B, 2, 2))
                       + 1j * rng.standard_normal((B, 2, 2)))
    J = torch.tensor(np.eye(2)[None, None] + 0.2 * (
        rng.standard_normal((1, N, 2, 2))
        + 1j * rng.standard_normal((1, N, 2, 2))))
    x = R.apply_jones(coh, J, bb)
    lev = diag.leverage(x, coh, J, bb, N, mu=1e-9)
    assert float(lev.min()) > -1e-6
    assert float(lev.max()) <= 8.0 + 1e-6
    # trace of hat matrix ~ number of effective parameters (8N - gauge)
    total = float(lev.sum())
    assert 0.5 * 8 * N < total < 8.2 * N


def test_fista_spatial_recovers_smooth_model():
    from sagecal_amd.consensus import fista
    rng = np.random.default_rng(1)
    M, P, G = 12, 10, 4
    Phi = torch.tensor(rng.standard_normal((M, G)))
    Ztrue = torch.tensor(rng.standard_normal((P, G))
                         + 1j * rng.standard_normal((P, G)))
    Zbar = (Ztrue @ Phi.T.to(Ztrue.dtype)).T   # [M, P]
    Z = fista.update_spatialreg_fista(Zbar, Phi, lam=1e-6, mu=1e-8,
                                      maxiter=400)
    err = float((Z - Ztrue).abs().max() / Ztrue.abs().max())
    assert err < 1e-2, err


def test_fista_l1_sparsifies():
    from sagecal_amd.consensus import fista
    rng = np.random.default_rng(2)
    M, P, G = 10, 6, 5
    Phi = torch.tensor(rng.standard_normal((M, G)))
    Zbar = torch.tensor(rng.standard_normal((M, P))
                        + 1j * rng.standard_normal((M, P))) * 0.01
    Z = fista.update_spatialreg_fista(Zbar, Phi, lam=0.0, mu=10.0,
                                      maxiter=100)
    assert float(Z.abs().max()) < 1e-6   # strong L1 kills tiny signal


def test_mdl_selects_true_order():
    from sagecal_amd.consensus import mdl, poly
    rng = np.random.default_rng(3)
    F, M, K = 8, 3, 16
    freqs = np.linspace(120e6, 170e6, F)
    B = poly.setup_polynomials(freqs, 145e6, 2, 0)
    Ztrue = torch.tensor(rng.standard_normal((M, 2, K))
                         + 1j * rng.standard_normal((M, 2, K)))
    Jb = torch.stack([torch.einsum('p,mpk->mk', B[f].to(Ztrue.dtype),
                                   Ztrue) for f in range(F)])
    Jb = Jb + 1e-4 * torch.tensor(rng.standard_normal(Jb.shape)
                                  + 1j * rng.standard_normal(Jb.shape))
    rho = torch.ones(M).double()
    best, scores = mdl.minimum_description_length(
        Jb, rho, freqs, 145e6, Kstart=1, Kfinish=5)
    assert best == 2, scores


def test_accel_proj_grad_quadratic():
    from sagecal_amd.consensus.fista import accel_proj_grad
    A = torch.tensor([[3.0, 1.0], [1.0, 2.0]])
    b = torch.tensor([1.0, -2.0])
    cost = lambda p: 0.5 * p @ A @ p - b @ p
    grad = lambda p: A @ p - b
    p = accel_proj_grad(cost, grad, torch.zeros(2), itmax=300)
    ref = torch.linalg.solve(A, b)
    assert float((p - ref).norm()) < 1e-3


def test_lbfgsb_respects_bounds():
    from sagecal_amd.solvers.lbfgs import lbfgs_fit_bounded
    # min (x-2)^2 + (y+3)^2 s.t. x,y in [-1, 1]
    def fg(p):
        g = 2 * (p - torch.tensor([2.0, -3.0]))
        f = ((p - torch.tensor([2.0, -3.0])) ** 2).sum()
        return f, g
    p, _, info = lbfgs_fit_bounded(fg, torch.zeros(2), [-1.0, -1.0],
                                   [1.0, 1.0], maxiter=60)
    torch.testing.assert_close(p, torch.tensor([1.0, -1.0]), atol=1e-6,
                               rtol=0)


def test_rosenbrock_lbfgs():
    """The reference's library demo (test/Dirac/demo.c:25-56): LBFGS on
    the extended Rosenbrock function converges to (1, ..., 1)."""
    from sagecal_amd.solvers.lbfgs import lbfgs_fit
    n = 8

    def fg(p):
        p = p.clone().requires_grad_(True)
        f = (100 * (p[1::2] - p[0::2] ** 2) ** 2
             + (1 - p[0::2]) ** 2).sum()
        g, = torch.autograd.grad(f, p)
        return f.detach(), g
    p0 = torch.full((n,), -1.2, dtype=torch.float64)
    p, _, info = lbfgs_fit(fg, p0, maxiter=400, m=7)
    torch.testing.assert_close(p, torch.ones(n, dtype=torch.float64),
                               atol=1e-5, rtol=0)


def test_whiten_and_ppm(tmp_path):
    from sagecal_amd.utils import taper, image
    u = torch.tensor([1e-7, 1e-6, 1e-5, 1e-3])
    v = torch.zeros(4, dtype=torch.float64)
    x = torch.ones(4, 2, 2, dtype=torch.complex128)
    xw, w = taper.whiten_data(x, u, v, 150e6)
    assert float(w[0]) < float(w[3])       # short baselines suppressed
    assert float(w[3]) == 1.0
    amp = np.abs(np.random.default_rng(0).standard_normal((16, 16)))
    p = str(tmp_path / 'model.ppm')
    image.write_ppm(p, amp)
    with open(p, 'rb') as f:
        assert f.read(2) == b'P6'
