"""RTR / NSD solver tests."""
import numpy as np
import pytest
import torch

from sagecal_amd.ops import reference as R
from sagecal_amd.solvers import lm as lm_mod
from sagecal_amd.solvers import rtr


def _problem(N=8, T=3, seed=0, noise=0.0):
    """Unpolarized (scalar) coherencies: C = s*I per baseline — the case
    where the quotient by the full U(2) ambiguity is exact (the manifold
    the reference's RTR is built for; with polarized C only the global
    phase is ambiguous)."""
    rng = np.random.default_rng(seed)
    pairs = [(p, q) for p in range(N) for q in range(p + 1, N)]
    B = len(pairs) * T
    bb = torch.tensor(pairs * T)
    s = torch.tensor(rng.standard_normal(B) + 1j * rng.standard_normal(B))
    coh = s[:, None, None] * torch.eye(2, dtype=torch.complex128)
    Jt = torch.tensor(np.eye(2)[None, None] + 0.25 * (
        rng.standard_normal((1, N, 2, 2))
        + 1j * rng.standard_normal((1, N, 2, 2))))
    x = R.apply_jones(coh, Jt, bb)
    if noise:
        x = x + noise * torch.tensor(
            rng.standard_normal(x.shape) + 1j * rng.standard_normal(x.shape))
    return x, coh, bb, Jt, N


def test_proj_removes_vertical():
    """Projection leaves horizontal directions and kills J*Om (vertical)."""
    rng = np.random.default_rng(1)
    J = torch.tensor(rng.standard_normal((2, 5, 2, 2))
                     + 1j * rng.standard_normal((2, 5, 2, 2)))
    Om = torch.tensor(rng.standard_normal((2, 2, 2))
                      + 1j * rng.standard_normal((2, 2, 2)))
    # make Om anti-Hermitian w.r.t. the metric: vertical directions are
    # J*Om with Om s.t. X^H X Om + Om X^H X Hermitian-skew... use the
    # definition directly: proj of a vertical vector must vanish
    Om = 0.5 * (Om - Om.conj().transpose(-1, -2))   # skew-Hermitian
    V = torch.einsum('nsij,njk->nsik', J, Om)
    PV = rtr._proj(J, V)
    assert float(PV.abs().max()) < 1e-8
    # idempotency
    Z = torch.tensor(rng.standard_normal(J.shape)
                     + 1j * rng.standard_normal(J.shape))
    P1 = rtr._proj(J, Z)
    P2 = rtr._proj(J, P1)
    torch.testing.assert_close(P1, P2, atol=1e-8, rtol=1e-8)


def test_rtr_converges():
    """Warm-started RTR (the SAGE loop always warm-starts it) converges to
    the noiseless optimum."""
    x, coh, bb, Jt, N = _problem()
    prob = lm_mod.LMProblem(x, coh, bb, N, 1, None)
    rng = np.random.default_rng(7)
    J0 = Jt + 0.1 * torch.tensor(rng.standard_normal(Jt.shape)
                                 + 1j * rng.standard_normal(Jt.shape))
    J, info = rtr.rtr_solve(prob, J0, maxiter=30, rsd_iters=2)
    assert float(info['final_cost'][0]) < 1e-8 * float(info['init_cost'][0])


def test_rtr_cold_start_reduces():
    """Cold start: monotone large reduction (may stop at a local basin,
    as the reference's RSD+RTR does; the SAGE EM loop provides warmth)."""
    x, coh, bb, Jt, N = _problem()
    prob = lm_mod.LMProblem(x, coh, bb, N, 1, None)
    J0 = torch.eye(2, dtype=torch.complex128).expand(1, N, 2, 2).clone()
    J, info = rtr.rtr_solve(prob, J0, maxiter=25, rsd_iters=4)
    assert float(info['final_cost'][0]) < 0.15 * float(info['init_cost'][0])


def test_rtr_robust_weighted():
    x, coh, bb, Jt, N = _problem(noise=1e-3, seed=3)
    w = torch.ones(x.shape[0], dtype=torch.float64)
    prob = lm_mod.LMProblem(x, coh, bb, N, 1, None, weights=w)
    rng = np.random.default_rng(8)
    J0 = Jt + 0.1 * torch.tensor(rng.standard_normal(Jt.shape)
                                 + 1j * rng.standard_normal(Jt.shape))
    J, info = rtr.rtr_solve(prob, J0, maxiter=25)
    assert float(info['final_cost'][0]) < 1e-3 * float(info['init_cost'][0])


def test_nsd_reduces_cost():
    x, coh, bb, Jt, N = _problem(seed=4)
    prob = lm_mod.LMProblem(x, coh, bb, N, 1, None)
    rng = np.random.default_rng(9)
    J0 = Jt + 0.15 * torch.tensor(rng.standard_normal(Jt.shape)
                                  + 1j * rng.standard_normal(Jt.shape))
    J, info = rtr.nsd_solve(prob, J0, maxiter=50)
    assert float(info['final_cost'][0]) < 0.05 * float(info['init_cost'][0])


def test_rtr_admm_pulls_to_target():
    """RTR with consensus terms: large rho pins the solution near BZ."""
    x, coh, bb, Jt, N = _problem(seed=5)
    BZ = Jt.clone()
    Y = torch.zeros_like(BZ)
    rho = torch.tensor([1e6])
    prob = lm_mod.LMProblem(x, coh, bb, N, 1, None,
                            admm=(rho, Y, BZ))
    rng = np.random.default_rng(9)
    J0 = Jt + 0.3 * torch.tensor(rng.standard_normal(Jt.shape)
                                 + 1j * rng.standard_normal(Jt.shape))
    J, info = rtr.rtr_solve(prob, J0, maxiter=20)
    # the projection leaves the vertical (unitary) frame free: compare
    # after Procrustes alignment (the reference pairs RTR-ADMM with the
    # manifold-average step for exactly this reason)
    from sagecal_amd.consensus import manifold
    A = sum(J[0, s].conj().T @ BZ[0, s] for s in range(N))
    U = manifold.polar_unitary(A)
    err = float((torch.einsum('sij,jk->sik', J[0], U) - BZ[0]).abs().mean())
    assert err < 0.05, err


def test_rtr_body_matches_eager():
    """The capture-safe poll-free RTR body (_rtr_body, used under graph
    capture) must converge like the eager rtr_solve on the same
    problem (run eagerly on CPU here; the graph mechanics themselves are
    GPU-only)."""
    from sagecal_amd.solvers import rtr as rtr_mod
    x, coh, bb, Jt, N = _problem()
    prob = lm_mod.LMProblem(x, coh, bb, N, 1, None)
    rng = np.random.default_rng(7)
    J0 = Jt + 0.1 * torch.tensor(rng.standard_normal(Jt.shape)
                                 + 1j * rng.standard_normal(Jt.shape))
    J_eager, info = rtr_mod.rtr_solve(prob, J0, maxiter=12)
    iw = rtr_mod._station_iw(prob.bb, prob.N, prob.x.device,
                             prob.x.real.dtype)
    J_body, cost, init_cost = rtr_mod._rtr_body(
        prob.x, prob.coh, prob.bb, prob.N, prob.nchunk, prob.chunk_rows,
        prob.weights, prob.layout, J0, iw, 12, 2, 12, prob.admm)
    assert float(cost.max()) <= 1.05 * float(
        info['final_cost'].max()) + 1e-6
    assert float(cost.max()) < 1e-6 * float(init_cost.max())
