"""Property-based tests (hypothesis) for the core math transforms the
solvers are built on: vecR/realify layouts, the damped-solve padding
contract, polynomial-basis invariants, and the shapelet product-scale
algebra."""
import math

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from sagecal_amd.ops import reference as R
from sagecal_amd import shapelet
from sagecal_amd.consensus import poly


@st.composite
def jones(draw, maxn=6):
    n = draw(st.integers(1, maxn))
    seed = draw(st.integers(0, 2 ** 31 - 1))
    rng = np.random.default_rng(seed)
    return torch.tensor(rng.standard_normal((1, n, 2, 2))
                        + 1j * rng.standard_normal((1, n, 2, 2)))


@given(jones())
@settings(max_examples=30, deadline=None)
def test_vecR_roundtrip(J):
    """vecR is the row-major interleaved re/im layout; inverting it
    recovers the Jones matrices exactly."""
    v = R.vecR(J)                          # [1, N, 8]
    back = torch.view_as_complex(
        v.reshape(J.shape[0], J.shape[1], 2, 2, 2).contiguous())
    assert torch.equal(back, J)


@given(st.integers(0, 2 ** 31 - 1), st.integers(1, 4))
@settings(max_examples=20, deadline=None)
def test_realify_is_complex_matmul(seed, n):
    """realify(A) acting on vecR-stacked reals equals the complex
    product: the 2x2-complex <-> 4x4-real embedding is a homomorphism."""
    rng = np.random.default_rng(seed)
    A = torch.tensor(rng.standard_normal((n, 2, 2))
                     + 1j * rng.standard_normal((n, 2, 2)))
    x = torch.tensor(rng.standard_normal((n, 2))
                     + 1j * rng.standard_normal((n, 2)))
    Ar = R.realify(A)                       # [n, 4, 4]
    xr = torch.stack([x.real[..., 0], x.imag[..., 0],
                      x.real[..., 1], x.imag[..., 1]], dim=-1)
    yr = (Ar @ xr.unsqueeze(-1)).squeeze(-1)
    y = (A @ x.unsqueeze(-1)).squeeze(-1)
    want = torch.stack([y.real[..., 0], y.imag[..., 0],
                        y.real[..., 1], y.imag[..., 1]], dim=-1)
    assert torch.allclose(yr, want, atol=1e-12)


@given(st.integers(0, 2 ** 31 - 1), st.integers(1, 4))
@settings(max_examples=20, deadline=None)
def test_antirealify_is_antilinear_map(seed, n):
    """antirealify(A) acting on stacked reals equals x -> A conj(x)."""
    rng = np.random.default_rng(seed)
    A = torch.tensor(rng.standard_normal((n, 2, 2))
                     + 1j * rng.standard_normal((n, 2, 2)))
    x = torch.tensor(rng.standard_normal((n, 2))
                     + 1j * rng.standard_normal((n, 2)))
    Ar = R.antirealify(A)
    xr = torch.stack([x.real[..., 0], x.imag[..., 0],
                      x.real[..., 1], x.imag[..., 1]], dim=-1)
    yr = (Ar @ xr.unsqueeze(-1)).squeeze(-1)
    y = (A @ x.conj().unsqueeze(-1)).squeeze(-1)
    want = torch.stack([y.real[..., 0], y.imag[..., 0],
                        y.real[..., 1], y.imag[..., 1]], dim=-1)
    assert torch.allclose(yr, want, atol=1e-12)


@given(st.integers(2, 8), st.integers(1, 4), st.sampled_from([0, 1, 2, 3]))
@settings(max_examples=25, deadline=None)
def test_poly_basis_shape_and_bernstein_partition(F, Npoly, ptype):
    """Basis is [F, Npoly]; Bernstein rows (type 2) sum to one (partition
    of unity), and every type's first column spans a constant direction
    so a frequency-independent gain is representable."""
    freqs = np.linspace(140e6, 160e6, F)
    B = poly.setup_polynomials(freqs, 150e6, Npoly, ptype)
    assert B.shape == (F, Npoly)
    assert torch.isfinite(B).all()
    if ptype == 2:
        assert torch.allclose(B.sum(dim=1),
                              torch.ones(F, dtype=B.dtype), atol=1e-9)
    # constant function representable: residual of ls-fit of ones is ~0
    # (pinv: Npoly may exceed F, making the system wide/rank-deficient)
    ones = torch.ones(F, 1, dtype=torch.float64)
    Bd = B.to(torch.float64)
    sol = torch.linalg.pinv(Bd) @ ones
    assert float((Bd @ sol - ones).abs().max()) < 1e-6


@given(st.floats(0.3, 3.0), st.floats(0.3, 3.0))
@settings(max_examples=20, deadline=None)
def test_product_scale_symmetric_and_contracting(a, b):
    g = shapelet.product_scale(a, b)
    assert abs(g - shapelet.product_scale(b, a)) < 1e-12
    assert g < min(a, b)  # products always NARROW the Gaussian
    # composing with an infinite-width factor is the identity
    assert abs(shapelet.product_scale(a, 1e9) - a) < 1e-6


@given(st.integers(0, 2 ** 31 - 1), st.integers(1, 3), st.integers(1, 3))
@settings(max_examples=10, deadline=None)
def test_shapelet_product_linearity(seed, Lf, Lg):
    """The Jones series product is bilinear: (aF) x G == a (F x G)."""
    rng = np.random.default_rng(seed)
    F = torch.tensor(rng.standard_normal((Lf * Lf, 2, 2))
                     + 1j * rng.standard_normal((Lf * Lf, 2, 2)))
    G = torch.tensor(rng.standard_normal((Lg * Lg, 2, 2))
                     + 1j * rng.standard_normal((Lg * Lg, 2, 2)))
    H1, g1 = shapelet.shapelet_product_jones(2.5 * F, G, 0.9, 1.2)
    H2, g2 = shapelet.shapelet_product_jones(F, G, 0.9, 1.2)
    assert g1 == g2
    assert torch.allclose(H1, 2.5 * H2, atol=1e-10)


@given(st.integers(0, 2 ** 31 - 1))
@settings(max_examples=10, deadline=None)
def test_lm_monotone_cost_reduction(seed):
    """LM on a random solvable (noiseless) calibration problem never
    increases the cost and reaches a meaningful reduction."""
    from sagecal_amd.solvers import lm as lm_mod
    rng = np.random.default_rng(seed)
    N, T = 6, 2
    pairs = [(p, q) for p in range(N) for q in range(p + 1, N)]
    bb = torch.tensor(pairs * T)
    B = len(pairs) * T
    coh = torch.tensor(rng.standard_normal((B, 2, 2))
                       + 1j * rng.standard_normal((B, 2, 2)))
    Jt = torch.tensor(np.eye(2)[None, None] + 0.3 * (
        rng.standard_normal((1, N, 2, 2))
        + 1j * rng.standard_normal((1, N, 2, 2))))
    x = R.apply_jones(coh, Jt, bb)
    prob = lm_mod.LMProblem(x, coh, bb, N, 1, None)
    J0 = torch.eye(2, dtype=torch.complex128)[None, None].expand(
        1, N, 2, 2).clone()
    J, info = lm_mod.lm_solve(prob, J0, maxiter=25)
    assert float(info['final_cost'][0]) <= float(info['init_cost'][0])
    assert float(info['final_cost'][0]) < 0.2 * float(
        info['init_cost'][0]) + 1e-12


@given(st.integers(0, 2 ** 31 - 1), st.floats(2.5, 25.0))
@settings(max_examples=15, deadline=None)
def test_student_t_weights_bounded(seed, nu):
    """IRLS Student's-t weights live in (0, (nu+8)/nu] and weight
    outliers strictly below clean points."""
    from sagecal_amd.ops import reference as RR
    rng = np.random.default_rng(seed)
    r = torch.tensor(rng.standard_normal((50, 2, 2))
                     + 1j * rng.standard_normal((50, 2, 2))) * 0.1
    r[0] *= 100.0                        # one outlier
    w = RR.update_weights(r, nu, p=8)
    assert float(w.min()) > 0
    assert float(w.max()) <= (nu + 8.0) / nu + 1e-9
    assert float(w[0]) < float(w[1:].min())


@given(jones())
@settings(max_examples=15, deadline=None)
def test_matC_inverts_vecR(J):
    assert torch.equal(R.matC(R.vecR(J)), J)


@given(st.integers(0, 2 ** 31 - 1), st.floats(2.0, 30.0))
@settings(max_examples=10, deadline=None)
def test_robust_cost_formula(seed, nu):
    """robust_cost = sum ln(1 + ||e||^2/nu) over baselines."""
    rng = np.random.default_rng(seed)
    N, B = 4, 12
    pairs = [(p, q) for p in range(N) for q in range(p + 1, N)]
    bb = torch.tensor((pairs * 3)[:B])
    coh = torch.tensor(rng.standard_normal((B, 2, 2))
                       + 1j * rng.standard_normal((B, 2, 2)))
    J = torch.tensor(np.eye(2)[None, None]
                     + 0.1 * rng.standard_normal((1, N, 2, 2)) + 0j)
    x = torch.tensor(rng.standard_normal((B, 2, 2))
                     + 1j * rng.standard_normal((B, 2, 2)))
    c, r = R.robust_cost(x, coh, J, bb, nu)
    e2 = (r.abs() ** 2).sum(dim=(-1, -2))
    want = float(torch.log1p(e2 / nu).sum())
    assert abs(float(c) - want) < 1e-10
    V = R.apply_jones(coh, J, bb)
    assert torch.allclose(r, x - V)
