"""Shapelet model tests: basis math, modes-file I/O, predict integration."""
import math
import os

import numpy as np
import pytest
import torch

from sagecal_amd import shapelet, sky
from sagecal_amd.ops import reference as R
from sagecal_amd.ops.reference import SourcePack


def test_mode00_is_gaussian():
    """Mode (0,0) gives the closed-form 2*pi*0.5*exp(-|uv|^2 b^2/2)."""
    u = torch.tensor([0.0, 100.0, -50.0], dtype=torch.float64)
    v = torch.tensor([0.0, 30.0, 80.0], dtype=torch.float64)
    w = torch.zeros(3, dtype=torch.float64)
    beta, n0 = 0.01, 3
    modes = [0.0] * 9
    modes[0] = 1.0
    out = shapelet.shapelet_contrib(u, v, w, 1.0, 1.0, 0.0, 1, 0, 1, 0,
                                    False, beta, n0, modes)
    expect = 2 * math.pi * 0.5 * torch.exp(-0.5 * (u**2 + v**2) * beta**2)
    torch.testing.assert_close(out.real, expect)
    assert float(out.imag.abs().max()) == 0.0


def test_uv_matches_image_dft():
    """uv-domain shapelet equals the 2-D DFT of its image-plane rendering
    (Fourier duality of the Gauss-Hermite basis; image_basis is the dual
    of uv_mode_vectors)."""
    rng = np.random.default_rng(0)
    n0, beta = 4, 2e-3
    modes = rng.standard_normal(n0 * n0)
    L = 256
    ext = 8 * beta / (2 * np.pi) * 6
    l = np.linspace(-ext, ext, L)
    dl = l[1] - l[0]
    ll, mm = np.meshgrid(l, l, indexing='ij')
    img = (shapelet.image_basis(ll.ravel(), mm.ravel(), n0, beta).numpy()
           @ modes)
    u = np.array([40.0, -25.0, 60.0])
    v = np.array([10.0, 45.0, -30.0])
    dft = []
    for k in range(3):
        ph = np.exp(2j * np.pi * (-u[k] * ll.ravel() + v[k] * mm.ravel()))
        dft.append((img * ph).sum() * dl * dl)
    dft = np.array(dft)
    got = shapelet.shapelet_contrib(
        torch.tensor(u), torch.tensor(v), torch.zeros(3), 1.0, 1.0, 0.0,
        1, 0, 1, 0, False, beta, n0, modes).numpy()
    np.testing.assert_allclose(got, dft, rtol=1e-6, atol=1e-9)


def test_modes_file_roundtrip(tmp_path):
    p = str(tmp_path / 'S1.fits.modes')
    modes = np.arange(9.0)
    shapelet.write_modes_file(p, 0, 0, 3, 0.01, modes)
    n0, beta, m2 = shapelet.read_modes_file(p)
    assert n0 == 3 and beta == pytest.approx(0.01)
    np.testing.assert_allclose(m2, modes)


def test_predict_with_shapelet_source(tmp_path):
    """Sky model with an S source + modes file flows through predict."""
    skyf = tmp_path / 'sky.txt'
    skyf.write_text(
        "S1 0 1 0 45 10 0 5.0 0 0 0 0 0 1 1 0 150e6\n"
        "P1 0 -1 0 44 50 0 2.0 0 0 0 0 0 0 0 0 150e6\n")
    (tmp_path / 'cluster.txt').write_text("0 1 S1 P1\n")
    shapelet.write_modes_file(str(tmp_path / 'S1.fits.modes'), 0, 0,
                              3, 5e-4, [1.0, 0.2, 0, 0.1, 0, 0, 0, 0, 0])
    clusters = sky.read_sky_cluster(str(skyf), str(tmp_path / 'cluster.txt'),
                                    0.0, np.pi / 4, 150e6)
    assert clusters[0].shapelets, "modes file not attached"
    pack = SourcePack(clusters)
    assert pack.shapelets
    rng = np.random.default_rng(1)
    B = 12
    u = torch.tensor(rng.uniform(-2e-6, 2e-6, B))
    v = torch.tensor(rng.uniform(-2e-6, 2e-6, B))
    w = torch.zeros(B, dtype=torch.float64)
    coh = R.predict_coh(pack, u, v, w, 150e6, 150e6, 0.0, 0.0, np.pi / 4)
    assert torch.isfinite(torch.view_as_real(coh)).all()
    # shapelet changes the prediction vs treating S1 as a point
    pack2 = SourcePack(clusters)
    pack2.shapelets = {}
    coh2 = R.predict_coh(pack2, u, v, w, 150e6, 150e6, 0.0, 0.0, np.pi / 4)
    assert float((coh - coh2).abs().max()) > 1e-3


def test_jones_product_exact_in_image_plane():
    """Zp(l,m) C(l,m) Zq(l,m)^H as a mode series: the Jones shapelet
    product (shapelet_product_jones) must reproduce the pointwise product
    of the factor series on a grid to machine precision."""
    rng = np.random.default_rng(3)
    Gc, Gz = 3, 2
    bc, bz = 1.1, 0.8

    def render(modes, beta, x, y):
        n0 = int(np.sqrt(modes.shape[0]))
        px = shapelet.hermite_phi(torch.tensor(x / beta), n0)
        py = shapelet.hermite_phi(torch.tensor(y / beta), n0)
        m = torch.tensor(modes).reshape(n0, n0, 2, 2)
        return torch.einsum('bk,bl,klij->bij',
                            py.to(torch.complex128),
                            px.to(torch.complex128), m)

    C = rng.standard_normal((Gc * Gc, 2, 2)) \
        + 1j * rng.standard_normal((Gc * Gc, 2, 2))
    Zp = rng.standard_normal((Gz * Gz, 2, 2)) \
        + 1j * rng.standard_normal((Gz * Gz, 2, 2))
    Zq = rng.standard_normal((Gz * Gz, 2, 2)) \
        + 1j * rng.standard_normal((Gz * Gz, 2, 2))
    CZq, g1 = shapelet.shapelet_product_jones(
        torch.tensor(C), torch.tensor(Zq), bc, bz, conj_g=True)
    H, g2 = shapelet.shapelet_product_jones(torch.tensor(Zp), CZq, bz, g1)
    x = np.linspace(-2.0, 2.0, 9)
    y = np.linspace(-2.0, 2.0, 9)
    xx, yy = np.meshgrid(x, y, indexing='ij')
    A = render(np.asarray(Zp), bz, xx.ravel(), yy.ravel())
    B = render(C, bc, xx.ravel(), yy.ravel())
    Dq = render(np.asarray(Zq), bz, xx.ravel(), yy.ravel())
    want = A @ B @ Dq.conj().transpose(-1, -2)
    got = render(H.numpy(), g2, xx.ravel(), yy.ravel())
    err = float((want - got).abs().max() / want.abs().max())
    assert err < 1e-10, err


def test_diffuse_coherencies_identity_gain():
    """With Z = identity for every station, the diffuse-predict path must
    equal the plain shapelet coherency of the same sky series."""
    rng = np.random.default_rng(5)
    N, n0, beta = 4, 3, 2e-3
    modes = rng.standard_normal(n0 * n0)
    # sky coherency series: unpolarized C = modes x I/2 per mode
    Cm = torch.zeros(n0 * n0, 2, 2, dtype=torch.complex128)
    Cm[:, 0, 0] = torch.tensor(modes, dtype=torch.complex128) * 0.5
    Cm[:, 1, 1] = torch.tensor(modes, dtype=torch.complex128) * 0.5
    # identity spatial model: only the (0,0) mode, normalized so the
    # rendered Z(l,m) ~ phi_0(x)phi_0(y) ... use a wide scale and scale
    # the mode so Z ~ I over the source extent
    Gz, bz = 1, 50.0 * beta
    phi0 = float(shapelet.hermite_phi(torch.zeros(1), 1)[0, 0])
    Z = torch.zeros(N, 1, 2, 2, dtype=torch.complex128)
    Z[:, 0, 0, 0] = 1.0 / phi0 ** 2
    Z[:, 0, 1, 1] = 1.0 / phi0 ** 2
    pairs = [(p, q) for p in range(N) for q in range(p + 1, N)]
    bb = torch.tensor(pairs)
    B = len(pairs)
    freq = 150e6
    u = torch.tensor(rng.standard_normal(B) * 50 / freq)
    v = torch.tensor(rng.standard_normal(B) * 50 / freq)
    w = torch.zeros(B)
    ll, mm, nn1 = 1e-3, -2e-3, 0.0
    got = shapelet.recalculate_diffuse_coherencies(
        u, v, w, bb, Z, bz, Cm, beta, ll, mm, nn1, freq, 0.0)
    env = shapelet.shapelet_contrib(
        u * freq, v * freq, w * freq, 1.0, 1.0, 0.0, 1, 0, 1, 0, False,
        beta, n0, modes)
    G = 2.0 * np.pi * (u * ll + v * mm + w * nn1)
    ph = torch.complex(torch.cos(G * freq), torch.sin(G * freq))
    want = (env * ph)[:, None, None] * 0.5 \
        * torch.eye(2, dtype=torch.complex128)
    err = float((got - want).abs().max() / want.abs().max())
    assert err < 5e-3, err


def test_decompose_image_roundtrip():
    """decompose_image recovers modes whose image rendering matches the
    input fluxes (buildsky shapelet fitting role)."""
    rng = np.random.default_rng(9)
    n0, beta = 3, 1.5e-3
    true = rng.standard_normal(n0 * n0)
    ext = 6 * beta / (2 * np.pi) * 4
    l = np.linspace(-ext, ext, 24)
    ll, mm = np.meshgrid(l, l, indexing='ij')
    img = shapelet.image_basis(ll.ravel(), mm.ravel(), n0, beta).numpy() \
        @ true
    got = shapelet.decompose_image(ll.ravel(), mm.ravel(), img, n0, beta)
    np.testing.assert_allclose(got, true, atol=1e-8)
