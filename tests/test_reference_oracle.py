"""Cross-validation against the REFERENCE implementation compiled from
/root/reference (SURVEY.md §6 / VERDICT r1 item 6): the reference's own
lbfgs.c and predict.c are built unmodified (tools/oracle/) and run as a
correctness oracle. This converts "self-consistent" into "matches the
reference" for the optimizer core and the coherency predict."""
import os
import subprocess
import shutil

import numpy as np
import pytest
import torch

REF = '/root/reference'
ORACLE_DIR = '/tmp/sagecal_oracle'
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope='module')
def oracle():
    if shutil.which('gcc') is None or not os.path.isdir(REF):
        pytest.skip('gcc or reference tree unavailable')
    r = subprocess.run(
        ['bash', os.path.join(REPO, 'tools', 'oracle', 'build_oracle.sh'),
         ORACLE_DIR], capture_output=True, text=True)
    if r.returncode != 0:
        pytest.skip(f'oracle build failed: {r.stderr[-500:]}')
    return ORACLE_DIR


def test_lbfgs_matches_reference_rosenbrock(oracle):
    """Reference lbfgs_fit (lbfgs.c:933, the test/Dirac/demo.c problem)
    vs sagecal_amd.solvers.lbfgs.lbfgs_fit on extended Rosenbrock."""
    out = subprocess.run([os.path.join(oracle, 'oracle_lbfgs'), '8',
                          '200'], capture_output=True, text=True,
                         check=True).stdout.split()
    ref_p = np.array([float(t) for t in out[:-1]])
    ref_cost = float(out[-1])
    assert np.allclose(ref_p, 1.0, atol=1e-6)
    assert ref_cost < 1e-12

    from sagecal_amd.solvers.lbfgs import lbfgs_fit

    def fg(p):
        x0, x1 = p[0::2], p[1::2]
        a = x1 - x0 ** 2
        b = 1.0 - x0
        f = (100.0 * a ** 2 + b ** 2).sum()
        g = torch.zeros_like(p)
        g[0::2] = -400.0 * x0 * a - 2.0 * b
        g[1::2] = 200.0 * a
        return float(f), g

    p0 = torch.tensor([-1.2, 1.0] * 4, dtype=torch.float64)
    p1, _, _ = lbfgs_fit(fg, p0, maxiter=200, m=7)
    # both implementations must find the same minimum
    assert torch.allclose(p1, torch.ones(8, dtype=torch.float64),
                          atol=1e-5)
    assert np.allclose(p1.numpy(), ref_p, atol=1e-5)


def _write_layout(path, N, bb, u, v, w, pack, freq0, fdelta, tdelta,
                  dec0):
    Nbase = u.shape[0]
    M = pack.M
    with open(path, 'w') as f:
        f.write(f"{N} {Nbase} {M} {freq0} {fdelta} {tdelta} {dec0}\n")
        for b in range(Nbase):
            f.write(f"{int(bb[b, 0])} {int(bb[b, 1])} "
                    f"{float(u[b]):.17g} {float(v[b]):.17g} "
                    f"{float(w[b]):.17g}\n")
        for ci in range(M):
            s0 = int(pack.cluster_off[ci])
            s1 = int(pack.cluster_off[ci + 1])
            f.write(f"{ci} {s1 - s0}\n")
            for s in range(s0, s1):
                f.write(f"{int(pack.stype[s])} "
                        f"{float(pack.ll[s]):.17g} "
                        f"{float(pack.mm[s]):.17g} "
                        f"{float(pack.nn1[s]):.17g} "
                        f"{float(pack.sI[s]):.17g} "
                        f"{float(pack.sQ[s]):.17g} "
                        f"{float(pack.sU[s]):.17g} "
                        f"{float(pack.sV[s]):.17g} "
                        f"{float(pack.eX[s]):.17g} "
                        f"{float(pack.eY[s]):.17g} "
                        f"{float(pack.eP[s]):.17g} "
                        f"{float(pack.cxi[s]):.17g} "
                        f"{float(pack.sxi[s]):.17g} "
                        f"{float(pack.cphi[s]):.17g} "
                        f"{float(pack.sphi[s]):.17g} "
                        f"{int(pack.use_proj[s])}\n")


def test_predict_matches_reference(oracle, tmp_path):
    """Reference precalculate_coherencies (predict.c:503) vs
    ops.reference.predict_coh on the same point+gaussian+disk+ring
    layout — value-by-value."""
    from sagecal_amd import sky
    from sagecal_amd.ops.reference import SourcePack, predict_coh

    rng = np.random.default_rng(11)
    N, Nbase = 8, 28
    freq0, fdelta, tdelta, dec0 = 150e6, 180e3, 10.0, np.pi / 4
    pairs = np.array([(p, q) for p in range(N) for q in range(p + 1, N)])
    u = rng.standard_normal(Nbase) * 300.0 / 3e8
    v = rng.standard_normal(Nbase) * 300.0 / 3e8
    w = rng.standard_normal(Nbase) * 30.0 / 3e8

    srcs = {}
    clist = []
    stypes = [0, 1, 2, 3]      # point, gaussian, disk, ring
    for ci, ty in enumerate(stypes):
        name = f's{ci}'
        s = sky.Source(
            name=name, ra=0.02 * (ci + 1), dec=dec0 + 0.01 * ci,
            sI=1.0 + 0.3 * ci, sQ=0.05, sU=0.02, sV=0.01,
            spec_idx=0.0, spec_idx1=0.0, spec_idx2=0.0, f0=freq0,
            stype=ty, eX=2e-3 if ty else 0.0, eY=1e-3 if ty else 0.0,
            eP=0.3 if ty else 0.0)
        srcs[name] = s
        clist.append((ci, 1, [name]))
    clusters = sky.build_clusters(srcs, clist, 0.0, dec0, freq0)
    pack = SourcePack(clusters)

    layout = tmp_path / 'layout.txt'
    _write_layout(layout, N, pairs, u, v, w, pack, freq0, fdelta, tdelta,
                  dec0)
    lines = subprocess.run([os.path.join(oracle, 'oracle_predict'),
                            str(layout)], capture_output=True, text=True,
                           check=True).stdout.strip().splitlines()
    vals = np.array([[float(t) for t in ln.split()] for ln in lines])
    M = pack.M
    ref = (vals[:, 0::2] + 1j * vals[:, 1::2]).reshape(Nbase, M, 2, 2)
    ref = np.transpose(ref, (1, 0, 2, 3))            # [M, B, 2, 2]

    ut = torch.tensor(u)
    vt = torch.tensor(v)
    wt = torch.tensor(w)
    # the reference applies NO time smearing in precalculate_coherencies
    # (time_smear's only call site, residual.c:434, is commented out, and
    # the GPU kernels ignore deltat) — our predict implements it as a
    # deliberate physics improvement; disable it here for the comparison
    ours = predict_coh(pack, ut, vt, wt, freq0, freq0, fdelta, 0.0,
                       dec0).numpy()
    scale = np.abs(ref).max()
    err = np.abs(ours - ref).max() / scale
    # 5e-9: torch.special bessel j0/j1 vs libm j0/j1 precision
    assert err < 5e-9, f"predict mismatch: rel err {err}"


def test_element_beam_matches_reference(oracle, tmp_path):
    """Reference eval_elementcoeffs (elementbeam.c:384, with the REAL
    LBA/HBA/ALO tables compiled in) vs beams.LofarElementCoeffs.basis
    on the ported .npz tables — the port + Laguerre-Gaussian basis are
    pinned to the reference value-by-value, including the frequency
    interpolation."""
    from sagecal_amd.beams import LofarElementCoeffs
    rng = np.random.default_rng(21)
    r = rng.uniform(0.0, np.pi / 2, 12)
    t = rng.uniform(-np.pi, np.pi, 12)
    pts = ''.join(f"{a:.17g} {b:.17g}\n" for a, b in zip(r, t))
    for kind, freq in (('lba', 55e6), ('hba', 155e6), ('alo', 2.3e7),
                       ('lba', 1e6), ('hba', 9e9)):   # incl. edge clamps
        out = subprocess.run(
            [os.path.join(oracle, 'oracle_element'), kind, str(freq)],
            input=pts, capture_output=True, text=True, check=True).stdout
        vals = np.array([[float(x) for x in ln.split()]
                         for ln in out.strip().splitlines()])
        ref_theta = vals[:, 0] + 1j * vals[:, 1]
        ref_phi = vals[:, 2] + 1j * vals[:, 3]
        ec = LofarElementCoeffs.load(kind)
        ct, cp = ec.at_freq(freq)
        B = ec.basis(r, t)
        ours_theta = B @ ct
        ours_phi = B @ cp
        scale = max(np.abs(ref_theta).max(), np.abs(ref_phi).max())
        assert np.allclose(ours_theta, ref_theta, atol=1e-12 * scale), \
            (kind, freq)
        assert np.allclose(ours_phi, ref_phi, atol=1e-12 * scale), \
            (kind, freq)


def test_coords_match_reference(oracle):
    """transforms.c oracle: GMST, radec2azel, precession vs coords.py."""
    from sagecal_amd import coords
    jd = 2456789.2345
    g_ref = float(subprocess.run(
        [os.path.join(oracle, 'oracle_misc'), 'gmst', str(jd)],
        capture_output=True, text=True, check=True).stdout)   # degrees
    g_ours = float(coords.jd_to_gmst(jd))                     # radians
    dg = abs((np.degrees(g_ours) - g_ref + 180.0) % 360.0 - 180.0)
    assert dg < 1e-3, dg      # sub-arcsecond-class GMST agreement

    ra, dec, lon, lat = 1.2, 0.8, 0.11, 0.92
    out = subprocess.run(
        [os.path.join(oracle, 'oracle_misc'), 'azel', str(ra), str(dec),
         str(lon), str(lat), str(jd)],
        capture_output=True, text=True, check=True).stdout.split()
    az_r, el_r = float(out[0]), float(out[1])
    az_o, el_o = coords.radec_to_azel_gmst(ra, dec, lon, lat, g_ours)
    assert abs((az_o - az_r + np.pi) % (2 * np.pi) - np.pi) < 1e-9
    assert abs(el_o - el_r) < 1e-9

    # precession rotation: the reference's NOVAS/Capitaine-2003 matrix
    # equals our IAU-1976 matrix transposed (NOVAS stores column-major)
    # to ~2e-7 (the model difference over ~15 yr). NOTE the reference's
    # precession() maps (ra,dec) through a COLATITUDE vector convention
    # (transforms.c: pos1=(cos ra sin dec, ...)), so end-to-end ra/dec
    # values are not directly comparable — the rotation matrix is.
    Tr = np.array([float(x) for x in subprocess.run(
        [os.path.join(oracle, 'oracle_misc'), 'pmatrix', str(jd)],
        capture_output=True, text=True, check=True).stdout.split()]
        ).reshape(3, 3)
    P = coords.precession_matrix(jd)
    assert np.abs(Tr - P.T).max() < 1e-6


def test_update_nu_matches_reference(oracle):
    """updatenu.c AECM grid search vs ops.reference.update_nu_aecm."""
    from sagecal_amd.ops.reference import update_nu_aecm
    rng = np.random.default_rng(3)
    w = torch.tensor(rng.gamma(3.0, 0.4, size=512))
    for nu_old in (2.0, 5.0, 12.0):
        sumlogw = float((torch.log(w) - w).mean())
        Nd, nulow, nuhigh, p = 100, 2.0, 30.0, 8
        nu_ref = float(subprocess.run(
            [os.path.join(oracle, 'oracle_misc'), 'nu', str(sumlogw),
             str(Nd), str(nulow), str(nuhigh), str(p), str(nu_old)],
            capture_output=True, text=True, check=True).stdout)
        nu_ours = update_nu_aecm(w, nu_old, nulow, nuhigh, Nd, p)
        # grids differ by endpoint convention: allow one cell
        assert abs(nu_ours - nu_ref) <= (nuhigh - nulow) / Nd + 1e-9, \
            (nu_old, nu_ours, nu_ref)


def test_shapelet_contrib_matches_reference(oracle, tmp_path):
    """shapelet.c shapelet_contrib (uv Gauss-Hermite envelope) vs
    shapelet.shapelet_contrib on the same modes."""
    from sagecal_amd import shapelet as shmod
    rng = np.random.default_rng(9)
    n0, beta = 3, 0.7
    eX, eY, eP = 1.3, 0.8, 0.4
    modes = rng.standard_normal(n0 * n0)
    mf = tmp_path / 'modes.txt'
    mf.write_text('\n'.join(f"{m:.17g}" for m in modes))
    uvw = rng.standard_normal((10, 3)) * 2.0
    pts = ''.join(f"{a:.17g} {b:.17g} {c:.17g}\n" for a, b, c in uvw)
    out = subprocess.run(
        [os.path.join(oracle, 'oracle_misc'), 'shapelet', str(n0),
         str(beta), str(eX), str(eY), str(eP), str(mf)],
        input=pts, capture_output=True, text=True, check=True).stdout
    vals = np.array([[float(x) for x in ln.split()]
                     for ln in out.strip().splitlines()])
    ref = vals[:, 0] + 1j * vals[:, 1]
    u = torch.tensor(uvw[:, 0])
    v = torch.tensor(uvw[:, 1])
    w = torch.tensor(uvw[:, 2])
    ours = shmod.shapelet_contrib(u, v, w, eX, eY, eP, 1.0, 0.0, 1.0,
                                  0.0, False, beta, n0, modes).numpy()
    assert np.allclose(ours, ref, atol=1e-10 * np.abs(ref).max()), \
        np.abs(ours - ref).max()


def test_consensus_poly_basis_matches_reference(oracle):
    """setup_polynomials types 0/1/2 (monomial / normalized / Bernstein,
    consensus_poly.c) vs consensus.poly.setup_polynomials."""
    from sagecal_amd.consensus import poly
    freqs = np.array([115e6, 125e6, 135e6, 152e6, 170e6])
    f0 = 140e6
    for ty in (0, 1, 2):
        pts = '\n'.join(f"{f:.17g}" for f in freqs)
        out = subprocess.run(
            [os.path.join(oracle, 'oracle_poly'), 'basis', '3',
             str(len(freqs)), str(ty), str(f0)],
            input=pts, capture_output=True, text=True, check=True).stdout
        ref = np.array([[float(x) for x in ln.split()]
                        for ln in out.strip().splitlines()])
        ours = poly.setup_polynomials(freqs, f0, 3, ty).numpy()
        assert np.allclose(ours, ref, atol=1e-12), (ty, ours - ref)


def test_bb_rho_matches_reference(oracle):
    """update_rho_bb (rho_bb_threadfn, consensus_poly.c:860-926) vs
    consensus.poly.update_rho_bb_ip on hybrid-chunk clusters — the
    alphaSD/alphaMG selection, correlation gate, accept window and the
    CONCATENATED per-cluster chunk accumulation."""
    from sagecal_amd.consensus import poly
    rng = np.random.default_rng(6)
    N, M = 4, 3
    nchunks = [1, 2, 3]
    Mt = sum(nchunks)
    L = 8 * N * Mt
    rho0 = np.array([5.0, 2.0, 7.0])
    # correlated deltas for clusters 0/2, anti-correlated for 1
    dJ = rng.standard_normal(L)
    dY = 0.8 * dJ + 0.1 * rng.standard_normal(L)
    off = 8 * N * nchunks[0]
    ln1 = 8 * N * nchunks[1]
    dY[off:off + ln1] = -dJ[off:off + ln1]
    Yh0 = rng.standard_normal(L)
    J0 = rng.standard_normal(L)
    Yh = Yh0 + dY
    J = J0 + dJ
    inp = '\n'.join(f"{rho0[ci]:.17g} {nchunks[ci]}" for ci in range(M))
    for buf in (Yh, Yh0, J, J0):
        inp += '\n' + '\n'.join(f"{x:.17g}" for x in buf)
    out = subprocess.run(
        [os.path.join(oracle, 'oracle_poly'), 'rhobb', str(N), str(M),
         '1000.0'], input=inp, capture_output=True, text=True,
        check=True).stdout.split()
    ref = np.array([float(x) for x in out])
    # ours: segment-summed inner products per cluster
    ip11 = np.zeros(M)
    ip12 = np.zeros(M)
    ip22 = np.zeros(M)
    o = 0
    for ci in range(M):
        ln = 8 * N * nchunks[ci]
        ip11[ci] = (dY[o:o + ln] ** 2).sum()
        ip12[ci] = (dY[o:o + ln] * dJ[o:o + ln]).sum()
        ip22[ci] = (dJ[o:o + ln] ** 2).sum()
        o += ln
    ours = poly.update_rho_bb_ip(
        torch.tensor(rho0), 1000.0, torch.tensor(ip11),
        torch.tensor(ip12), torch.tensor(ip22)).numpy()
    assert np.allclose(ours, ref, atol=1e-10), (ours, ref)
    # anti-correlated cluster kept its old rho
    assert ours[1] == rho0[1]


def test_find_prod_inverse_matches_reference(oracle):
    """find_prod_inverse_full (per-cluster Npoly x Npoly pseudo-inverse
    of sum_f rho_fk B_f B_f^T, consensus_poly.c:465 + prod_inv_threadfn)
    vs consensus.poly.find_prod_inverse — including a rank-deficient
    case (Npoly > Nf) where the pinv cutoff semantics matter."""
    from sagecal_amd.consensus import poly
    rng = np.random.default_rng(8)
    for Npoly, Nf, M in ((3, 5, 2), (3, 2, 2)):   # full rank + deficient
        freqs = 120e6 + 1e7 * np.arange(Nf)
        f0 = float(np.mean(freqs))
        rho_fk = rng.uniform(0.5, 4.0, size=(M, Nf))   # [M, Nf]
        inp = '\n'.join(f"{f:.17g}" for f in freqs)
        # reference layout rho[k + f*M]
        for f in range(Nf):
            for k in range(M):
                inp += '\n' + f"{rho_fk[k, f]:.17g}"
        out = subprocess.run(
            [os.path.join(oracle, 'oracle_poly'), 'bii', str(Npoly),
             str(Nf), str(M), '0', str(f0)],
            input=inp, capture_output=True, text=True, check=True).stdout
        vals = np.array([[float(x) for x in ln.split()]
                         for ln in out.strip().splitlines()])
        ref = vals.reshape(M, Npoly, Npoly)
        B = poly.setup_polynomials(freqs, f0, Npoly, 0)
        ours = poly.find_prod_inverse(
            B, torch.tensor(rho_fk)).numpy()
        assert np.allclose(ours, ref, atol=1e-8), \
            (Npoly, Nf, np.abs(ours - ref).max())


def test_residuals_multifreq_matches_reference(oracle, tmp_path):
    """calculate_residuals_multifreq (residual.c:940): per-channel
    re-predict with the SOLUTIONS applied and spectral-index flux
    scaling, against our predict+apply path — this also pins the
    reference's in-memory 8-reals-per-station parameter ordering to
    solutions.jones_to_ref_vec (the solution-file layout)."""
    from sagecal_amd import sky, solutions
    from sagecal_amd.ops.reference import SourcePack, predict_coh, \
        apply_jones

    rng = np.random.default_rng(31)
    N, tilesz, Nchan = 6, 2, 2
    pairs = np.array([(p, q) for p in range(N) for q in range(p + 1, N)])
    Nbase = len(pairs)
    rows = Nbase * tilesz
    freqs = np.array([140e6, 160e6])
    freq0 = 150e6
    fdelta, tdelta, dec0 = 10e6, 0.0, np.pi / 4

    srcs = {}
    clist = []
    for ci in range(2):
        names = []
        for si in range(2):
            nm = f's{ci}_{si}'
            srcs[nm] = sky.Source(
                name=nm, ra=0.02 * (ci + 1) + 0.005 * si,
                dec=dec0 + 0.01 * ci - 0.004 * si,
                sI=1.0 + 0.3 * ci + 0.1 * si, sQ=0.05, sU=0.02,
                sV=0.01, spec_idx=-0.7 + 0.2 * ci, f0=148e6,
                stype=1 if si else 0, eX=2e-3 if si else 0.0,
                eY=1e-3 if si else 0.0, eP=0.3 if si else 0.0)
            names.append(nm)
        clist.append((ci, 1, names))
    clusters = sky.build_clusters(srcs, clist, 0.0, dec0, freq0)
    pack = SourcePack(clusters)
    M = pack.M

    u = torch.tensor(rng.standard_normal(rows) * 300.0 / 3e8)
    v = torch.tensor(rng.standard_normal(rows) * 300.0 / 3e8)
    w = torch.tensor(rng.standard_normal(rows) * 30.0 / 3e8)
    bb = torch.tensor(np.tile(pairs, (tilesz, 1)))
    J = torch.tensor(np.eye(2)[None, None] + 0.25 * (
        rng.standard_normal((M, N, 2, 2))
        + 1j * rng.standard_normal((M, N, 2, 2))))
    x = torch.tensor(rng.standard_normal((Nchan, rows, 2, 2))
                     + 1j * rng.standard_normal((Nchan, rows, 2, 2)))

    # ---- layout file for the oracle ----
    L = [f"{N} {Nbase} {tilesz} {M} {Nchan} {fdelta} {tdelta} {dec0} "
         f"-1 0.0"]
    L += [f"{f:.17g}" for f in freqs]
    for b in range(rows):
        L.append(f"{int(bb[b, 0])} {int(bb[b, 1])} {float(u[b]):.17g} "
                 f"{float(v[b]):.17g} {float(w[b]):.17g}")
    for ci in range(M):
        s0, s1 = int(pack.cluster_off[ci]), int(pack.cluster_off[ci + 1])
        L.append(f"{ci} {s1 - s0}")
        for s in range(s0, s1):
            L.append(
                f"{int(pack.stype[s])} {float(pack.ll[s]):.17g} "
                f"{float(pack.mm[s]):.17g} {float(pack.nn1[s]):.17g} "
                f"{float(pack.sI[s]):.17g} {float(pack.sQ[s]):.17g} "
                f"{float(pack.sU[s]):.17g} {float(pack.sV[s]):.17g} "
                f"{float(pack.eX[s]):.17g} {float(pack.eY[s]):.17g} "
                f"{float(pack.eP[s]):.17g} {float(pack.cxi[s]):.17g} "
                f"{float(pack.sxi[s]):.17g} {float(pack.cphi[s]):.17g} "
                f"{float(pack.sphi[s]):.17g} {int(pack.use_proj[s])} "
                f"{float(pack.f0[s]):.17g} "
                f"{float(pack.spec_idx[s]):.17g} "
                f"{float(pack.spec_idx1[s]):.17g} "
                f"{float(pack.spec_idx2[s]):.17g} "
                f"{float(pack.sI0[s]):.17g} {float(pack.sQ0[s]):.17g} "
                f"{float(pack.sU0[s]):.17g} {float(pack.sV0[s]):.17g}")
    for ci in range(M):
        pvec = solutions.jones_to_ref_vec(J[ci]).numpy()
        L += [f"{val:.17g}" for val in pvec]
    # x: XX(re,im),XY,YX,YY per baseline, per timeslot, per channel
    for fi in range(Nchan):
        xf = torch.view_as_real(x[fi]).reshape(rows, 8).numpy()
        for b in range(rows):
            L += [f"{val:.17g}" for val in xf[b]]
    layout = tmp_path / 'res_layout.txt'
    layout.write_text('\n'.join(L) + '\n')

    out = subprocess.run([os.path.join(oracle, 'oracle_residual'),
                          str(layout)], capture_output=True, text=True,
                         check=True).stdout.split()
    vals = np.array([float(t) for t in out]).reshape(Nchan, rows, 2, 2,
                                                     2)
    ref = torch.view_as_complex(torch.tensor(vals))

    # ours: x - sum_ci J_ci C_ci(f) J_ci^H per channel (per-channel
    # flux scaling handled inside predict_coh via freq != freq0)
    fdelta_ch = fdelta / Nchan
    ours = x.clone()
    for fi in range(Nchan):
        coh = predict_coh(pack, u, v, w, float(freqs[fi]), freq0,
                          fdelta_ch, tdelta, dec0)
        for ci in range(M):
            ours[fi] -= apply_jones(coh[ci], J[ci:ci + 1], bb)
    scale = float(ref.abs().max())
    err = float((ours - ref).abs().max()) / scale
    assert err < 1e-10, f"residual mismatch: rel err {err}"


def test_residual_correction_matches_reference(oracle, tmp_path):
    """The ccid MMSE-style correction (residual.c:125-196 mat_invert +
    G1 x G2^H application, incl. the det guard) vs
    sage.correct_residuals."""
    from sagecal_amd import sky, solutions
    from sagecal_amd.ops.reference import SourcePack, predict_coh, \
        apply_jones
    from sagecal_amd.solvers import sage as sage_mod

    rng = np.random.default_rng(41)
    N, tilesz, Nchan = 5, 1, 1
    pairs = np.array([(p, q) for p in range(N) for q in range(p + 1, N)])
    Nbase = len(pairs)
    rows = Nbase * tilesz
    freqs = np.array([150e6])
    freq0, fdelta, tdelta, dec0 = 150e6, 5e6, 0.0, np.pi / 4
    rho_c = 0.3

    srcs = {'a': sky.Source(name='a', ra=0.02, dec=dec0 + 0.01, sI=1.2,
                            sQ=0.0, sU=0.0, sV=0.0, f0=freq0, stype=0)}
    clusters = sky.build_clusters(srcs, [(7, 1, ['a'])], 0.0, dec0,
                                  freq0)
    pack = SourcePack(clusters)

    u = torch.tensor(rng.standard_normal(rows) * 300.0 / 3e8)
    v = torch.tensor(rng.standard_normal(rows) * 300.0 / 3e8)
    w = torch.tensor(rng.standard_normal(rows) * 30.0 / 3e8)
    bb = torch.tensor(np.tile(pairs, (tilesz, 1)))
    J = torch.tensor(np.eye(2)[None, None] + 0.3 * (
        rng.standard_normal((1, N, 2, 2))
        + 1j * rng.standard_normal((1, N, 2, 2))))
    x = torch.tensor(rng.standard_normal((Nchan, rows, 2, 2))
                     + 1j * rng.standard_normal((Nchan, rows, 2, 2)))

    L = [f"{N} {Nbase} {tilesz} 1 {Nchan} {fdelta} {tdelta} {dec0} "
         f"7 {rho_c}"]
    L += [f"{f:.17g}" for f in freqs]
    for b in range(rows):
        L.append(f"{int(bb[b, 0])} {int(bb[b, 1])} {float(u[b]):.17g} "
                 f"{float(v[b]):.17g} {float(w[b]):.17g}")
    L.append("7 1")
    s = 0
    L.append(
        f"{int(pack.stype[s])} {float(pack.ll[s]):.17g} "
        f"{float(pack.mm[s]):.17g} {float(pack.nn1[s]):.17g} "
        f"{float(pack.sI[s]):.17g} 0 0 0 0 0 0 1 0 1 0 0 "
        f"{float(pack.f0[s]):.17g} 0 0 0 {float(pack.sI0[s]):.17g} "
        f"0 0 0")
    L += [f"{val:.17g}"
          for val in solutions.jones_to_ref_vec(J[0]).numpy()]
    xf = torch.view_as_real(x[0]).reshape(rows, 8).numpy()
    for b in range(rows):
        L += [f"{val:.17g}" for val in xf[b]]
    layout = tmp_path / 'corr_layout.txt'
    layout.write_text('\n'.join(L) + '\n')
    out = subprocess.run([os.path.join(oracle, 'oracle_residual'),
                          str(layout)], capture_output=True, text=True,
                         check=True).stdout.split()
    ref = torch.view_as_complex(torch.tensor(
        np.array([float(t) for t in out]).reshape(Nchan, rows, 2, 2,
                                                  2)))

    coh = predict_coh(pack, u, v, w, freq0, freq0, fdelta, tdelta, dec0)
    res = x[0] - apply_jones(coh[0], J, bb)
    corr = sage_mod.correct_residuals(
        res.unsqueeze(0), J[0, bb[:, 0]], J[0, bb[:, 1]], rho=rho_c)
    err = float((corr[0] - ref[0]).abs().max() / ref.abs().max())
    assert err < 1e-10, f"correction mismatch: rel err {err}"
