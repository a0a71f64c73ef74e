"""buildsky / restore / FITS round-trip tests."""
import numpy as np
import pytest

from sagecal_amd.utils import fits as fitsio


def test_fits_roundtrip(tmp_path):
    rng = np.random.default_rng(0)
    img = rng.standard_normal((32, 48))
    p = str(tmp_path / 'x.fits')
    fitsio.write_fits_image(p, img, crval=(10.0, 45.0),
                            cdelt=(-0.01, 0.01))
    img2, hdr = fitsio.read_fits_image(p)
    np.testing.assert_allclose(img2, img)
    assert hdr['NAXIS1'] == 48 and hdr['NAXIS2'] == 32
    assert hdr['CRVAL2'] == pytest.approx(45.0)
    # pixel <-> sky round trip
    ra, dec = fitsio.pix_to_radec(hdr, 30.0, 10.0)
    x, y = fitsio.radec_to_pix(hdr, ra, dec)
    assert x == pytest.approx(30.0, abs=1e-6)
    assert y == pytest.approx(10.0, abs=1e-6)


def _make_test_image(tmp_path, fluxes=(10.0, 6.0, 3.0)):
    img = np.zeros((64, 64))
    pos = [(16, 20), (40, 44), (50, 12)]
    for (y, x), f in zip(pos, fluxes):
        img[y, x] = f
        img[y + 1, x] = f * 0.3
        img[y, x + 1] = f * 0.3
    img += np.random.default_rng(1).standard_normal((64, 64)) * 0.01
    p = str(tmp_path / 'img.fits')
    fitsio.write_fits_image(p, img, crval=(0.0, 45.0),
                            cdelt=(-0.01, 0.01))
    return p, pos


def test_buildsky_finds_sources(tmp_path):
    from sagecal_amd.apps import buildsky
    p, pos = _make_test_image(tmp_path)
    outsky = str(tmp_path / 'sky.txt')
    outcl = str(tmp_path / 'cl.txt')
    rc = buildsky.main(['-f', p, '-s', outsky, '-c', outcl, '-Q', '2'])
    assert rc == 0
    from sagecal_amd import sky
    srcs = sky.read_sky_model(outsky)
    assert len(srcs) == 3
    fluxes = sorted(s.sI for s in srcs.values())
    assert fluxes[-1] == pytest.approx(10 * 1.6, rel=0.1)
    clist = sky.read_cluster_file(outcl)
    assert 1 <= len(clist) <= 2


def test_restore_roundtrip(tmp_path):
    """buildsky -> restore: rendered image peaks where sources were."""
    from sagecal_amd.apps import buildsky, restore
    p, pos = _make_test_image(tmp_path)
    outsky = str(tmp_path / 'sky.txt')
    outcl = str(tmp_path / 'cl.txt')
    buildsky.main(['-f', p, '-s', outsky, '-c', outcl, '-Q', '2'])
    outf = str(tmp_path / 'model.fits')
    rc = restore.main(['-f', p, '-s', outsky, '-c', outcl, '-o', outf])
    assert rc == 0
    img, hdr = fitsio.read_fits_image(outf)
    # brightest rendered pixel near the brightest input source
    y, x = np.unravel_index(np.argmax(img), img.shape)
    assert abs(y - pos[0][0]) <= 1 and abs(x - pos[0][1]) <= 1
    assert img.sum() == pytest.approx(19.0 * 1.6, rel=0.15)


def _gauss_img(shape, comps, noise=0.002, seed=2):
    """comps: list of (flux, cy, cx, sy, sx)."""
    yy, xx = np.mgrid[:shape[0], :shape[1]]
    img = np.zeros(shape)
    for f, cy, cx, sy, sx in comps:
        A = f / (2 * np.pi * sy * sx)
        img += A * np.exp(-0.5 * (((yy - cy) / sy) ** 2
                                  + ((xx - cx) / sx) ** 2))
    img += np.random.default_rng(seed).standard_normal(shape) * noise
    return img


def test_multicomponent_island_fit():
    """A blended double inside ONE island must come out as TWO components
    with the right positions/fluxes (fitmultipixels.c multi-fit + model
    selection)."""
    from sagecal_amd.apps.buildsky import find_islands, fit_island_multi
    img = _gauss_img((48, 48), [(12.0, 22.0, 19.0, 2.0, 2.0),
                                (8.0, 25.0, 25.0, 2.0, 2.0)])
    islands = find_islands(img, threshold=0.05)
    assert len(islands) == 1
    comps = fit_island_multi(img, *islands[0], maxfits=5)
    assert len(comps) == 2
    comps.sort(key=lambda c: -c['flux'])
    assert abs(comps[0]['cy'] - 22.0) < 0.3
    assert abs(comps[0]['cx'] - 19.0) < 0.3
    assert abs(comps[1]['cy'] - 25.0) < 0.3
    assert abs(comps[1]['cx'] - 25.0) < 0.3
    assert abs(comps[0]['flux'] - 12.0) / 12.0 < 0.15
    assert abs(comps[1]['flux'] - 8.0) / 8.0 < 0.15


def test_model_order_selection_prefers_single():
    """A single Gaussian island must NOT be over-fit with extra
    components (AIC stops at k=1)."""
    from sagecal_amd.apps.buildsky import find_islands, fit_island_multi
    img = _gauss_img((40, 40), [(10.0, 20.0, 20.0, 2.5, 1.8)])
    islands = find_islands(img, threshold=0.05)
    comps = fit_island_multi(img, *islands[0], maxfits=5)
    assert len(comps) == 1
    assert abs(comps[0]['flux'] - 10.0) / 10.0 < 0.1


def test_merge_close_components():
    from sagecal_amd.apps.buildsky import merge_close
    srcs = [dict(stype='P', ra=0.0, dec=0.5, flux=4.0, eX=0, eY=0, eP=0),
            dict(stype='P', ra=1e-5, dec=0.5, flux=2.0, eX=0, eY=0, eP=0),
            dict(stype='P', ra=0.01, dec=0.5, flux=1.0, eX=0, eY=0, eP=0)]
    out = merge_close(srcs, rd=1.0, beam_rad=1e-4)
    assert len(out) == 2
    big = max(out, key=lambda s: s['flux'])
    assert big['flux'] == pytest.approx(6.0)
    # flux-weighted position between the two merged components
    assert 0.0 < big['ra'] < 1e-5


def test_negative_and_rescale_cli(tmp_path):
    from sagecal_amd.apps import buildsky
    img = -_gauss_img((40, 40), [(10.0, 20.0, 20.0, 2.0, 2.0)],
                      noise=0.001)
    p = str(tmp_path / 'neg.fits')
    fitsio.write_fits_image(p, img, crval=(0.0, 45.0),
                            cdelt=(-0.01, 0.01))
    outsky = str(tmp_path / 'sky.txt')
    outcl = str(tmp_path / 'cl.txt')
    rc = buildsky.main(['-f', p, '-s', outsky, '-c', outcl, '-Q', '1',
                        '-N', '-q', '1'])
    assert rc == 0
    rows = [l for l in open(outsky) if not l.startswith('#')]
    assert len(rows) >= 1
    flux = float(rows[0].split()[7])
    assert flux < 0          # -N reports negative flux
    assert abs(abs(flux) - 10.0) / 10.0 < 0.2


def test_multifrequency_spectral_index(tmp_path):
    """-d directory of per-frequency planes: fluxes fit per plane with
    common positions/shapes, spectral index recovered from the log-log
    fit (buildmultisky.c)."""
    from sagecal_amd.apps import buildsky
    d = tmp_path / 'planes'
    d.mkdir()
    f0s = [130e6, 150e6, 170e6]
    si_true = -0.7
    fref = np.exp(np.mean(np.log(f0s)))
    for i, f0 in enumerate(f0s):
        flux = 10.0 * (f0 / fref) ** si_true
        img = _gauss_img((40, 40), [(flux, 20.0, 20.0, 2.0, 2.0)],
                         noise=0.001, seed=i)
        p = str(d / f'plane{i}.fits')
        fitsio.write_fits_image(p, img, crval=(0.0, 45.0),
                                cdelt=(-0.01, 0.01), freq=f0)
    outsky = str(tmp_path / 'sky.txt')
    rc = buildsky.main(['-d', str(d), '-s', outsky,
                        '-c', str(tmp_path / 'cl.txt'), '-Q', '1'])
    assert rc == 0
    rows = [l.split() for l in open(outsky) if not l.startswith('#')]
    assert len(rows) == 1
    flux = float(rows[0][7])
    si = float(rows[0][11])
    assert abs(flux - 10.0) / 10.0 < 0.1
    assert abs(si - si_true) < 0.1, si


def test_restore_with_solutions_scales_flux(tmp_path):
    """restore -p: cluster fluxes scaled by the mean |J|^2 of that
    cluster's solutions (restore.c withsol path)."""
    from sagecal_amd.apps import buildsky, restore
    from sagecal_amd import solutions, sky as skymod
    from sagecal_amd.solvers import sage
    from sagecal_amd.ops.reference import SourcePack
    import torch
    img = _gauss_img((48, 48), [(8.0, 24.0, 24.0, 2.0, 2.0)],
                     noise=0.001)
    p = str(tmp_path / 'f.fits')
    fitsio.write_fits_image(p, img, crval=(0.0, 45.0),
                            cdelt=(-0.01, 0.01))
    skyf = str(tmp_path / 'sky.txt')
    clf = str(tmp_path / 'cl.txt')
    assert buildsky.main(['-f', p, '-s', skyf, '-c', clf, '-Q', '1']) == 0
    clusters = skymod.read_sky_cluster(skyf, clf, 0.0, np.deg2rad(45.0),
                                       150e6)
    pack = SourcePack(clusters)
    state = sage.CalState(pack, 6)
    state.J *= 2.0                         # |J|^2 = 4 -> flux x4
    sol = str(tmp_path / 'sol.txt')
    w = solutions.SolutionWriter(sol, 150e6, 1e5, 1.0, 6, state.M,
                                 state.Mt)
    w.write_tile(state)
    w.close()
    out1 = str(tmp_path / 'o1.fits')
    out2 = str(tmp_path / 'o2.fits')
    assert restore.main(['-f', p, '-s', skyf, '-c', clf,
                         '-o', out1]) == 0
    assert restore.main(['-f', p, '-s', skyf, '-c', clf, '-p', sol,
                         '-o', out2]) == 0
    r1, _ = fitsio.read_fits_image(out1)
    r2, _ = fitsio.read_fits_image(out2)
    ratio = r2.sum() / r1.sum()
    assert abs(ratio - 4.0) < 0.2, ratio


def test_duchamp_mask_islands_by_value():
    """A Duchamp mask groups pixels by object ID (mask value), not by
    connectivity: one object with two disconnected parts stays ONE
    island; two touching objects stay TWO islands (buildsky.c mask
    semantics)."""
    from sagecal_amd.apps.buildsky import find_islands
    img = np.zeros((20, 20))
    mask = np.zeros((20, 20))
    # object 1: two disconnected patches
    mask[2:4, 2:4] = 1
    mask[10:12, 10:12] = 1
    # objects 2 and 3: touching blocks with different ids
    mask[15:17, 2:5] = 2
    mask[15:17, 5:8] = 3
    isl = find_islands(img, mask=mask)
    assert len(isl) == 3
    sizes = sorted(len(ys) for ys, xs in isl)
    assert sizes == [6, 6, 8]


def test_hull_penalty_helpers():
    """Convex hull + outside-distance (hull.c / inside_hull penalty,
    fitpixels.c:533)."""
    from sagecal_amd.apps.buildsky import convex_hull, \
        outside_hull_distance
    ys, xs = np.mgrid[0:5, 0:5]
    hull = convex_hull(ys.ravel(), xs.ravel())
    assert hull is not None
    # interior and vertex points: zero penalty
    assert outside_hull_distance(hull, 2.0, 2.0) == 0.0
    assert outside_hull_distance(hull, 0.0, 0.0) <= 1e-12
    # a point 3 pixels beyond an edge
    assert abs(outside_hull_distance(hull, 7.0, 2.0) - 3.0) < 1e-9
    # degenerate (collinear) islands give None -> no penalty
    assert convex_hull(np.zeros(4), np.arange(4.0)) is None
    assert outside_hull_distance(None, 9.0, 9.0) == 0.0


def test_restore_fft_shapelet_matches_direct(tmp_path):
    """FFT-convolved shapelet rendering (fft.c path): with a narrow PSF
    the FFT path converges to the direct basis evaluation; with a wide
    PSF the peak smooths down but flux spreads, peak position intact."""
    from sagecal_amd.apps.restore import render_shapelet_fft
    from sagecal_amd import shapelet as shmod
    ny = nx = 40
    pscale = 2e-5
    yy, xx = np.mgrid[0:ny, 0:nx]
    lg = (xx - nx / 2) * pscale
    mg = (yy - ny / 2) * pscale
    n0, beta = 3, 8e-5
    modes = np.zeros(9)
    modes[0] = 1.0
    modes[4] = 0.3
    direct = shmod.image_basis(lg.ravel(), mg.ravel(), n0,
                               beta).numpy() @ modes
    direct = direct.reshape(ny, nx)
    narrow = render_shapelet_fft(lg, mg, n0, beta, modes, 1.0, 1e-3,
                                 1e-3)
    assert np.allclose(narrow, direct, atol=1e-3 * np.abs(direct).max())
    # wide PSF: peak position intact; shape is smoother (the unit-peak
    # beam RAISES extended-source peaks — Jy/beam convention)
    wide = render_shapelet_fft(lg, mg, n0, beta, modes, 1.0, 6.0, 6.0)
    assert np.unravel_index(np.argmax(wide), wide.shape) == \
        np.unravel_index(np.argmax(direct), direct.shape)
    def width(a):
        a = np.abs(a)
        yy2, xx2 = np.mgrid[0:a.shape[0], 0:a.shape[1]]
        cy = (a * yy2).sum() / a.sum()
        cx = (a * xx2).sum() / a.sum()
        return float(np.sqrt((a * ((yy2 - cy) ** 2
                                   + (xx2 - cx) ** 2)).sum() / a.sum()))
    assert width(wide) > width(direct) + 0.3   # convolution broadens
