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
