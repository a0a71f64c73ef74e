"""Whole-toolchain walkthrough (the reference's selfcal tutorial,
Docs/source/tutorial.rst): synthetic image -> buildsky -> create_clusters
-> simulate visibilities from the extracted model (-a 1) -> calibrate
them -> residuals near zero -> restore the model into an image."""
import numpy as np
import pytest

from sagecal_amd.utils import fits as fitsio
from sagecal_amd import msdata


def test_selfcal_toolchain(tmp_path):
    from sagecal_amd.apps import buildsky, create_clusters, sagecal, \
        restore
    # 1. synthetic sky image with three point-ish sources
    img = np.zeros((64, 64))
    for (y, x), f in zip([(16, 20), (40, 44), (50, 12)],
                         (10.0, 6.0, 3.0)):
        img[y, x] = f
        img[y + 1, x] = 0.3 * f
        img[y, x + 1] = 0.3 * f
    img += np.random.default_rng(1).standard_normal((64, 64)) * 0.01
    fits_in = str(tmp_path / 'field.fits')
    fitsio.write_fits_image(fits_in, img, crval=(0.0, 45.0),
                            cdelt=(-0.01, 0.01))
    # 2. buildsky -> sky model; create_clusters -> cluster file
    skyf = str(tmp_path / 'sky.txt')
    clf0 = str(tmp_path / 'cl0.txt')
    assert buildsky.main(['-f', fits_in, '-s', skyf, '-c', clf0,
                          '-Q', '2']) == 0
    clf = str(tmp_path / 'cl.txt')
    assert create_clusters.main(['-s', skyf, '-c', clf, '-Q', '2']) == 0
    # 3. an empty synthetic MS at the same phase centre, then SIMULATE
    # visibilities from the extracted model (-a 1)
    msf = str(tmp_path / 'obs.npz')
    msdata.make_synthetic_npz(msf, N=10, tilesz=4, Ntime=4, Nchan=2,
                              pack=None, ra0=0.0,
                              dec0=np.deg2rad(45.0))
    assert sagecal.main(['-d', msf, '-s', skyf, '-c', clf, '-a', '1',
                         '-t', '4', '-O', 'data']) == 0
    z = np.load(msf)
    assert np.abs(z['data']).mean() > 0
    # 4. calibrate the simulated data against the same model: residuals
    # collapse (unit gains are the truth)
    assert sagecal.main(['-d', msf, '-s', skyf, '-c', clf, '-t', '4',
                         '-e', '2', '-g', '8', '-j', '1', '-l', '0',
                         '-O', 'res']) == 0
    z = np.load(msf)
    assert np.abs(z['res']).mean() < 0.05 * np.abs(z['data']).mean()
    # 5. restore the model into an image: flux reappears at the sources
    fits_out = str(tmp_path / 'model.fits')
    assert restore.main(['-f', fits_in, '-s', skyf, '-c', clf,
                         '-o', fits_out]) == 0
    rimg, _ = fitsio.read_fits_image(fits_out)
    assert rimg.max() > 1.0
    assert rimg[14:19, 18:23].sum() > rimg[30:35, 30:35].sum()
