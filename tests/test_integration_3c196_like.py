"""End-to-end scenario modeled on the reference's own test case
(test/Calibration/dosage.sh + 3c196.sky.txt.cluster): a bright
multi-component target cluster with NEGATIVE id and 2 hybrid chunks,
outlier clusters with 3-term spectral indices (-F 1), uv cut (-x),
residual correction toward the target (-k -1 -J 0), LBFGS polish (-l),
solutions file write + warm restart (-q). This is the workflow a
reference user runs day-to-day, exercised through the same CLI.
"""
import numpy as np
import pytest
import torch

from sagecal_amd import sky, msdata
from sagecal_amd.ops.reference import SourcePack

SKY_F1 = """\
# name h m s d m s I Q U V si si1 si2 RM eX eY eP f0
PT1 0 0 30 45 10 0 12.5 0 0 0 -0.43 0.09 0 0 0 0 0 143e6
PT2 0 0 31 45 12 0 6.8 0 0 0 -1.00 0.73 0 0 0 0 0 143e6
PT3 0 0 29 45 8 0 8.1 0 0 0 -0.53 -0.42 0 0 0 0 0 143e6
PO1 0 6 0 44 40 0 4.1 0 0 0 -0.7 0 0 0 0 0 0 143e6
PO2 23 54 0 45 40 0 3.5 0 0 0 0.1 0 0 0 0 0 0 143e6
"""
CLUSTER = """\
# negative id: target cluster, solved but kept in the residual
-1 2 PT1 PT2 PT3
2 1 PO1
3 1 PO2
"""


def test_full_reference_workflow(tmp_path):
    from sagecal_amd.apps import sagecal as app
    skyf = str(tmp_path / 'sky.txt')
    clf = str(tmp_path / 'cl.txt')
    open(skyf, 'w').write(SKY_F1)
    open(clf, 'w').write(CLUSTER)
    clusters = sky.read_sky_cluster(skyf, clf, 0.0, np.pi / 4, 150e6,
                                    fmt=1)
    assert [c.cluster_id for c in clusters] == [-1, 2, 3]
    assert clusters[0].nchunk == 2
    pack = SourcePack(clusters)
    msf = str(tmp_path / 'obs.npz')
    msdata.make_synthetic_npz(msf, N=12, tilesz=6, Ntime=6, Nchan=2,
                              pack=pack, noise_sigma=2e-3, seed=11,
                              ra0=0.0, dec0=np.pi / 4)
    sol = str(tmp_path / 'solutions.txt')
    args = ['-d', msf, '-s', skyf, '-c', clf, '-F', '1', '-t', '6',
            '-e', '3', '-g', '10', '-l', '5', '-m', '7', '-j', '5',
            '-x', '5.0', '-k', '-1', '-p', sol, '-O', 'res']
    assert app.main(args) == 0
    z = np.load(msf)
    # residual keeps the target-cluster flux (negative id): compare to
    # the outlier-free model power
    res = np.abs(z['res']).mean()
    data = np.abs(z['data']).mean()
    assert np.isfinite(z['res']).all()
    assert 0.05 * data < res < 1.2 * data
    # solutions round-trip and warm restart converges immediately
    from sagecal_amd import solutions
    hdr, tiles = solutions.read_solutions(sol)
    assert len(tiles) == 1
    Mt = 2 + 1 + 1
    assert tiles[0].shape == (Mt, 12, 2, 2)
    args2 = ['-d', msf, '-s', skyf, '-c', clf, '-F', '1', '-t', '6',
             '-e', '1', '-g', '4', '-l', '0', '-j', '5', '-q', sol,
             '-x', '5.0', '-k', '-1', '-O', 'res2']
    assert app.main(args2) == 0
    z2 = np.load(msf)
    res2 = np.abs(z2['res2']).mean()
    assert res2 < 1.3 * res          # warm start holds the solution
