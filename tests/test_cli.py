"""End-to-end CLI tests: calibrate / simulate / stochastic on an NpzMS,
solution-file round trip (the reference's test/Calibration smoke tests
re-imagined as an automated suite)."""
import os

import numpy as np
import pytest
import torch

from sagecal_amd import sky, msdata, solutions
from sagecal_amd.ops.reference import SourcePack


SKY = """\
P1C1 0 0 30 45 10 0 2.5 0 0 0 -0.7 0 0 0 0 150e6
P2C1 0 2 0 44 50 0 1.5 0 0 0 0.1 0 0 0 0 150e6
P3C2 0 -1 30 45 20 0 3.0 0 0 0 -0.3 0 0 0 0 150e6
"""
CLUSTER = "1 1 P1C1 P2C1\n2 1 P3C2\n"


@pytest.fixture
def obs(tmp_path):
    skyf = tmp_path / 'sky.txt'
    skyf.write_text(SKY)
    clf = tmp_path / 'cluster.txt'
    clf.write_text(CLUSTER)
    clusters = sky.read_sky_cluster(str(skyf), str(clf), 0.0, np.pi / 4,
                                    150e6)
    pack = SourcePack(clusters)
    msf = tmp_path / 'obs.npz'
    msdata.make_synthetic_npz(str(msf), N=8, tilesz=4, Ntime=4, Nchan=2,
                              pack=pack, bandwidth=50e3, noise_sigma=1e-3,
                              seed=5, ra0=0.0, dec0=np.pi / 4)
    return tmp_path, str(skyf), str(clf), str(msf)


def test_cli_calibrate(obs):
    from sagecal_amd.apps import sagecal as app
    tmp, skyf, clf, msf = obs
    sol = str(tmp / 'sol.txt')
    rc = app.main(['-d', msf, '-s', skyf, '-c', clf, '-p', sol,
                   '-t', '4', '-e', '6', '-g', '10', '-j', '3', '-l', '0'])
    assert rc == 0
    # solutions file exists and parses
    hdr, tiles = solutions.read_solutions(sol)
    assert hdr['N'] == 8 and hdr['Mt'] == 2
    assert len(tiles) == 1
    # residual column written and smaller than data
    z = np.load(msf)
    assert 'residual' in z.files
    assert np.abs(z['residual']).mean() < 0.3 * np.abs(z['data']).mean()


def test_cli_simulate_roundtrip(obs):
    """-a 1 writes a noise-free model; calibrating against it must fit."""
    from sagecal_amd.apps import sagecal as app
    tmp, skyf, clf, msf = obs
    rc = app.main(['-d', msf, '-s', skyf, '-c', clf, '-a', '1',
                   '-t', '4', '-O', 'model'])
    assert rc == 0
    z = np.load(msf)
    assert 'model' in z.files
    assert np.abs(z['model']).mean() > 0.1


def test_cli_simulate_subtract(obs):
    from sagecal_amd.apps import sagecal as app
    tmp, skyf, clf, msf = obs
    rc = app.main(['-d', msf, '-s', skyf, '-c', clf, '-a', '3',
                   '-t', '4', '-O', 'subtracted'])
    assert rc == 0
    z = np.load(msf)
    # with identity solutions, data - model removes most signal
    assert np.abs(z['subtracted']).mean() < np.abs(z['data']).mean()


def test_cli_stochastic(obs):
    from sagecal_amd.apps import sagecal as app
    tmp, skyf, clf, msf = obs
    rc = app.main(['-d', msf, '-s', skyf, '-c', clf, '-N', '3', '-M', '2',
                   '-w', '2', '-t', '4', '-l', '8', '-O', 'res_st'])
    assert rc == 0
    z = np.load(msf)
    assert np.abs(z['res_st']).mean() < 0.5 * np.abs(z['data']).mean()


def test_cli_warm_start(obs):
    """-q warm start: second run starting from written solutions."""
    from sagecal_amd.apps import sagecal as app
    tmp, skyf, clf, msf = obs
    sol = str(tmp / 'sol.txt')
    app.main(['-d', msf, '-s', skyf, '-c', clf, '-p', sol, '-t', '4',
              '-e', '4', '-j', '3', '-l', '0'])
    rc = app.main(['-d', msf, '-s', skyf, '-c', clf, '-q', sol, '-t', '4',
                   '-e', '1', '-j', '3', '-l', '0', '-O', 'res2'])
    assert rc == 0
    z = np.load(msf)
    assert np.abs(z['res2']).mean() < 0.3 * np.abs(z['data']).mean()


def _mpi_worker(rank, world, tmpdir):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = '29541'
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    import torch.distributed as dist
    dist.init_process_group('gloo', rank=rank, world_size=world)
    try:
        from sagecal_amd.apps import sagecal_mpi as app
        rc = app.main(['-f', os.path.join(tmpdir, 'mslist.txt'),
                       '-s', os.path.join(tmpdir, 'sky.txt'),
                       '-c', os.path.join(tmpdir, 'cluster.txt'),
                       '-t', '4', '-A', '4', '-P', '2', '-j', '3',
                       '-e', '2', '-g', '8', '-r', '2.0', '-M',
                       '-X', '0.01,1e-4,1,10,2', '-u', '0.0',
                       '-p', os.path.join(tmpdir, 'sol.txt')])
        assert rc == 0
        # warm start (-q) from the solutions just written (new round-2
        # MPI path): one quick ADMM pass must run clean
        rc = app.main(['-f', os.path.join(tmpdir, 'mslist.txt'),
                       '-s', os.path.join(tmpdir, 'sky.txt'),
                       '-c', os.path.join(tmpdir, 'cluster.txt'),
                       '-t', '4', '-A', '2', '-P', '2', '-j', '3',
                       '-e', '1', '-g', '6', '-r', '2.0',
                       '-q', os.path.join(tmpdir, 'sol.txt.rank')
                       + str(rank), '-O', 'res_q'])
        assert rc == 0
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_sagecal_mpi_two_bands(tmp_path):
    """2-rank sagecal-mpi analog over gloo: both band residual files
    written, residuals reduced, per-rank J and global Z solution files
    parse back."""
    import os as _os
    import torch.multiprocessing as mp
    (tmp_path / 'sky.txt').write_text(SKY)
    (tmp_path / 'cluster.txt').write_text(CLUSTER)
    names = []
    for r, f0 in enumerate((150e6, 152e6)):
        clusters = sky.read_sky_cluster(str(tmp_path / 'sky.txt'),
                                        str(tmp_path / 'cluster.txt'),
                                        0.0, np.pi / 4, f0)
        pack = SourcePack(clusters)
        msf = tmp_path / f'band{r}.npz'
        msdata.make_synthetic_npz(str(msf), N=8, tilesz=4, Ntime=4,
                                  Nchan=2, pack=pack, freq0=f0,
                                  bandwidth=50e3, noise_sigma=1e-3,
                                  seed=7, ra0=0.0, dec0=np.pi / 4)
        names.append(str(msf))
    (tmp_path / 'mslist.txt').write_text('\n'.join(names))
    mp.spawn(_mpi_worker, args=(2, str(tmp_path)), nprocs=2, join=True)
    for r in range(2):
        z = np.load(names[r])
        assert 'residual' in z.files
        assert np.abs(z['residual']).mean() < \
            0.4 * np.abs(z['data']).mean(), f"band {r}"
        # warm-started run also reduced residuals
        assert np.abs(z['res_q']).mean() < \
            0.4 * np.abs(z['data']).mean(), f"band {r} warm"
    from sagecal_amd import solutions
    hdr, tiles = solutions.read_solutions(str(tmp_path / 'sol.txt.rank0'))
    assert tiles and tiles[0].shape[-2:] == (2, 2)
    hdrz, ztiles = solutions.read_global_z(str(tmp_path / 'sol.txt.Z'))
    assert ztiles, "global Z solution file missing/empty"
    # reference Z header (sagecal_master.cpp:517): freq Npoly N Mo Mt
    assert hdrz['Npoly'] == 2 and hdrz['N'] == hdr['N']
    assert hdrz['Mt'] >= hdrz['M']
    spf = tmp_path / 'spatial_sol.txt'
    assert spf.exists() and len(spf.read_text().splitlines()) > 2
    assert (tmp_path / 'sol.txt.spatial.ppm').exists()
    # Npoly=2, Z per effective cluster: the Z file carries Npoly times
    # the J file's values per tile
    assert sum(t.numel() for t in ztiles) == \
        2 * sum(t.numel() for t in tiles)


def test_cli_dochan_and_diag(obs):
    from sagecal_amd.apps import sagecal as app
    tmp, skyf, clf, msf = obs
    rc = app.main(['-d', msf, '-s', skyf, '-c', clf, '-t', '4', '-e', '3',
                   '-j', '3', '-l', '0', '-b', '1', '-O', 'res_chan'])
    assert rc == 0
    z = np.load(msf)
    assert np.abs(z['res_chan']).mean() < 0.4 * np.abs(z['data']).mean()
    rc = app.main(['-d', msf, '-s', skyf, '-c', clf, '-t', '4', '-e', '1',
                   '-j', '3', '-l', '0', '-i', '1', '-O', 'lev'])
    assert rc == 0
    z = np.load(msf)
    assert np.isfinite(z['lev']).all()
    assert z['lev'].real.max() > 0


def test_cli_correct_phase_only(obs):
    from sagecal_amd.apps import sagecal as app
    tmp, skyf, clf, msf = obs
    rc = app.main(['-d', msf, '-s', skyf, '-c', clf, '-t', '4', '-e', '3',
                   '-j', '3', '-l', '0', '-k', '1', '-J', '1',
                   '-O', 'res_corr'])
    assert rc == 0
    z = np.load(msf)
    assert np.isfinite(z['res_corr']).all()


def test_sagefit_with_flags():
    """Flagged rows contribute nothing; solution still converges on the
    valid part."""
    import torch
    from sagecal_amd.solvers import sage
    from sagecal_amd.constants import SM_LM_LBFGS
    from tests.test_sage import setup_ms
    ms, pack = setup_ms(M=2, noise=1e-4)
    tile = ms.load_tile(0)
    # flag 20% of rows with garbage data
    rng = np.random.default_rng(0)
    bad = torch.zeros(tile.x.shape[0], dtype=torch.bool)
    bad[rng.choice(tile.x.shape[0], tile.x.shape[0] // 5,
                   replace=False)] = True
    tile.x[bad] = 1000.0
    tile.flags = bad
    bb = ms.bb_tensor()
    state = sage.CalState(pack, ms.N)
    cohs = sage.precalc_coherencies(pack, tile)
    opts = sage.SageSolveOptions(max_emiter=3, max_iter=12,
                                 solver_mode=SM_LM_LBFGS, joint_iters=4)
    res0, res1 = sage.sagefit(state, cohs, tile, bb, opts)
    assert res1 < 0.05 * res0


def test_npz_multi_tile_and_uvwriter(tmp_path):
    """Multi-tile observation: per-tile solutions blocks; uvwriter runs."""
    from sagecal_amd.apps import sagecal as app, uvwriter
    (tmp_path / 'sky.txt').write_text(SKY)
    (tmp_path / 'cl.txt').write_text(CLUSTER)
    clusters = sky.read_sky_cluster(str(tmp_path / 'sky.txt'),
                                    str(tmp_path / 'cl.txt'),
                                    0.0, np.pi / 4, 150e6)
    pack = SourcePack(clusters)
    msf = str(tmp_path / 'obs2.npz')
    msdata.make_synthetic_npz(msf, N=8, tilesz=3, Ntime=6, Nchan=2,
                              pack=pack, bandwidth=50e3, noise_sigma=1e-3,
                              seed=8, ra0=0.0, dec0=np.pi / 4)
    sol = str(tmp_path / 'sol2.txt')
    rc = app.main(['-d', msf, '-s', str(tmp_path / 'sky.txt'),
                   '-c', str(tmp_path / 'cl.txt'), '-p', sol, '-t', '3',
                   '-e', '5', '-j', '3', '-l', '0'])
    assert rc == 0
    hdr, tiles = solutions.read_solutions(sol)
    assert len(tiles) == 2          # two solution intervals
    rc = uvwriter.main(['-d', msf])
    assert rc == 0


def test_cli_beam_and_whiten(tmp_path):
    """-B 1 (array-beam predict from MS element layouts) and -W 1
    (NCP pre-whitening) run end-to-end; calibration against beam-free
    synthetic data still reduces residuals (beam ~ unity at the phase
    centre pointing)."""
    from sagecal_amd.apps import sagecal as app
    (tmp_path / 'sky.txt').write_text(SKY)
    (tmp_path / 'cluster.txt').write_text(CLUSTER)
    clusters = sky.read_sky_cluster(str(tmp_path / 'sky.txt'),
                                    str(tmp_path / 'cluster.txt'),
                                    0.0, np.pi / 4, 150e6)
    pack = SourcePack(clusters)
    msf = str(tmp_path / 'obs.npz')
    msdata.make_synthetic_npz(msf, N=8, tilesz=4, Ntime=4, Nchan=2,
                              pack=pack, noise_sigma=1e-3, seed=3,
                              ra0=0.0, dec0=np.pi / 4)
    # embed a small element layout per station
    z = dict(np.load(msf))
    rng = np.random.default_rng(0)
    z['element_enu'] = rng.uniform(-10, 10, (8, 16, 3))
    z['element_enu'][:, :, 2] = 0.0
    z['lon'], z['lat'] = 0.0, np.pi / 4
    np.savez_compressed(msf, **z)
    rc = app.main(['-d', msf, '-s', str(tmp_path / 'sky.txt'),
                   '-c', str(tmp_path / 'cluster.txt'), '-t', '4',
                   '-e', '2', '-g', '8', '-j', '3', '-l', '0',
                   '-B', '1', '-W', '1', '-O', 'resb'])
    assert rc == 0
    out = np.load(msf)
    assert 'resb' in out.files
    assert np.isfinite(out['resb']).all()
    # per-channel full beam (-B 5 = array+element WB)
    rc = app.main(['-d', msf, '-s', str(tmp_path / 'sky.txt'),
                   '-c', str(tmp_path / 'cluster.txt'), '-t', '4',
                   '-e', '1', '-g', '6', '-j', '3', '-l', '0',
                   '-B', '5', '-O', 'resb5'])
    assert rc == 0
    out = np.load(msf)
    assert np.isfinite(out['resb5']).all()


def test_cli_precesses_with_ms_epoch(tmp_path):
    """An MS carrying jd0 precesses the catalogue + phase centre to the
    observation epoch (data.cpp:1616 behavior): calibration still
    converges, and the loaded cluster directions differ from the
    J2000 ones."""
    from sagecal_amd.apps import sagecal as app
    (tmp_path / 'sky.txt').write_text(SKY)
    (tmp_path / 'cluster.txt').write_text(CLUSTER)
    clusters = sky.read_sky_cluster(str(tmp_path / 'sky.txt'),
                                    str(tmp_path / 'cluster.txt'),
                                    0.0, np.pi / 4, 150e6)
    pack = SourcePack(clusters)
    msf = str(tmp_path / 'obs.npz')
    msdata.make_synthetic_npz(msf, N=8, tilesz=4, Ntime=4, Nchan=2,
                              pack=pack, noise_sigma=1e-3, seed=5,
                              ra0=0.0, dec0=np.pi / 4)
    z = dict(np.load(msf))
    z['jd0'] = 2462000.5          # ~2028: a few arcmin of precession
    np.savez_compressed(msf, **z)
    cl2 = sky.read_sky_cluster(str(tmp_path / 'sky.txt'),
                               str(tmp_path / 'cluster.txt'),
                               0.0, np.pi / 4, 150e6, jd=2462000.5)
    # differential precession is small: lmn shift well under 1e-4 but
    # nonzero
    d = abs(float(cl2[0].ll[0]) - float(clusters[0].ll[0]))
    assert 0 < d < 1e-4
    rc = app.main(['-d', msf, '-s', str(tmp_path / 'sky.txt'),
                   '-c', str(tmp_path / 'cluster.txt'), '-t', '4',
                   '-e', '2', '-g', '8', '-j', '3', '-l', '0',
                   '-O', 'resp'])
    assert rc == 0
    out = np.load(msf)
    assert np.abs(out['resp']).mean() < 0.4 * np.abs(out['data']).mean()


def test_cli_mslist_processes_all(tmp_path):
    """-f MSlist runs the calibration on EVERY listed MS (main.cpp MS
    loop), writing residuals into each."""
    from sagecal_amd.apps import sagecal as app
    (tmp_path / 'sky.txt').write_text(SKY)
    (tmp_path / 'cluster.txt').write_text(CLUSTER)
    clusters = sky.read_sky_cluster(str(tmp_path / 'sky.txt'),
                                    str(tmp_path / 'cluster.txt'),
                                    0.0, np.pi / 4, 150e6)
    pack = SourcePack(clusters)
    names = []
    for i in range(2):
        msf = str(tmp_path / f'm{i}.npz')
        msdata.make_synthetic_npz(msf, N=8, tilesz=4, Ntime=4, Nchan=2,
                                  pack=pack, noise_sigma=1e-3, seed=10 + i,
                                  ra0=0.0, dec0=np.pi / 4)
        names.append(msf)
    (tmp_path / 'list.txt').write_text('\n'.join(names))
    rc = app.main(['-f', str(tmp_path / 'list.txt'),
                   '-s', str(tmp_path / 'sky.txt'),
                   '-c', str(tmp_path / 'cluster.txt'), '-t', '4',
                   '-e', '2', '-g', '8', '-j', '3', '-l', '0',
                   '-O', 'resm'])
    assert rc == 0
    for msf in names:
        out = np.load(msf)
        assert 'resm' in out.files, msf
        assert np.abs(out['resm']).mean() < \
            0.4 * np.abs(out['data']).mean(), msf


def test_annotate_and_convert(tmp_path):
    """annotate produces a DS9 region per source; convert_skymodel LSM ->
    BBS -> LSM round-trips positions/fluxes/spectral index."""
    from sagecal_amd.apps import annotate, convert_skymodel
    from sagecal_amd import sky as skymod
    (tmp_path / 'sky.txt').write_text(SKY)
    (tmp_path / 'cluster.txt').write_text(CLUSTER)
    reg = str(tmp_path / 'out.reg')
    rc = annotate.main(['-s', str(tmp_path / 'sky.txt'),
                        '-c', str(tmp_path / 'cluster.txt'),
                        '-o', reg, '-n'])
    assert rc == 0
    lines = open(reg).read()
    srcs = skymod.read_sky_model(str(tmp_path / 'sky.txt'))
    assert lines.count('circle') + lines.count('ellipse') == len(srcs)
    assert 'fk5' in lines
    # LSM -> BBS -> LSM
    bbs = str(tmp_path / 'model.bbs')
    lsm2 = str(tmp_path / 'sky2.txt')
    assert convert_skymodel.main(['-l', '-i', str(tmp_path / 'sky.txt'),
                                  '-o', bbs]) == 0
    assert convert_skymodel.main(['-b', '-i', bbs, '-o', lsm2]) == 0
    s2 = skymod.read_sky_model(lsm2)
    assert len(s2) == len(srcs)
    a = sorted(srcs.values(), key=lambda s: s.sI)
    b = sorted(s2.values(), key=lambda s: s.sI)
    for x, y in zip(a, b):
        dra = (x.ra - y.ra + np.pi) % (2 * np.pi) - np.pi
        assert abs(dra) < 1e-6
        assert abs(x.dec - y.dec) < 1e-6
        assert abs(x.sI - y.sI) < 1e-5
        assert abs(x.spec_idx - y.spec_idx) < 1e-3


def test_create_clusters(tmp_path):
    from sagecal_amd.apps import create_clusters
    from sagecal_amd import sky as skymod
    (tmp_path / 'sky.txt').write_text(SKY)
    out = str(tmp_path / 'cl2.txt')
    rc = create_clusters.main(['-s', str(tmp_path / 'sky.txt'),
                               '-c', out, '-Q', '2', '-t', '2'])
    assert rc == 0
    clist = skymod.read_cluster_file(out)
    assert len(clist) == 2
    allnames = [n for _, _, ns in clist for n in ns]
    assert sorted(allnames) == ['P1C1', 'P2C1', 'P3C2']
    assert all(nc == 2 for _, nc, _ in clist)
    # the written file drives a calibration end-to-end
    clusters = skymod.read_sky_cluster(str(tmp_path / 'sky.txt'), out,
                                       0.0, np.pi / 4, 150e6)
    assert sum(c.nsrc for c in clusters) == 3


def test_change_freq(tmp_path):
    from sagecal_amd.apps import change_freq
    msf = str(tmp_path / 'a.npz')
    msdata.make_synthetic_npz(msf, N=5, tilesz=2, Ntime=2, Nchan=2)
    out = str(tmp_path / 'b.npz')
    rc = change_freq.main(['-d', msf, '-f', '160e6', '-o', out])
    assert rc == 0
    z = np.load(out)
    assert abs(float(np.mean(z['freqs'])) - 160e6) < 1.0
    ms = msdata.NpzMS(out)
    assert abs(ms.freq0 - 160e6) < 1.0


def _mpi_mux_worker(rank, world, tmpdir):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = '29558'
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    import torch.distributed as dist
    dist.init_process_group('gloo', rank=rank, world_size=world)
    try:
        from sagecal_amd.apps import sagecal_mpi as app
        rc = app.main(['-f', os.path.join(tmpdir, 'mslist4.txt'),
                       '-s', os.path.join(tmpdir, 'sky.txt'),
                       '-c', os.path.join(tmpdir, 'cluster.txt'),
                       '-t', '4', '-A', '8', '-P', '2', '-j', '3',
                       '-e', '2', '-g', '8', '-r', '2.0'])
        assert rc == 0
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_sagecal_mpi_multiplexed_four_bands(tmp_path):
    """4 MSs on 2 ranks: the CLI takes the multiplexed path (rotating
    bands per ADMM iteration) and writes residuals into every MS."""
    import torch.multiprocessing as mp
    (tmp_path / 'sky.txt').write_text(SKY)
    (tmp_path / 'cluster.txt').write_text(CLUSTER)
    names = []
    for r, f0 in enumerate((148e6, 150e6, 152e6, 154e6)):
        clusters = sky.read_sky_cluster(str(tmp_path / 'sky.txt'),
                                        str(tmp_path / 'cluster.txt'),
                                        0.0, np.pi / 4, f0)
        pack = SourcePack(clusters)
        msf = tmp_path / f'mband{r}.npz'
        msdata.make_synthetic_npz(str(msf), N=8, tilesz=4, Ntime=4,
                                  Nchan=2, pack=pack, freq0=f0,
                                  bandwidth=50e3, noise_sigma=1e-3,
                                  seed=20 + r, ra0=0.0, dec0=np.pi / 4)
        names.append(str(msf))
    (tmp_path / 'mslist4.txt').write_text('\n'.join(names))
    mp.spawn(_mpi_mux_worker, args=(2, str(tmp_path)), nprocs=2,
             join=True)
    for msf in names:
        z = np.load(msf)
        assert 'residual' in z.files, msf
        assert np.abs(z['residual']).mean() < \
            0.4 * np.abs(z['data']).mean(), msf


def test_cli_beam_with_real_lofar_tables(obs):
    """-B 2 with the REAL LOFAR element tables, selected from the MS
    metadata key `elem_type` (LOFAR_ANTENNA_FIELD analog,
    data.cpp:268-288) or --elem-type: simulate WITH the LBA beam, then
    calibrating with the SAME tables fits far better than with the
    wrong (HBA) tables — the dipole model matters and flows through."""
    from sagecal_amd.apps import sagecal as app
    tmp, skyf, clf, msf = obs
    z = dict(np.load(msf))
    rng = np.random.default_rng(2)
    z['element_enu'] = rng.uniform(-20, 20, (8, 8, 3))
    z['elem_type'] = 'lba'
    np.savez(msf, **z)
    # simulate the LBA-beamed model (auto elem_type from the MS)
    rc = app.main(['-d', msf, '-s', skyf, '-c', clf, '-a', '1',
                   '-t', '4', '-B', '2', '-O', 'beamed'])
    assert rc == 0
    z = dict(np.load(msf))
    model = z['beamed']
    assert np.abs(model).mean() > 1e-3
    # make the beamed model the DATA (plus small noise)
    z['data'] = (model + 1e-3 * (rng.standard_normal(model.shape)
                                 + 1j * rng.standard_normal(model.shape))
                 ).astype(model.dtype)
    np.savez(msf, **z)
    res = {}
    for tag, extra in (('lba', []), ('hba', ['--elem-type', 'hba'])):
        rc = app.main(['-d', msf, '-s', skyf, '-c', clf, '-t', '4',
                       '-e', '6', '-g', '10', '-j', '3', '-l', '0',
                       '-B', '2', '-O', f'r_{tag}'] + extra)
        assert rc == 0
        zz = np.load(msf)
        res[tag] = float(np.abs(zz[f'r_{tag}']).mean())
    data_mean = float(np.abs(np.load(msf)['data']).mean())
    # right tables: near-noise residual; wrong tables: markedly worse
    assert res['lba'] < 0.05 * data_mean, res
    assert res['hba'] > 3.0 * res['lba'], res


def test_sagecal_mpi_diffuse_spatial_model(tmp_path, monkeypatch):
    """-D cluster_id,gamma with -X: the designated shapelet cluster's
    coherencies are re-predicted from the FISTA spatial model at the
    admm cadence (recalculate_diffuse_coherencies hook,
    sagecal_slave.cpp:669-694) — world=1 run, residual still reduced,
    hook actually fired."""
    from sagecal_amd.apps import sagecal_mpi as app
    from sagecal_amd import shapelet as shmod
    skyf = tmp_path / 'sky.txt'
    skyf.write_text(SKY + "SDIF 0 1 0 45 5 0 2.0 0 0 0 0 0 "
                          "1e-3 1e-3 0 150e6\n")
    clf = tmp_path / 'cluster.txt'
    clf.write_text(CLUSTER + "3 1 SDIF\n")
    # shapelet modes file next to the sky file (readsky.c convention)
    shmod.write_modes_file(str(tmp_path / 'SDIF.fits.modes'),
                           0.25, 0.79, 2, 5e-4,
                           np.array([1.0, 0.2, 0.1, 0.05]))
    clusters = sky.read_sky_cluster(str(skyf), str(clf), 0.0, np.pi / 4,
                                    150e6)
    pack = SourcePack(clusters)
    assert pack.shapelets, "shapelet source did not load"
    msf = tmp_path / 'band0.npz'
    msdata.make_synthetic_npz(str(msf), N=8, tilesz=4, Ntime=4, Nchan=2,
                              pack=pack, bandwidth=50e3,
                              noise_sigma=1e-3, seed=9, ra0=0.0,
                              dec0=np.pi / 4)
    (tmp_path / 'mslist.txt').write_text(str(msf))
    fired = {'n': 0}
    from sagecal_amd.consensus.admm import ConsensusADMM
    orig = ConsensusADMM.spatial_update

    def spy(self, *a, **k):
        fired['n'] += 1
        return orig(self, *a, **k)
    monkeypatch.setattr(ConsensusADMM, 'spatial_update', spy)
    rc = app.main(['-f', str(tmp_path / 'mslist.txt'), '-s', str(skyf),
                   '-c', str(clf), '-t', '4', '-A', '4', '-P', '1',
                   '-j', '3', '-e', '2', '-g', '8', '-r', '2.0',
                   '-X', '0.01,1e-4,1,10,2', '-u', '0.0', '-D', '3,0.1',
                   '-F', '0', '-i', '1'])
    assert rc == 0
    assert fired['n'] >= 1
    z = np.load(str(msf))
    # -i 1: the output column carries influence-function leverage
    assert 'residual' in z.files
    assert np.isfinite(z['residual']).all()
    assert z['residual'].real.max() > 0


def _fed_worker(rank, world, tmpdir):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = '29561'
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    import torch.distributed as dist
    dist.init_process_group('gloo', rank=rank, world_size=world)
    try:
        from sagecal_amd.apps import sagecal_mpi as app
        rc = app.main(['-f', os.path.join(tmpdir, 'mslist.txt'),
                       '-s', os.path.join(tmpdir, 'sky.txt'),
                       '-c', os.path.join(tmpdir, 'cluster.txt'),
                       '-t', '4', '-N', '2', '-M2', '2', '-w', '2',
                       '-l', '8', '-A', '2', '-P', '2', '--fed-alpha', '0.1'])
        assert rc == 0
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_sagecal_mpi_federated_stochastic(tmp_path):
    """-N in sagecal-mpi: the federated stochastic mode (the reference's
    sagecal_stochastic_master/slave dispatch) runs 2 ranks over gloo,
    writes reduced residuals on both bands."""
    import torch.multiprocessing as mp
    (tmp_path / 'sky.txt').write_text(SKY)
    (tmp_path / 'cluster.txt').write_text(CLUSTER)
    names = []
    for r, f0 in enumerate((150e6, 152e6)):
        clusters = sky.read_sky_cluster(str(tmp_path / 'sky.txt'),
                                        str(tmp_path / 'cluster.txt'),
                                        0.0, np.pi / 4, f0)
        pack = SourcePack(clusters)
        msf = tmp_path / f'fband{r}.npz'
        msdata.make_synthetic_npz(str(msf), N=8, tilesz=4, Ntime=4,
                                  Nchan=4, pack=pack, freq0=f0,
                                  bandwidth=50e3, noise_sigma=1e-3,
                                  seed=11, ra0=0.0, dec0=np.pi / 4)
        names.append(str(msf))
    (tmp_path / 'mslist.txt').write_text('\n'.join(names))
    mp.spawn(_fed_worker, args=(2, str(tmp_path)), nprocs=2, join=True)
    for r in range(2):
        z = np.load(names[r])
        assert 'residual' in z.files
        assert np.abs(z['residual']).mean() < \
            0.6 * np.abs(z['data']).mean(), f"band {r}"
