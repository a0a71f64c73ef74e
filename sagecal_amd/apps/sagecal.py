"""sagecal CLI — single-node calibration app.

Mirrors the reference CLI (/root/reference/src/MS/main.cpp:110-308: same
single-letter flags and the three run modes of fullbatch_mode.cpp /
minibatch_mode.cpp / minibatch_consensus_mode.cpp). MS input is the NpzMS
container (this environment has no casacore; the backend interface in
msdata.py keeps a python-casacore MS backend pluggable).

Usage:
  sagecal.py -d obs.npz -s sky.txt -c cluster.txt [-p sol.txt] [-t 10] ...
  sagecal.py -d obs.npz -s sky.txt -c cluster.txt -N 2 -M 2 -w 4   (stochastic)
  sagecal.py -d obs.npz -s sky.txt -c cluster.txt -a 1 [-p sol.txt] (simulate)
"""
import argparse
import sys
import time

import numpy as np
import torch


def build_argparser():
    ap = argparse.ArgumentParser(
        prog='sagecal', description=__doc__,
        formatter_class=argparse.RawDescriptionHelpFormatter)
    ap.add_argument('-d', dest='ms', help='MS (npz container)')
    ap.add_argument('-f', dest='mslist', help='text file with MS names')
    ap.add_argument('-s', dest='sky', required=True, help='sky model file')
    ap.add_argument('-c', dest='cluster', required=True,
                    help='cluster file')
    ap.add_argument('-p', dest='solfile',
                    help='solutions file (write; read when simulating)')
    ap.add_argument('-F', dest='format', type=int, default=0,
                    help='sky format: 0 LSM, 1 LSM 3-order spectra')
    ap.add_argument('-I', dest='incol', default='data')
    ap.add_argument('-O', dest='outcol', default='residual')
    ap.add_argument('-e', dest='max_emiter', type=int, default=3)
    ap.add_argument('-g', dest='max_iter', type=int, default=10)
    ap.add_argument('-l', dest='max_lbfgs', type=int, default=10)
    ap.add_argument('-m', dest='lbfgs_m', type=int, default=7)
    ap.add_argument('-n', dest='nthreads', type=int, default=6)
    ap.add_argument('-t', dest='tilesz', type=int, default=10)
    ap.add_argument('-a', dest='dosim', type=int, default=0,
                    help='0 calibrate; 1 simulate; 2 sim+add; 3 sim+sub')
    ap.add_argument('-z', dest='ignfile', help='ignore-cluster file')
    ap.add_argument('-b', dest='dochan', type=int, default=0)
    ap.add_argument('-B', dest='dobeam', type=int, default=0,
                    help='beam in predict: 1 array, 2 array+element, '
                         '3 element; 4/5/6 same but per channel '
                         '(MS must carry element_enu; predict_withbeam.c '
                         '/ DOBEAM_* + _WB modes)')
    ap.add_argument('--elem-type', dest='elem_type', default='auto',
                    choices=['auto', 'synthetic', 'lba', 'hba', 'alo'],
                    help='element dipole model for -B 2/3/5/6: the real '
                         'LOFAR LBA/HBA or ALO coefficient tables '
                         '(elementcoeff.h port) or the synthetic '
                         'pattern; auto reads the MS key `elem_type`, '
                         'falling back to synthetic (the reference '
                         'reads this from LOFAR_ANTENNA_FIELD, '
                         'data.cpp:268-288)')
    ap.add_argument('-W', dest='whiten', type=int, default=0,
                    help='1: pre-whiten data with the NCP uv taper '
                         '(whiten_data, updatenu.c)')
    ap.add_argument('-E', dest='gpupredict', type=int, default=None,
                    help='1: use GPU (default: auto)')
    ap.add_argument('-x', dest='min_uvcut', type=float, default=0.0)
    ap.add_argument('-y', dest='max_uvcut', type=float, default=1e9)
    ap.add_argument('-k', dest='ccid', type=int, default=-99999,
                    help='correct residuals with this cluster id')
    ap.add_argument('-o', dest='rho_corr', type=float, default=1e-9)
    ap.add_argument('-J', dest='phase_only', type=int, default=0,
                    help='1: phase-only correction')
    ap.add_argument('-i', dest='dodiag', type=int, default=0,
                    help='1: replace output with influence diagnostics')
    ap.add_argument('-j', dest='solver_mode', type=int, default=5,
                    help='0 OSLM,1 LM,2 OSRLM,3 RLM,4 RTR,5 RRTR,6 NSD')
    ap.add_argument('-L', dest='nulow', type=float, default=2.0)
    ap.add_argument('-H', dest='nuhigh', type=float, default=30.0)
    ap.add_argument('-q', dest='initsol', help='initial solutions file')
    ap.add_argument('-N', dest='epochs', type=int, default=0,
                    help='>0: stochastic calibration epochs')
    ap.add_argument('-M', dest='minibatches', type=int, default=1)
    ap.add_argument('-w', dest='minibands', type=int, default=1)
    ap.add_argument('--multifreq', action='store_true',
                    help='per-channel gradient across each mini-band '
                         '(lbfgs_multifreq semantics) instead of the '
                         'channel-average fit')
    ap.add_argument('-A', dest='nadmm', type=int, default=1,
                    help='consensus updates per epoch (stochastic -w>1)')
    ap.add_argument('-P', dest='npoly', type=int, default=2)
    ap.add_argument('-Q', dest='polytype', type=int, default=0)
    ap.add_argument('-r', dest='admm_rho', type=float, default=5.0)
    ap.add_argument('-U', dest='use_global', type=int, default=0,
                    help='1: residuals from the consensus polynomial '
                         'solution (stochastic -w > 1)')
    ap.add_argument('-R', dest='randomize', type=int, default=0,
                    help='1: alternate EM sweeps reallocate LM iterations '
                         'to high-error groups (lmfit.c weighted_iter)')
    ap.add_argument('-V', dest='verbose', action='store_true')
    return ap


def load_context(args):
    from .. import sky as skymod, msdata, solutions
    from ..ops.reference import SourcePack
    use_gpu = torch.cuda.is_available() if args.gpupredict is None \
        else bool(args.gpupredict)
    device = 'cuda:0' if use_gpu else 'cpu'
    dtype = torch.float32 if use_gpu else torch.float64
    torch.set_num_threads(max(1, args.nthreads))
    ms = msdata.open_ms(args.ms, tilesz=args.tilesz, device=device,
                        dtype=dtype, data_col=args.incol)
    ignore = skymod.read_ignore_file(args.ignfile) if args.ignfile else ()
    clusters = skymod.read_sky_cluster(args.sky, args.cluster, ms.ra0,
                                       ms.dec0, ms.freq0, fmt=args.format,
                                       ignore_ids=ignore,
                                       jd=getattr(ms, 'jd0', None))
    pack = SourcePack(clusters)
    return ms, pack, clusters, device, dtype


def _predict_with_beam(ms, pack, tile, ti, args):
    """Coherencies with the station array beam applied (-B 1;
    predict_visibilities_multifreq_withbeam role). The MS must carry the
    element layout: NpzMS key `element_enu` [N, E, 3] (m) plus optional
    `lon`/`lat` (rad) — the analog of the LOFAR ANTENNA_FIELD tables the
    reference reads (data.cpp:268-288)."""
    from .. import beams
    z = getattr(ms, '_z', {})
    if 'element_enu' not in z:
        raise SystemExit(
            "-B 1 requires element layouts in the MS (key 'element_enu')")
    lon = float(z['lon']) if 'lon' in z else 0.0
    lat = float(z['lat']) if 'lat' in z else 0.92
    cfg = beams.ArrayConfig(list(np.asarray(z['element_enu'])), lon, lat,
                            ms.ra0, ms.dec0)
    T = ms.tilesz
    t0 = float(z['tmjd0']) if 'tmjd0' in z else 56789.0
    tmjd = t0 + (ti * T + np.arange(T) + 0.5) * ms.tdelta / 86400.0
    mode = args.dobeam
    et = getattr(args, 'elem_type', 'auto')
    if et == 'auto':
        et = str(z['elem_type']) if 'elem_type' in z else 'synthetic'
    coeffs = None if et == 'synthetic' else et   # name -> LOFAR tables
    if mode <= 3:
        return beams.predict_coh_withbeam(
            pack, tile.u, tile.v, tile.w, tile.freq0, tile.freq0,
            tile.fdelta, tile.tdelta, tile.dec0, cfg, tmjd,
            ms.bb_tensor(), ms.Nbase, T, mode=mode, coeffs=coeffs)
    # -B 4/5/6: per-channel beam (DOBEAM_*_WB): evaluate the beam at each
    # channel frequency and average, like the discrete channel model
    mode -= 3
    fdelta_ch = tile.fdelta / len(tile.freqs)
    acc = None
    for f in tile.freqs:
        c = beams.predict_coh_withbeam(
            pack, tile.u, tile.v, tile.w, float(f), tile.freq0,
            fdelta_ch, tile.tdelta, tile.dec0, cfg, tmjd,
            ms.bb_tensor(), ms.Nbase, T, mode=mode, coeffs=coeffs)
        acc = c if acc is None else acc + c
    return acc / len(tile.freqs)


def _predict_channel_with_beam(ms, pack, tile, ti, args, freq,
                               fdelta_ch):
    """One channel's beamed coherencies for the residual path (-B with
    multifreq residuals)."""
    from .. import beams
    z = getattr(ms, '_z', {})
    lon = float(z['lon']) if 'lon' in z else 0.0
    lat = float(z['lat']) if 'lat' in z else 0.92
    cfg = beams.ArrayConfig(list(np.asarray(z['element_enu'])), lon, lat,
                            ms.ra0, ms.dec0)
    T = ms.tilesz
    t0 = float(z['tmjd0']) if 'tmjd0' in z else 56789.0
    tmjd = t0 + (ti * T + np.arange(T) + 0.5) * ms.tdelta / 86400.0
    mode = args.dobeam
    if mode > 3:
        mode -= 3
    et = getattr(args, 'elem_type', 'auto')
    if et == 'auto':
        et = str(z['elem_type']) if 'elem_type' in z else 'synthetic'
    coeffs = None if et == 'synthetic' else et
    return beams.predict_coh_withbeam(
        pack, tile.u, tile.v, tile.w, freq, tile.freq0, fdelta_ch,
        tile.tdelta, tile.dec0, cfg, tmjd, ms.bb_tensor(), ms.Nbase, T,
        mode=mode, coeffs=coeffs)


def uv_flags(tile, args):
    """Baseline uv-cut flags (predict.c flag=2 semantics for -x/-y)."""
    uvlen = torch.sqrt(tile.u ** 2 + tile.v ** 2) * tile.freq0
    return tile.flags | (uvlen < args.min_uvcut) | (uvlen > args.max_uvcut)


def run_calibration(args):
    from ..solvers import sage
    from .. import solutions
    ms, pack, clusters, device, dtype = load_context(args)
    cdtype = torch.complex64 if dtype == torch.float32 else torch.complex128
    state = sage.CalState(pack, ms.N, device=device, dtype=cdtype)
    if args.initsol:
        hdr, tiles = solutions.read_solutions(args.initsol)
        if tiles:
            state.J = solutions.reorder_read_tile(
                tiles[0], state.nchunks).to(device=device, dtype=cdtype)
    opts = sage.SageSolveOptions(
        max_emiter=args.max_emiter, max_iter=args.max_iter,
        solver_mode=args.solver_mode, robust_nulow=args.nulow,
        robust_nuhigh=args.nuhigh, lbfgs_iters=args.max_lbfgs if
        args.max_lbfgs > 0 else 0, randomize=bool(args.randomize))
    opts.lbfgs_m = args.lbfgs_m
    writer = None
    if args.solfile:
        writer = solutions.SolutionWriter(
            args.solfile, ms.freq0, ms.fdelta,
            ms.tilesz * ms.tdelta / 60.0, ms.N, state.M, state.Mt)
    pinit = state.J.clone()
    for ti, tile in enumerate(ms.tiles()):
        t0 = time.time()
        flags = uv_flags(tile, args)
        if args.whiten:
            from ..utils import taper
            tile.x, _ = taper.whiten_data(tile.x, tile.u, tile.v,
                                          tile.freq0)
        if args.dobeam:
            cohs = _predict_with_beam(ms, pack, tile, ti, args)
        else:
            cohs = sage.precalc_coherencies(pack, tile)
        if device != 'cpu':
            cohs = cohs.to(torch.complex64)
        bb = ms.bb_tensor(device=device)
        res0, res1 = sage.sagefit(state, cohs, tile, bb, opts, flags=flags)
        ccid = args.ccid if args.ccid != -99999 else None
        if args.phase_only and ccid is not None:
            # phase-only: normalize the correcting cluster's J to unit
            # amplitude before inversion (residual.c phase-only path)
            ids = getattr(pack, 'cluster_ids', list(range(state.M)))
            match = [i for i, c in enumerate(ids) if c == ccid]
            if match:
                o = state.chunk_off[match[0]]
                nc = state.nchunks[match[0]]
                Jc = state.J[o:o + nc]
                state.J[o:o + nc] = Jc / Jc.abs().clamp_min(1e-12)
        coh_fn = None
        if args.dobeam:
            # beamed residuals (calculate_residuals_multifreq_withbeam
            # role): per-channel coherencies WITH the station beam, so
            # the subtracted model matches what was calibrated against
            fdch = tile.fdelta / len(tile.freqs)
            coh_fn = (lambda f, _t=tile, _ti=ti:
                      _predict_channel_with_beam(ms, pack, _t, _ti, args,
                                                 f, fdch))
        xres = sage.calculate_residuals_multifreq(
            state, pack, tile, bb, ccid=ccid, rho=args.rho_corr,
            coh_fn=coh_fn)
        if args.dochan:
            # per-channel refinement (-b 1, fullbatch_mode.cpp:453-499):
            # polish each channel's solutions with a short joint LBFGS and
            # recompute that channel's residual
            from ..solvers import lbfgs as lbfgs_mod
            from ..ops import dispatch as dops
            fdelta_ch = tile.fdelta / len(tile.freqs)
            for fi, f in enumerate(tile.freqs):
                if coh_fn is not None:   # beamed per-channel (-B + -b)
                    cohs_f = coh_fn(float(f))
                else:
                    cohs_f = dops.predict_coh(pack, tile.u, tile.v,
                                              tile.w, float(f),
                                              tile.freq0, fdelta_ch,
                                              tile.tdelta, tile.dec0)
                if cohs_f.dtype != state.J.dtype:
                    cohs_f = cohs_f.to(state.J.dtype)

                class _T:  # channel view of the tile
                    pass
                tf = _T()
                tf.x = tile.xo[fi]
                tf.tilesz, tf.Nbase = tile.tilesz, tile.Nbase
                st2 = sage.CalState(pack, ms.N, device=tile.x.device,
                                    dtype=state.J.dtype)
                st2.J = state.J.clone()
                st2.nu = state.nu.clone()
                o2 = sage.SageSolveOptions(
                    max_emiter=1, max_iter=4,
                    solver_mode=args.solver_mode,
                    lbfgs_iters=max(args.max_lbfgs, 4))
                lbfgs_mod.polish(st2, cohs_f, tf, bb, o2)
                V = sage.total_model(st2, cohs_f, bb, tile.tilesz,
                                     tile.Nbase)
                xres[fi] = tile.xo[fi] - V
        if args.dodiag:
            from ..solvers import diagnostics as diagmod
            cohs0 = sage.precalc_coherencies(pack, tile)
            if cohs0.dtype != state.J.dtype:
                cohs0 = cohs0.to(state.J.dtype)
            lev = diagmod.influence_map(state, cohs0, tile, bb)
            xres = lev[None, :, None, None].expand_as(xres).to(xres.dtype)
        ms.write_column(args.outcol, ti, xres)
        if writer:
            writer.write_tile(state)
        # divergence safeguard (fullbatch_mode.cpp:622-632)
        if not np.isfinite(res1) or (res0 > 0 and res1 > 5 * res0):
            state.J = pinit.clone()
        mean_nu = float(state.nu.mean())
        if args.verbose:
            print(f"tile {ti}: solve took {time.time() - t0:.2f} s")
        print(f"tile {ti}: residual {res0:.6f} -> {res1:.6f}, mean nu "
              f"{mean_nu:.1f} ({time.time() - t0:.1f}s)")
    ms.save()
    if writer:
        writer.close()


def run_simulation(args):
    """-a 1/2/3: predict (optionally corrupted by solutions), write/add/
    subtract (fullbatch_mode.cpp:536-591)."""
    from ..solvers import sage
    from ..ops import dispatch as ops
    from .. import solutions
    ms, pack, clusters, device, dtype = load_context(args)
    cdtype = torch.complex64 if dtype == torch.float32 else torch.complex128
    state = sage.CalState(pack, ms.N, device=device, dtype=cdtype)
    if args.solfile:
        hdr, tiles = solutions.read_solutions(args.solfile)
        if tiles:
            state.J = solutions.reorder_read_tile(
                tiles[0], state.nchunks).to(device=device, dtype=cdtype)
    bb = ms.bb_tensor(device=device)
    for ti, tile in enumerate(ms.tiles()):
        V = torch.zeros_like(tile.xo)
        fdelta_ch = tile.fdelta / len(tile.freqs)
        for fi, f in enumerate(tile.freqs):
            if args.dobeam:
                # predict WITH the station beam
                # (predict_visibilities_multifreq_withbeam,
                # fullbatch_mode.cpp:538-559)
                cohs = _predict_channel_with_beam(ms, pack, tile, ti,
                                                  args, float(f),
                                                  fdelta_ch)
            else:
                cohs = ops.predict_coh(pack, tile.u, tile.v, tile.w,
                                       float(f), tile.freq0, fdelta_ch,
                                       tile.tdelta, tile.dec0)
            if cohs.dtype != cdtype:
                cohs = cohs.to(cdtype)
            V[fi] = sage.total_model(state, cohs, bb, tile.tilesz,
                                     tile.Nbase)
        if args.dosim == 2:
            out = tile.xo + V
        elif args.dosim == 3:
            out = tile.xo - V
        else:
            out = V
        ms.write_column(args.outcol, ti, out)
        print(f"tile {ti}: simulated (mode {args.dosim})")
    ms.save()


def run_stochastic(args):
    """-N epochs: minibatch (bandpass-consensus when -w > 1) calibration
    (minibatch_mode.cpp / minibatch_consensus_mode.cpp)."""
    if getattr(args, 'dobeam', 0):
        raise SystemExit(
            'sagecal: -B with stochastic mode (-N) is not supported yet; '
            'failing loudly instead of silently calibrating beam-free '
            '(ROUND3_NOTES)')
    from ..solvers.stochastic import MinibatchConsensusCalibration
    ms, pack, clusters, device, dtype = load_context(args)
    cdtype = torch.complex64 if dtype == torch.float32 else torch.complex128
    cal = MinibatchConsensusCalibration(
        pack, ms.N, ms.freqs, nsolbw=args.minibands, Npoly=args.npoly,
        poly_type=args.polytype, rho=args.admm_rho, device=device,
        dtype=cdtype, multifreq=getattr(args, 'multifreq', False))
    bb = ms.bb_tensor(device=device)
    for ti, tile in enumerate(ms.tiles()):
        for ep in range(args.epochs):
            # -A extra consensus sweeps per epoch
            # (minibatch_consensus_mode.cpp admm loop)
            for _ in range(max(1, args.nadmm)):
                cal.epoch(tile, bb, nmb=args.minibatches,
                          lbfgs_iters=args.max_lbfgs,
                          robust_nu=(args.nulow + args.nuhigh) / 2)
        xres = cal.residuals(tile, bb, use_global=bool(args.use_global))
        ms.write_column(args.outcol, ti, xres)
        print(f"tile {ti}: stochastic epochs {args.epochs}, "
              f"res {float(xres.abs().pow(2).mean()):.6f}")
    ms.save()


def _run_one(args):
    if args.dosim > 0:
        run_simulation(args)
    elif args.epochs > 0:
        run_stochastic(args)
    else:
        run_calibration(args)


def main(argv=None):
    args = build_argparser().parse_args(argv)
    names = [args.ms] if args.ms else []
    if args.mslist:
        with open(args.mslist) as f:
            names += [l.strip() for l in f if l.strip()]
    if not names:
        print("need -d MS or -f MSlist", file=sys.stderr)
        return 1
    base_sol = args.solfile
    # -f MSlist: every MS processed in sequence (main.cpp loops the
    # mslist the same way; per-MS solution files keep the outputs apart)
    for i, name in enumerate(names):
        args.ms = name
        if base_sol and len(names) > 1:
            args.solfile = f"{base_sol}.ms{i}"
        _run_one(args)
    return 0


if __name__ == '__main__':
    sys.exit(main())
