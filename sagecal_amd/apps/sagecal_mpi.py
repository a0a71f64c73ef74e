"""sagecal-mpi CLI — distributed multi-band consensus calibration.

The reference's MPI hub (src/MPI/sagecal_master.cpp + sagecal_slave.cpp)
re-designed for one rank per GPU over RCCL/xGMI (torchrun launcher, gloo on
CPU): every rank loads its own sub-band MS; the master's Z-update runs
replicated on all ranks after one fused all-reduce (consensus/admm.py).

Launch:
  torchrun --nnodes 1 --nproc-per-node W --master-addr 127.0.0.1 \\
      sagecal_mpi.py -f mslist.txt -s sky.txt -c cluster.txt -A 10 -P 2

-f: a file listing one MS path per line, OR a glob pattern. With as many
MSs as ranks, rank r takes entry r. With MORE MSs than ranks, each rank
owns names[rank::world] and rotates through them one ADMM iteration at a
time (sagecal_master.cpp:1055 Scurrent multiplexing; this path currently
solves with the core consensus options — -G/-X/-p extras apply to the
one-band-per-rank mode).
"""
import argparse
import os
import sys

import numpy as np
import torch
import torch.distributed as dist


def build_argparser():
    ap = argparse.ArgumentParser(prog='sagecal-mpi')
    ap.add_argument('-f', dest='mslist', required=True,
                    help='file listing per-band MS paths (one per rank), '
                         'or a glob pattern like "*.ms" (the reference '
                         'master discovers MSs by pattern)')
    ap.add_argument('-s', dest='sky', required=True)
    ap.add_argument('-c', dest='cluster', required=True)
    ap.add_argument('-p', dest='solfile', help='per-rank solutions file')
    ap.add_argument('-F', dest='format', type=int, default=0)
    ap.add_argument('-e', dest='max_emiter', type=int, default=3)
    ap.add_argument('-g', dest='max_iter', type=int, default=10)
    ap.add_argument('-l', dest='max_lbfgs', type=int, default=0)
    ap.add_argument('-t', dest='tilesz', type=int, default=10)
    ap.add_argument('-j', dest='solver_mode', type=int, default=5)
    ap.add_argument('-A', dest='nadmm', type=int, default=10)
    ap.add_argument('-P', dest='npoly', type=int, default=2)
    ap.add_argument('-Q', dest='polytype', type=int, default=0)
    ap.add_argument('-r', dest='admm_rho', type=float, default=5.0)
    ap.add_argument('-G', dest='rhofile',
                    help='per-cluster regularization file')
    ap.add_argument('-a', '-C', dest='use_bb', type=int, default=0,
                    help='>0: Barzilai-Borwein adaptive rho (the '
                         'reference flag is -C, MPI/main.cpp)')
    ap.add_argument('-U', dest='use_global', type=int, default=0,
                    help='1: residuals from the global solution B Z')
    ap.add_argument('-X', dest='spatial',
                    help='L2,L1,order,fista_iters,cadence (spatial reg)')
    ap.add_argument('-u', dest='spatial_alpha', type=float, default=0.0)
    ap.add_argument('-L', dest='nulow', type=float, default=2.0)
    ap.add_argument('-H', dest='nuhigh', type=float, default=30.0)
    ap.add_argument('-n', dest='nthreads', type=int, default=6)
    ap.add_argument('-m', dest='lbfgs_m', type=int, default=7)
    ap.add_argument('-x', dest='min_uvcut', type=float, default=0.0)
    ap.add_argument('-y', dest='max_uvcut', type=float, default=1e9)
    ap.add_argument('-k', dest='ccid', type=int, default=-99999,
                    help='correct residuals with this cluster id')
    ap.add_argument('-o', dest='rho_corr', type=float, default=1e-9)
    ap.add_argument('-K', dest='nskip', type=int, default=0,
                    help='skip this many solution intervals first')
    ap.add_argument('-T', dest='nend', type=int, default=0,
                    help='>0: stop after this solution interval')
    ap.add_argument('-W', dest='whiten', type=int, default=0)
    ap.add_argument('-B', dest='dobeam', type=int, default=0,
                    help='beam in predict (as in sagecal: 1 array, '
                         '2 array+element, 3 element; the reference '
                         'sagecal-mpi clamps >3 to 1, MPI/main.cpp:147)')
    ap.add_argument('--elem-type', dest='elem_type', default='auto',
                    choices=['auto', 'synthetic', 'lba', 'hba', 'alo'])
    ap.add_argument('-D', dest='diffuse',
                    help='cluster_id,gamma: treat this cluster as the '
                         'diffuse foreground model when -X is on — its '
                         'coherencies are re-predicted with the FISTA '
                         'spatial model each cadence '
                         '(recalculate_diffuse_coherencies, '
                         'sagecal_slave.cpp:669-694; MPI/main.cpp:76)')
    ap.add_argument('-M', dest='mdl', action='store_true',
                    help='evaluate AIC/MDL over polynomial orders 1..-P '
                         'after the first tile and print the suggestion '
                         '(reference -M, sagecal_master.cpp:992)')
    ap.add_argument('-I', dest='incol', default=None,
                    help='input data column (reference -I)')
    ap.add_argument('-J', dest='phase_only', type=int, default=0,
                    help='1: phase-only correction (with -k)')
    ap.add_argument('-q', dest='initsol', default=None,
                    help='warm-start solutions file (per rank)')
    ap.add_argument('-R', dest='randomize', type=int, default=0,
                    help='randomized/weighted EM iteration allocation')
    ap.add_argument('-N', dest='epochs', type=int, default=0,
                    help='>0: FEDERATED stochastic calibration — every '
                         'rank runs minibatch bandpass consensus and the '
                         'per-rank Z are manifold-averaged across ranks '
                         '(sagecal_stochastic_master/slave role)')
    ap.add_argument('-M2', dest='minibatches', type=int, default=2,
                    help='time minibatches per epoch (stochastic mode; '
                         'the reference letter -M is taken by our MDL '
                         'advisory flag)')
    ap.add_argument('-w', dest='minibands', type=int, default=1,
                    help='mini-bands for stochastic bandpass mode')
    ap.add_argument('--fed-alpha', dest='fed_alpha', type=float,
                    default=0.1,
                    help='federated averaging strength alpha '
                         '(find_prod_inverse_full_fed, '
                         'sagecal_stochastic_slave.cpp:563)')
    ap.add_argument('-E', dest='gpupredict', type=int, default=1,
                    help='accepted for reference-CLI compatibility '
                         '(model prediction runs on the GPU whenever '
                         'one is present)')
    ap.add_argument('-S', dest='heapsize', type=float, default=0,
                    help='accepted for reference-CLI compatibility '
                         '(GPU heap MB — not applicable here)')
    ap.add_argument('-i', dest='dodiag', type=int, default=0,
                    help='1: replace output with influence diagnostics '
                         '(reference MPI -i)')
    ap.add_argument('-O', dest='outcol', default='residual')
    ap.add_argument('-V', dest='verbose', action='store_true')
    return ap


def _run_multiplexed(args, names, rank, world, device, dtype, cdtype):
    from .. import sky as skymod, msdata
    from ..ops.reference import SourcePack
    from ..solvers import sage
    from ..consensus.admm import MultiplexedADMM
    my_names = names[rank::world]
    mss, bandsets, tilesets, my_ids, f0_local = [], [], [], [], []
    for i, name in enumerate(my_names):
        ms = msdata.open_ms(name, tilesz=args.tilesz, device=device,
                            dtype=dtype)
        mss.append(ms)
        my_ids.append(rank + i * world)
        f0_local.append(ms.freq0)
    # gather every band's centre frequency
    F = len(names)
    use_gpu = device != 'cpu'
    f0s = torch.zeros(F, device=device if use_gpu else 'cpu')
    for bi, f0 in zip(my_ids, f0_local):
        f0s[bi] = f0
    if world > 1:
        dist.all_reduce(f0s)
    f0s = f0s.cpu()
    freq0_global = float(f0s.mean())
    clusters = skymod.read_sky_cluster(args.sky, args.cluster,
                                       mss[0].ra0, mss[0].dec0,
                                       freq0_global, fmt=args.format)
    pack = SourcePack(clusters)
    rho = torch.full((pack.M,), args.admm_rho)
    opts = sage.SageSolveOptions(
        max_emiter=args.max_emiter, max_iter=args.max_iter,
        solver_mode=args.solver_mode, robust_nulow=args.nulow,
        robust_nuhigh=args.nuhigh, lbfgs_iters=max(args.max_lbfgs, 0))
    for ms in mss:
        bandsets.append({'state': sage.CalState(pack, ms.N,
                                                device=device,
                                                dtype=cdtype),
                         'freq0': ms.freq0})
    adm = MultiplexedADMM(bandsets, my_ids, f0s.tolist(), freq0_global,
                          rank, world, Npoly=args.npoly,
                          poly_type=args.polytype, rho=rho)
    for ti_all in range(min(ms.n_tiles() for ms in mss)):
        tilesets = []
        for ms in mss:
            tile = ms.load_tile(ti_all)
            cohs = sage.precalc_coherencies(pack, tile)
            if cohs.dtype != cdtype:
                cohs = cohs.to(cdtype)
            tilesets.append({'cohs': cohs, 'tile': tile,
                             'bb': ms.bb_tensor(device=device)})
        res = adm.run(tilesets, opts, n_admm=args.nadmm)
        for k, ms in enumerate(mss):
            st = bandsets[k]['state']
            xres = sage.calculate_residuals_multifreq(st, pack,
                                                      tilesets[k]['tile'],
                                                      tilesets[k]['bb'])
            ms.write_column(args.outcol, ti_all, xres)
        if rank == 0 or args.verbose:
            rr = {bi: (round(a, 4), round(b, 4))
                  for bi, (a, b) in res.items()}
            print(f"rank {rank} tile {ti_all}: bands {rr}")
    for ms in mss:
        ms.save()
    if world > 1:
        dist.destroy_process_group()
    return 0


def _run_federated_stochastic(args, ms, pack, device, cdtype, rank,
                              world):
    """-N epochs in sagecal-mpi: each rank runs stochastic bandpass
    consensus on its own band; per-rank Z is pulled toward the
    manifold-averaged global Z with strength -S alpha
    (sagecal_stochastic_slave.cpp:563 find_prod_inverse_full_fed +
    :799-878 X update; master averaging
    sagecal_stochastic_master.cpp:340-350)."""
    from ..solvers.stochastic import MinibatchConsensusCalibration
    cal = MinibatchConsensusCalibration(
        pack, ms.N, ms.freqs, nsolbw=args.minibands, Npoly=args.npoly,
        poly_type=args.polytype, rho=args.admm_rho, device=device,
        dtype=cdtype, fed_alpha=args.fed_alpha if world > 1 else 0.0,
        world=world, rank=rank)
    bb = ms.bb_tensor(device=device)
    for ti, tile in enumerate(ms.tiles()):
        if ti < args.nskip:
            continue
        if args.nend > 0 and ti >= args.nend:
            break
        for ep in range(args.epochs):
            for _ in range(max(1, args.nadmm)):
                cal.epoch(tile, bb, nmb=args.minibatches,
                          lbfgs_iters=max(args.max_lbfgs, 8),
                          robust_nu=(args.nulow + args.nuhigh) / 2)
        xres = cal.residuals(tile, bb,
                             use_global=bool(args.use_global))
        ms.write_column(args.outcol, ti, xres)
        if rank == 0 or args.verbose:
            print(f"rank {rank} tile {ti}: federated stochastic "
                  f"epochs {args.epochs}, "
                  f"res {float(xres.abs().pow(2).mean()):.6f}")
    ms.save()
    if world > 1 and dist.is_initialized():
        dist.destroy_process_group()
    return 0


def _diffuse_hook(args, pack, tile, bb):
    """Build the diffuse-model recalculation hook for -D cluster_id
    (reference sp_diffuse_id): replaces the designated cluster's
    coherencies with the spatial-model shapelet predict each cadence."""
    spec = getattr(args, 'diffuse', None)
    if not spec or not getattr(args, 'spatial', None):
        return None
    ddid = int(str(spec).split(',')[0])
    import torch as _t
    ids = getattr(pack, 'cluster_ids', [])
    match = [i for i, c in enumerate(ids) if c == ddid]
    if not match:
        print(f"sagecal-mpi: -D cluster id {ddid} not found; ignored")
        return None
    ci = match[0]
    s0, s1 = int(pack.cluster_off[ci]), int(pack.cluster_off[ci + 1])
    gi = next((g for g in range(s0, s1)
               if g in getattr(pack, 'shapelets', {})), None)
    if gi is None:
        print(f"sagecal-mpi: -D cluster {ddid} has no shapelet source; "
              f"ignored")
        return None
    n0, beta_c, modes = pack.shapelets[gi]
    Cm = _t.zeros(n0 * n0, 2, 2, dtype=_t.complex128)
    Cm[:, 0, 0] = _t.as_tensor(np.asarray(modes), dtype=_t.float64)
    Cm[:, 1, 1] = Cm[:, 0, 0]
    lmn = (float(pack.ll[gi]), float(pack.mm[gi]), float(pack.nn1[gi]))

    def hook(adm, cohs):
        out = adm.diffuse_coherencies(tile.u, tile.v, tile.w, bb, Cm,
                                      beta_c, lmn, tile.freq0,
                                      tile.fdelta)
        if out is not None:
            cohs[ci] = out.to(device=cohs.device, dtype=cohs.dtype)
    return hook


def main(argv=None):
    args = build_argparser().parse_args(argv)
    rank = int(os.environ.get('RANK', '0'))
    world = int(os.environ.get('WORLD_SIZE', '1'))
    local_rank = int(os.environ.get('LOCAL_RANK', str(rank)))
    use_gpu = torch.cuda.is_available()
    if world > 1 and not dist.is_initialized():
        dist.init_process_group('nccl' if use_gpu else 'gloo')
    if use_gpu:
        torch.cuda.set_device(local_rank)
    device = f'cuda:{local_rank}' if use_gpu else 'cpu'
    dtype = torch.float32 if use_gpu else torch.float64
    cdtype = torch.complex64 if use_gpu else torch.complex128

    from .. import sky as skymod, msdata, solutions
    from ..ops.reference import SourcePack
    from ..solvers import sage
    from ..consensus.admm import ConsensusADMM

    import glob as _glob
    if os.path.isfile(args.mslist):
        with open(args.mslist) as f:
            names = [l.strip() for l in f if l.strip()]
    else:
        names = sorted(_glob.glob(args.mslist))
        if not names:
            print(f"no MS matches {args.mslist}", file=sys.stderr)
            return 1
    torch.set_num_threads(max(1, args.nthreads))
    if len(names) > world:
        # more MSs than ranks: this rank owns names[rank::world] and
        # rotates through them per ADMM iteration
        # (sagecal_master.cpp:1055 Scurrent multiplexing)
        return _run_multiplexed(args, names, rank, world, device, dtype,
                                cdtype)
    my_ms = names[rank % len(names)]
    mskw = {}
    if getattr(args, 'incol', None):
        mskw['data_col'] = args.incol
    ms = msdata.open_ms(my_ms, tilesz=args.tilesz, device=device,
                        dtype=dtype, **mskw)
    clusters = skymod.read_sky_cluster(args.sky, args.cluster, ms.ra0,
                                       ms.dec0, ms.freq0, fmt=args.format,
                                       jd=getattr(ms, 'jd0', None))
    pack = SourcePack(clusters)
    state = sage.CalState(pack, ms.N, device=device, dtype=cdtype)
    if getattr(args, 'initsol', None):
        hdr_i, tiles_i = solutions.read_solutions(args.initsol)
        if tiles_i:
            state.J = solutions.reorder_read_tile(
                tiles_i[0], state.nchunks).to(device=device, dtype=cdtype)
    if getattr(args, 'epochs', 0) > 0:
        # federated stochastic mode (-N>0): per-rank minibatch bandpass
        # consensus + manifold-averaged global Z across ranks (the
        # reference's sagecal_stochastic_master/slave dispatch,
        # MPI/main.cpp:336-370)
        return _run_federated_stochastic(args, ms, pack, device, cdtype,
                                         rank, world)

    rho = torch.full((pack.M,), args.admm_rho)
    if args.rhofile:
        arho, _ = skymod.read_arho_file(args.rhofile, clusters)
        rho = torch.tensor(arho, dtype=torch.float64)
        rho[rho == 0] = args.admm_rho

    # gather every band's centre frequency (TAG_MSAUX metadata exchange);
    # NCCL needs device tensors for collectives
    f0s = torch.zeros(world, device=device if use_gpu else 'cpu')
    f0s[rank] = ms.freq0
    if world > 1:
        dist.all_reduce(f0s)
    f0s = f0s.cpu()
    freq0_global = float(f0s.mean())

    spatial = None
    centroids = None
    if args.spatial:
        v = args.spatial.split(',')
        spatial = (float(v[0]), float(v[1]), int(v[2]), int(v[3]),
                   int(v[4]))
        centroids = (np.array([c.ll.mean() for c in clusters]),
                     np.array([c.mm.mean() for c in clusters]))
    adm = ConsensusADMM(state, f0s.tolist(), freq0_global, rank, world,
                        Npoly=min(args.npoly, world), poly_type=args.polytype,
                        rho=rho, use_bb=bool(args.use_bb), spatial=spatial,
                        spatial_alpha=args.spatial_alpha,
                        centroids=centroids)
    opts = sage.SageSolveOptions(
        max_emiter=args.max_emiter, max_iter=args.max_iter,
        solver_mode=args.solver_mode, robust_nulow=args.nulow,
        robust_nuhigh=args.nuhigh,
        lbfgs_iters=max(args.max_lbfgs, 0),
        randomize=bool(getattr(args, 'randomize', 0)))
    opts.lbfgs_m = args.lbfgs_m
    writer = None
    zwriter = None
    if args.solfile:
        writer = solutions.SolutionWriter(
            f"{args.solfile}.rank{rank}", ms.freq0, ms.fdelta,
            ms.tilesz * ms.tdelta / 60.0, ms.N, state.M, state.Mt)
        if rank == 0:
            # global Z solution file (the master's write,
            # sagecal_master.cpp:1165): Z is replicated on every rank
            zwriter = solutions.GlobalZWriter(
                f"{args.solfile}.Z", freq0_global, ms.N, state.M,
                state.Mt, adm.Npoly)
    bb = ms.bb_tensor(device=device)
    for ti, tile in enumerate(ms.tiles()):
        if ti < args.nskip:
            continue
        if args.nend > 0 and ti >= args.nend:
            break
        if args.whiten:
            from ..utils import taper
            tile.x, _ = taper.whiten_data(tile.x, tile.u, tile.v,
                                          tile.freq0)
        # uv cuts flag baselines out of the solve (predict.c flag=2)
        uvlen = torch.sqrt(tile.u ** 2 + tile.v ** 2) * tile.freq0
        flags = tile.flags | (uvlen < args.min_uvcut) | \
            (uvlen > args.max_uvcut)
        if getattr(args, 'dobeam', 0):
            from .sagecal import _predict_with_beam
            if args.dobeam > 3:       # MPI/main.cpp:147 clamp
                args.dobeam = 1
            cohs = _predict_with_beam(ms, pack, tile, ti, args)
        else:
            cohs = sage.precalc_coherencies(pack, tile)
        if cohs.dtype != cdtype:
            cohs = cohs.to(cdtype)
        # TAG_FRATIO: weigh this band's rho by its unflagged fraction
        adm.set_fratio(float((~flags).float().mean()))
        dhook = _diffuse_hook(args, pack, tile, bb)
        res0, res1 = adm.run(cohs, tile, bb, opts, n_admm=args.nadmm,
                             flags=flags,
                             verbose=args.verbose and rank == 0,
                             diffuse_hook=dhook)
        if args.use_global:
            state.J = adm.global_solution()
        ccid = args.ccid if args.ccid != -99999 else None
        if getattr(args, 'phase_only', 0) and ccid is not None:
            # phase-only correction (-J, residual.c phase-only path)
            ids = getattr(pack, 'cluster_ids', list(range(state.M)))
            match = [i for i, c in enumerate(ids) if c == ccid]
            if match:
                o = state.chunk_off[match[0]]
                nc = state.nchunks[match[0]]
                Jc = state.J[o:o + nc]
                state.J[o:o + nc] = Jc / Jc.abs().clamp_min(1e-12)
        coh_fn = None
        if getattr(args, 'dobeam', 0):
            from .sagecal import _predict_channel_with_beam
            fdch = tile.fdelta / len(tile.freqs)
            coh_fn = (lambda f, _t=tile, _ti=ti:
                      _predict_channel_with_beam(ms, pack, _t, _ti,
                                                 args, f, fdch))
        xres = sage.calculate_residuals_multifreq(state, pack, tile, bb,
                                                  ccid=ccid,
                                                  rho=args.rho_corr,
                                                  coh_fn=coh_fn)
        if getattr(args, 'dodiag', 0):
            from ..solvers import diagnostics as diagmod
            cohs0 = cohs if cohs.dtype == state.J.dtype \
                else cohs.to(state.J.dtype)
            lev = diagmod.influence_map(state, cohs0, tile, bb)
            xres = lev[None, :, None, None].expand_as(xres).to(xres.dtype)
        ms.write_column(args.outcol, ti, xres)
        if writer:
            writer.write_tile(state)
        if zwriter:
            zwriter.write_tile(adm.Z)
        if args.mdl and ti == 0:
            # gather every band's J and score polynomial orders 1..Npoly
            from ..consensus import mdl as mdl_mod
            Jflat = (state.J.reshape(state.Mt, -1)
                     [[state.chunk_off[c] for c in range(state.M)]])
            Jb = torch.view_as_real(Jflat).reshape(1, state.M, -1).double()
            if world > 1:
                gath = [torch.zeros_like(Jb) for _ in range(world)]
                dist.all_gather(gath, Jb)
                Jall = torch.cat(gath, dim=0)
            else:
                Jall = Jb
            best, scores = mdl_mod.minimum_description_length(
                torch.view_as_complex(
                    Jall.reshape(world, state.M, -1, 2)
                    .contiguous()).cpu(),
                rho.double().cpu(), f0s.double().cpu(), freq0_global,
                polytype=args.polytype, Kstart=1,
                Kfinish=min(args.npoly, world))
            if rank == 0:
                for k, (mv, av) in sorted(scores.items()):
                    print(f"MDL: Npoly={k}: MDL={mv:.3f} AIC={av:.3f}")
                print(f"MDL: suggested Npoly={best}")
        if rank == 0 or args.verbose:
            print(f"rank {rank} tile {ti}: res {res0:.6f} -> {res1:.6f} "
                  f"rho[0]={float(adm.rho[0]):.2f}")
    ms.save()
    if writer:
        writer.close()
    if zwriter:
        zwriter.close()
    if rank == 0 and args.solfile and args.spatial:
        Zsp = adm.spatial_coefficients()
        if Zsp is not None:
            # render the spatial model amplitude as PPM
            # (plot_spatial_model / pngoutput.c DEBUG dump made a
            # first-class output)
            from ..utils import image as img_mod
            from ..consensus import fista as fista_mod
            beta = float(max(np.max(np.abs(centroids[0])),
                             np.max(np.abs(centroids[1])), 1e-3))
            order = int(args.spatial.split(',')[2])
            ext = 2.0 * beta
            img_mod.plot_spatial_model(
                f"{args.solfile}.spatial.ppm", Zsp,
                lambda l, m: fista_mod.spatial_basis(l, m, order, beta),
                grid=64, extent=ext)
            # spatial_<solfile>: rows p of the [P, 2G] real coefficient
            # matrix (sagecal_master.cpp spatial write)
            import os as _os
            d, b = _os.path.split(args.solfile)
            with open(_os.path.join(d, f"spatial_{b}"), 'w') as fh:
                fh.write("# spatial model coefficients: P x 2G\n")
                for p in range(Zsp.shape[0]):
                    fh.write(str(p) + ''.join(
                        f" {v.real:e} {v.imag:e}" for v in
                        Zsp[p].tolist()) + "\n")
    if world > 1:
        dist.destroy_process_group()
    return 0


if __name__ == '__main__':
    sys.exit(main())
