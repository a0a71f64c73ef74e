#!/usr/bin/env python3
"""Flagship benchmark: visibilities/sec calibrated, 64-station x
10-direction LM + Student's-t robust noise (BASELINE.json config).

One step = full calibration of one solution interval (tile):
  coherency predict -> SAGE EM (robust LM, fixed schedule) -> joint
  refinement -> per-channel residual computation.

Run:  python bench.py [--gpus N] [--steps K] [--warmup W]
Multi-GPU (driver): torchrun --nproc-per-node N bench.py --gpus N ...
  each rank calibrates its own frequency sub-band (the sagecal-mpi
  frequency-parallel decomposition) coupled by consensus-ADMM polynomial
  allreduce over RCCL/xGMI; weak scaling (fixed work per GPU).
"""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch


def build_problem(args, device, dtype, rank=0, world=1):
    from sagecal_amd import sky, msdata
    from sagecal_amd.ops.reference import SourcePack

    srcs, clist = sky.make_synthetic_sky(
        M=args.dirs, nsrc_per_cluster=args.srcs, seed=17)
    # Batch P solution intervals per step via the hybrid time-chunk
    # machinery (reference -t tilesz + cluster chunk column,
    # lmfit.c:893-967): every cluster solves P independent Jones
    # solutions, one per `tilesz`-slot interval — P x em_group problems
    # per batched LM solve, sized to fill 256 CUs.
    P = max(1, getattr(args, 'intervals', 1))
    # clamp the interval batch so the resident per-cluster coherencies
    # stay within a sane fraction of 288 GB HBM (or CPU RAM): cohs are
    # [dirs, rows, 2, 2] complex64 + data ~3x that
    nb = args.stations * (args.stations - 1) // 2
    bytes_per_iv = nb * args.tilesz * (args.dirs + 3) * 32
    cap = max(1, int(96e9 // max(bytes_per_iv, 1)))
    if P > cap:
        print(f"# bench: clamping --intervals {P} -> {cap} "
              f"(coherency residency budget)", file=sys.stderr)
        P = cap
        args.intervals = cap
    if P > 1:
        clist = [(cid, P * nchunk, names) for cid, nchunk, names in clist]
    nshap = getattr(args, 'shapelet_dirs', 0)
    if nshap:
        # make the first sources of the first nshap clusters shapelets
        from sagecal_amd import shapelet as shmod
        rng_s = np.random.default_rng(23)
        for ci in range(min(nshap, len(clist))):
            name = clist[ci][2][0]
            s = srcs[name]
            s.stype = 4
            s.eX = s.eY = 1.0
            s.sh_n0 = 3
            s.sh_beta = 1e-3
            s.sh_coeff = rng_s.standard_normal(9) * [1, .3, .1, .3, .2,
                                                     .05, .1, .05, .02]
    clusters = sky.build_clusters(srcs, clist, 0.0, np.pi / 4,
                                  args.freq0)
    pack = SourcePack(clusters)
    # per-rank frequency sub-band (sagecal-mpi: one MS band per worker)
    freq0 = args.freq0 + rank * args.bandwidth
    # synthetic truth is generated in fp64; for large arrays generate on
    # the GPU (fp64 on device) — CPU generation at 512 stations takes
    # minutes and starves the bench setup
    nb_rows = nb * args.tilesz * P
    gen_dev = device if (device != 'cpu' and nb_rows >= 2**21) else 'cpu'
    if gen_dev != 'cpu':
        pack.to(gen_dev)
    ms = msdata.SyntheticMS(
        N=args.stations, tilesz=args.tilesz * P, Ntime=args.tilesz * P,
        Nchan=args.chan, freq0=freq0, bandwidth=args.bandwidth,
        tdelta=10.0, pack=pack, seed=101 + rank,
        noise_sigma=5e-3, robust_noise=4.0,
        device=gen_dev, dtype=torch.float64)
    tile = ms.load_tile(0)
    if device != 'cpu':
        cdt = torch.complex64 if dtype == torch.float32 else torch.complex128
        tile.u = tile.u.to(device)
        tile.v = tile.v.to(device)
        tile.w = tile.w.to(device)
        tile.x = tile.x.to(device=device, dtype=cdt)
        tile.xo = tile.xo.to(device=device, dtype=cdt)
        tile.flags = tile.flags.to(device)
    bb = ms.bb_tensor(device=device)
    return pack, ms, tile, bb


def run_step(state, pack, tile, bb, opts, args, device, adm=None):
    """One full tile calibration (the benchmark unit of work).

    With consensus (adm): the sagecal-mpi per-tile ADMM loop — n_admm
    outer iterations, each one EM sweep + fused Z all-reduce over
    RCCL/xGMI; world=1 runs the identical loop without communication, so
    the scaling series is apples-to-apples (weak scaling over sub-bands).
    """
    from sagecal_amd.solvers import sage
    state.reset()
    state.nu.fill_(2.0)
    cohs = sage.precalc_coherencies(pack, tile)
    if device != 'cpu':
        cohs = cohs.to(torch.complex64)
    if adm is not None:
        adm.Y.zero_()
        adm.Z.zero_()
        adm.Yhat_prev = None
        adm.J_prev = None
        res0, res1 = adm.run(cohs, tile, bb, opts, n_admm=args.emiter)
    else:
        res0, res1 = sage.sagefit(state, cohs, tile, bb, opts)
    xres = sage.calculate_residuals_multifreq(state, pack, tile, bb)
    return res0, res1, xres


def run_bandpass(args, device, dtype, rank, world):
    """Config 4: stochastic-LBFGS bandpass calibration with mini-band
    consensus (+ federated averaging across ranks). One step = one epoch
    over time minibatches of all mini-bands."""
    import torch.distributed as tdist
    from sagecal_amd import sky, msdata
    from sagecal_amd.ops.reference import SourcePack
    from sagecal_amd.solvers.stochastic import MinibatchConsensusCalibration
    srcs, clist = sky.make_synthetic_sky(M=args.dirs,
                                         nsrc_per_cluster=args.srcs,
                                         seed=17)
    clusters = sky.build_clusters(srcs, clist, 0.0, np.pi / 4, args.freq0)
    pack = SourcePack(clusters)
    if device != 'cpu':
        pack.to(device)
    cdt = torch.complex64 if dtype == torch.float32 else torch.complex128
    ms = msdata.SyntheticMS(
        N=args.stations, tilesz=args.tilesz, Ntime=args.tilesz,
        Nchan=args.chan, freq0=args.freq0 + rank * args.bandwidth,
        bandwidth=args.bandwidth, tdelta=10.0, pack=pack, seed=303 + rank,
        noise_sigma=5e-3, device=device,
        dtype=torch.float32 if device != 'cpu' else torch.float64)
    tile = ms.load_tile(0)
    bb = ms.bb_tensor()
    cal = MinibatchConsensusCalibration(
        pack, args.stations, ms.freqs, nsolbw=args.nsolbw, Npoly=2,
        rho=1.0, device=device, dtype=cdt,
        fed_alpha=0.1 if world > 1 else 0.0, world=world, rank=rank,
        multifreq=not args.band_average)
    def step():
        cal.epoch(tile, bb, nmb=2, lbfgs_iters=6, robust_nu=10.0)
    return step, tile


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--gpus', type=int, default=1)
    ap.add_argument('--steps', type=int, default=4)
    ap.add_argument('--warmup', type=int, default=1)
    ap.add_argument('--stations', type=int, default=64)
    ap.add_argument('--dirs', type=int, default=10)
    ap.add_argument('--srcs', type=int, default=5)
    ap.add_argument('--tilesz', type=int, default=60)
    ap.add_argument('--intervals', type=int, default=16,
                    help='solution intervals batched per step (hybrid '
                         'time-chunks; each interval is tilesz slots)')
    ap.add_argument('--chan', type=int, default=8)
    ap.add_argument('--freq0', type=float, default=150e6)
    ap.add_argument('--bandwidth', type=float, default=180e3)
    ap.add_argument('--emiter', type=int, default=2)
    ap.add_argument('--maxiter', type=int, default=6)
    ap.add_argument('--robust-outer', type=int, default=1)
    ap.add_argument('--joint', type=int, default=0)
    ap.add_argument('--lbfgs-polish', type=int, default=10,
                    help='final joint LBFGS iterations per sagefit '
                         '(lmfit.c:1019 polish; 0 = off)')
    ap.add_argument('--em-group', type=int, default=10)
    ap.add_argument('--npoly', type=int, default=2)
    ap.add_argument('--solver', choices=['lm', 'rtr'], default='lm',
                    help='rtr: Riemannian trust-region (SKA config 5)')
    ap.add_argument('--mode', choices=['sage', 'bandpass'], default='sage',
                    help='bandpass: stochastic-LBFGS 256-chan consensus '
                         '(BASELINE config 4)')
    ap.add_argument('--nsolbw', type=int, default=8)
    ap.add_argument('--band-average', action='store_true',
                    help='bandpass mode: fit each mini-band to its '
                         'channel AVERAGE instead of the reference-'
                         'faithful per-channel gradient '
                         '(robust_batchmode_lbfgs.c:942 loops channels)')
    ap.add_argument('--shapelet-dirs', type=int, default=0,
                    help='make this many clusters shapelet (config 5)')
    ap.add_argument('--admm-rho', type=float, default=5.0)
    ap.add_argument('--no-consensus', action='store_true')
    ap.add_argument('--cpu', action='store_true')
    args = ap.parse_args()

    rank = int(os.environ.get('RANK', '0'))
    world = int(os.environ.get('WORLD_SIZE', '1'))
    local_rank = int(os.environ.get('LOCAL_RANK', str(rank)))
    dist = world > 1
    if dist:
        # N concurrent ranks on one host each generating synthetic data
        # on CPU: divide the cores instead of oversubscribing N-fold
        torch.set_num_threads(max(1, (os.cpu_count() or 8) // world))
    if dist:
        backend = os.environ.get('SAGECAL_BENCH_BACKEND')
        if backend is None:
            backend = 'nccl' if torch.cuda.is_available() else 'gloo'
        torch.distributed.init_process_group(backend=backend)

    use_gpu = torch.cuda.is_available() and not args.cpu
    if use_gpu:
        ngpu = torch.cuda.device_count()
        torch.cuda.set_device(local_rank % ngpu)
        device = f'cuda:{local_rank % ngpu}'
        dtype = torch.float32
    else:
        device = 'cpu'
        dtype = torch.float64

    from sagecal_amd.solvers import sage
    from sagecal_amd.constants import SM_RLM_RLBFGS, SM_RTR_OSRLM_RLBFGS

    if args.mode == 'bandpass':
        stepfn, tile = run_bandpass(args, device, dtype, rank, world)

        def sync():
            if use_gpu:
                torch.cuda.synchronize()
            if dist:
                torch.distributed.barrier()
        for _ in range(args.warmup):
            stepfn()
        sync()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            stepfn()
        sync()
        elapsed = time.perf_counter() - t0
        if dist:
            te = torch.tensor([elapsed],
                              device=device if use_gpu else 'cpu')
            torch.distributed.all_reduce(
                te, op=torch.distributed.ReduceOp.MAX)
            elapsed = float(te)
        Nbase = args.stations * (args.stations - 1) // 2
        vis_per_step = Nbase * args.tilesz * args.chan
        if rank == 0:
            print(json.dumps({
                'metric': 'visibilities/sec calibrated',
                'value': vis_per_step * args.steps * world / elapsed,
                'unit': 'vis/s', 'n_gpus': world if use_gpu else 0,
                'steps': args.steps, 'warmup': args.warmup,
                'ms_per_step': elapsed / args.steps * 1e3,
                'higher_is_better': True, 'scaling': 'weak',
                'vs_baseline': None,
                'dtype': 'fp32' if use_gpu else 'fp64',
                'data': 'synthetic',
                'config': {'model': f'{args.stations}-station stochastic-'
                           f'LBFGS bandpass, {args.chan} chan, '
                           f'{args.nsolbw} mini-bands',
                           'gradient': ('band-average'
                                        if args.band_average
                                        else 'per-channel'),
                           'global_batch': vis_per_step,
                           'seq_len': args.tilesz,
                           'parallelism': f'federated dp{world}'}}))
        if dist:
            torch.distributed.destroy_process_group()
        return

    pack, ms, tile, bb = build_problem(args, device, dtype, rank, world)
    cdtype = torch.complex64 if dtype == torch.float32 else torch.complex128
    state = sage.CalState(pack, args.stations, device=device, dtype=cdtype)
    adm = None
    if not args.no_consensus:
        from sagecal_amd.consensus.admm import ConsensusADMM
        freqs_all = [args.freq0 + r * args.bandwidth for r in range(world)]
        adm = ConsensusADMM(state, freqs_all, args.freq0, rank, world,
                            Npoly=min(args.npoly, world),
                            rho=torch.full((pack.M,), args.admm_rho))
    opts = sage.SageSolveOptions(
        max_emiter=1 if adm is not None else args.emiter,
        max_iter=args.maxiter,
        solver_mode=(SM_RTR_OSRLM_RLBFGS if args.solver == 'rtr'
                     else SM_RLM_RLBFGS),
        robust_outer=args.robust_outer, em_group=args.em_group,
        joint_iters=args.joint, lbfgs_iters=args.lbfgs_polish)

    def sync():
        if use_gpu:
            torch.cuda.synchronize()
        if dist:
            torch.distributed.barrier()

    # warmup
    for _ in range(args.warmup):
        run_step(state, pack, tile, bb, opts, args, device, adm)
    sync()
    t0 = time.perf_counter()
    res = None
    for _ in range(args.steps):
        res = run_step(state, pack, tile, bb, opts, args, device, adm)
    sync()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    if dist:
        te = torch.tensor([elapsed], device=device if use_gpu else 'cpu')
        torch.distributed.all_reduce(te, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(te)

    Nbase = args.stations * (args.stations - 1) // 2
    vis_per_step = Nbase * args.tilesz * args.intervals * args.chan
    ms_per_step = elapsed / args.steps * 1e3
    value = vis_per_step * args.steps * world / elapsed

    if rank == 0:
        out = {
            'metric': 'visibilities/sec calibrated',
            'value': value,
            'unit': 'vis/s',
            'n_gpus': world if use_gpu else 0,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': ms_per_step,
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': None,
            'dtype': 'fp32' if dtype == torch.float32 else 'fp64',
            'data': 'synthetic',
            'config': {
                'model': f'{args.stations}-station x {args.dirs}-direction '
                         f'robust LM (SAGE)',
                'stations': args.stations, 'directions': args.dirs,
                'tilesz': args.tilesz, 'channels': args.chan,
                'intervals_per_step': args.intervals,
                'global_batch': vis_per_step,
                'seq_len': args.tilesz,
                'parallelism': f'consensus-admm freq-band dp{world}',
                'res0': res[0] if res else None,
                'res1': res[1] if res else None,
            },
        }
        print(json.dumps(out))
    if dist:
        torch.distributed.destroy_process_group()


if __name__ == '__main__':
    main()
