"""Phase-level timing of one bench step on GPU (predict / EM sweeps /
residual), to direct optimization. Run on a GPU box:
    python tools/profile_step.py [--tilesz 60] ...
"""
import argparse
import sys
import time
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def t_sync():
    torch.cuda.synchronize()
    return time.perf_counter()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--stations', type=int, default=64)
    ap.add_argument('--dirs', type=int, default=10)
    ap.add_argument('--srcs', type=int, default=5)
    ap.add_argument('--tilesz', type=int, default=60)
    ap.add_argument('--intervals', type=int, default=16)
    ap.add_argument('--chan', type=int, default=8)
    ap.add_argument('--emiter', type=int, default=2)
    ap.add_argument('--maxiter', type=int, default=8)
    ap.add_argument('--em-group', type=int, default=10)
    ap.add_argument('--reps', type=int, default=2)
    args = ap.parse_args()
    sys.argv = [sys.argv[0]]

    import bench
    from sagecal_amd.solvers import sage
    from sagecal_amd.constants import SM_RLM_RLBFGS

    class A:
        pass
    a = A()
    a.__dict__.update(stations=args.stations, dirs=args.dirs, srcs=args.srcs,
                      tilesz=args.tilesz, chan=args.chan, freq0=150e6,
                      bandwidth=180e3, intervals=args.intervals)
    dev = 'cuda:0'
    pack, ms, tile, bb = bench.build_problem(a, dev, torch.float32)
    state = sage.CalState(pack, args.stations, device=dev,
                          dtype=torch.complex64)
    opts = sage.SageSolveOptions(max_emiter=args.emiter,
                                 max_iter=args.maxiter,
                                 solver_mode=SM_RLM_RLBFGS,
                                 robust_outer=1, em_group=args.em_group)
    # warm
    cohs = sage.precalc_coherencies(pack, tile).to(torch.complex64)
    sage.sagefit(state, cohs, tile, bb, opts)
    sage.calculate_residuals_multifreq(state, pack, tile, bb)

    for rep in range(args.reps):
        state.reset()
        t0 = t_sync()
        cohs = sage.precalc_coherencies(pack, tile).to(torch.complex64)
        t1 = t_sync()
        res0, res1 = sage.sagefit(state, cohs, tile, bb, opts)
        t2 = t_sync()
        xres = sage.calculate_residuals_multifreq(state, pack, tile, bb)
        t3 = t_sync()
        print(f"rep{rep}: predict {1e3*(t1-t0):8.2f} ms | sage "
              f"{1e3*(t2-t1):8.2f} ms | residual {1e3*(t3-t2):8.2f} ms | "
              f"res {res0:.3f}->{res1:.4f}")


if __name__ == '__main__':
    main()
