"""Stage-by-stage timing of the SKA config-5 bench path (512 stn, RTR)
with incremental flushes to gpurun_out/ska_stages.log so a timeout still
shows where the time went. Run via gpurun."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import torch

LOG = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), 'gpurun_out', 'ska_stages.log')
os.makedirs(os.path.dirname(LOG), exist_ok=True)
_t0 = time.perf_counter()


def log(msg):
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    with open(LOG, 'a') as f:
        f.write(f"[{time.perf_counter() - _t0:8.1f}s] {msg}\n")


def main():
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument('--stations', type=int, default=512)
    ap.add_argument('--dirs', type=int, default=20)
    ap.add_argument('--graph', type=int, default=1)
    ap.add_argument('--emiter', type=int, default=1)
    ap.add_argument('--maxiter', type=int, default=6)
    args = ap.parse_args()
    if not args.graph:
        os.environ['SAGECAL_RTR_GRAPH'] = '0'
    os.environ['SAGECAL_TRACE'] = '1'
    import bench
    from sagecal_amd.solvers import sage
    from sagecal_amd.constants import SM_RTR_OSRLM_RLBFGS

    class A:
        pass
    a = A()
    a.__dict__.update(stations=args.stations, dirs=args.dirs, srcs=5,
                      tilesz=60, chan=8, freq0=150e6, bandwidth=180e3,
                      intervals=1, shapelet_dirs=2)
    dev = 'cuda:0' if torch.cuda.is_available() else 'cpu'
    log(f"start dev={dev}")
    pack, ms, tile, bb = bench.build_problem(
        a, dev, torch.float32 if dev != 'cpu' else torch.float64)
    log(f"build_problem done rows={tile.x.shape[0]}")
    state = sage.CalState(pack, args.stations, device=dev,
                          dtype=torch.complex64 if dev != 'cpu'
                          else torch.complex128)
    cohs = sage.precalc_coherencies(pack, tile)
    if dev != 'cpu':
        cohs = cohs.to(torch.complex64)
    log(f"precalc done {tuple(cohs.shape)}")
    opts = sage.SageSolveOptions(
        max_emiter=args.emiter, max_iter=args.maxiter,
        solver_mode=SM_RTR_OSRLM_RLBFGS, robust_outer=1, em_group=10)
    r0, r1 = sage.sagefit(state, cohs, tile, bb, opts)
    log(f"sagefit emiter={args.emiter} done res {r0:.3f}->{r1:.4f}")
    t = time.perf_counter()
    r0, r1 = sage.sagefit(state, cohs, tile, bb, opts)
    log(f"sagefit again (warm) {time.perf_counter() - t:.1f}s "
        f"res {r0:.3f}->{r1:.4f}")


if __name__ == '__main__':
    main()
