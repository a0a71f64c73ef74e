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


def gen_phases(stations=512, dirs=20):
    """Fine-grained timing of SyntheticMS generation on the current
    device — pinpoints the config-5 build_problem wall (ROUND3 P0)."""
    import numpy as np
    from sagecal_amd import sky, msdata
    from sagecal_amd.ops import reference as R
    from sagecal_amd.ops.reference import SourcePack
    dev = 'cuda:0' if torch.cuda.is_available() else 'cpu'
    srcs, clist = sky.make_synthetic_sky(M=dirs, nsrc_per_cluster=5,
                                         seed=17)
    clusters = sky.build_clusters(srcs, clist, 0.0, np.pi / 4, 150e6)
    pack = SourcePack(clusters)
    pack.to(dev)
    log(f"gen_phases start dev={dev} N={stations} M={dirs}")
    ms = msdata.SyntheticMS(N=stations, tilesz=60, Ntime=60, Nchan=8,
                            freq0=150e6, bandwidth=180e3, tdelta=10.0,
                            pack=None, seed=7, noise_sigma=0.0,
                            device=dev, dtype=torch.float64)
    ub, vb, wb = ms.uvw_for(0)
    log("uvw_for done")
    u = torch.tensor(ub, dtype=torch.float64, device=dev)
    v = torch.tensor(vb, dtype=torch.float64, device=dev)
    w = torch.tensor(wb, dtype=torch.float64, device=dev)
    bb = ms.bb_tensor()
    log(f"uvw->device + bb done rows={u.shape[0]}")
    coh = R.predict_coh(pack, u, v, w, 150e6, 150e6, 2e4, 10.0,
                        np.pi / 4)
    log(f"ONE channel predict_coh (torch fp64) done {tuple(coh.shape)}")
    Jt = ms.true_jones(dirs, 0).to(dev)
    J1 = Jt[:, bb[:, 0]]
    J2h = Jt[:, bb[:, 1]].conj().transpose(-1, -2)
    log("J gather done")
    xo = ((J1 @ coh.to(J1.dtype)) @ J2h).sum(dim=0)
    log("c128 bmm apply done")
    from sagecal_amd.ops import dispatch as disp
    if disp.have_ext() and dev != 'cpu':
        coh2 = disp.predict_coh(pack, u, v, w, 150e6, 150e6, 2e4, 10.0,
                                np.pi / 4)
        log("ONE channel predict via HIP kernel done")
        xo2 = ((J1.to(torch.complex64) @ coh2)
               @ J2h.to(torch.complex64)).sum(dim=0)
        log("c64 bmm apply done")
        err = float((xo2.to(torch.complex128) - xo).abs().max()
                    / xo.abs().max())
        log(f"kernel-vs-torch gen rel err {err:.2e}")
    import numpy as _np
    rng = _np.random.default_rng(1)
    nre = rng.standard_normal((8, u.shape[0], 2, 2, 2))
    log("numpy noise standard_normal done")
    noise = torch.tensor(nre[..., 0] + 1j * nre[..., 1], device=dev)
    log("noise to device done")


def main():
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument('--stations', type=int, default=512)
    ap.add_argument('--dirs', type=int, default=20)
    ap.add_argument('--graph', type=int, default=1)
    ap.add_argument('--emiter', type=int, default=1)
    ap.add_argument('--maxiter', type=int, default=6)
    ap.add_argument('--phase', choices=['all', 'gen'], default='all')
    args = ap.parse_args()
    if args.phase == 'gen':
        gen_phases(args.stations, args.dirs)
        return
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
