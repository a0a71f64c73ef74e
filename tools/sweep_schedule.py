"""Sweep solver schedules on the bench config: wall time vs residual
quality, to pick the production schedule (fixed quality target, maximum
throughput)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def main():
    dev = 'cuda:0'
    import bench
    from sagecal_amd.solvers import sage
    from sagecal_amd.constants import SM_RLM_RLBFGS

    class A: pass
    a = A(); a.__dict__.update(stations=64, dirs=10, srcs=5, tilesz=60,
                               chan=8, freq0=150e6, bandwidth=180e3,
                               intervals=16, shapelet_dirs=0)
    pack, ms, tile, bb = bench.build_problem(a, dev, torch.float32)

    configs = [
        # (emiter, maxiter, robust_outer, em_group, lbfgs_polish, label)
        (2, 6, 1, 10, 10, 'em2-g10-lb10 (prod)'),
        (2, 6, 1, 10, 6, 'em2-g10-lb6'),
        (1, 6, 1, 10, 14, 'em1-g10-lb14'),
        (2, 6, 1, 5, 10, 'em2-g5-lb10'),
        (3, 6, 1, 10, 10, 'em3-g10-lb10'),
        (2, 8, 1, 10, 10, 'em2-it8-lb10'),
    ]
    for emiter, maxiter, ro, eg, lb, label in configs:
        opts = sage.SageSolveOptions(
            max_emiter=emiter, max_iter=maxiter,
            solver_mode=SM_RLM_RLBFGS, robust_outer=ro, em_group=eg,
            lbfgs_iters=lb)
        state = sage.CalState(pack, 64, device=dev, dtype=torch.complex64)
        cohs = sage.precalc_coherencies(pack, tile).to(torch.complex64)
        sage.sagefit(state, cohs, tile, bb, opts)   # warm
        state.reset(); state.nu.fill_(2.0)
        torch.cuda.synchronize(); t0 = time.perf_counter()
        res0, res1 = sage.sagefit(state, cohs, tile, bb, opts)
        torch.cuda.synchronize(); t1 = time.perf_counter()
        print(f"{label:20s} em={emiter} it={maxiter} g={eg} lb={lb}: "
              f"{1e3*(t1-t0):7.1f} ms  res {res0:.3f}->{res1:.4f}")


if __name__ == '__main__':
    main()
