"""Micro-benchmark the pieces of one batched-LM iteration on GPU."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch


def timeit(fn, n=50):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    dev = 'cuda:0'
    import bench
    from sagecal_amd.ops.hip_host import BaselineLayout, jtj_jtr, \
        model_cost_per_chunk
    from sagecal_amd.solvers import lm as lm_mod

    class A: pass
    a = A(); a.__dict__.update(stations=64, dirs=10, srcs=5, tilesz=60,
                               chan=8, freq0=150e6, bandwidth=180e3)
    pack, ms, tile, bb = bench.build_problem(a, dev, torch.float32)
    N = 64; T = 60; Nbase = ms.Nbase
    from sagecal_amd.solvers import sage
    state = sage.CalState(pack, N, device=dev, dtype=torch.complex64)
    cohs = sage.precalc_coherencies(pack, tile).to(torch.complex64)
    # group of 2 clusters problem
    nseg = 2
    lay = BaselineLayout(bb, Nbase, T, nseg, N, dev)
    x2 = torch.cat([tile.x, tile.x])
    c2 = torch.cat([cohs[0], cohs[1]])
    J = state.J[:2].clone()
    bb2 = torch.cat([bb, bb])
    rows = torch.cat([torch.zeros(tile.x.shape[0], dtype=torch.long, device=dev),
                      torch.ones(tile.x.shape[0], dtype=torch.long, device=dev)])

    print("B rows:", x2.shape[0])
    t = timeit(lambda: jtj_jtr(x2, c2, J, bb2, N, None, rows, 2, lay))
    print(f"jtj_jtr (accum+expand): {t:.3f} ms")
    JtJ, Jtr, cost = jtj_jtr(x2, c2, J, bb2, N, None, rows, 2, lay)
    eye = torch.eye(512, device=dev).unsqueeze(0)
    mu = torch.ones(2, device=dev)
    t = timeit(lambda: JtJ + mu[:, None, None] * eye)
    print(f"A = JtJ + mu*eye:       {t:.3f} ms")
    A_ = JtJ + mu[:, None, None] * eye
    def chol():
        L, info = torch.linalg.cholesky_ex(A_)
        return torch.cholesky_solve(Jtr.unsqueeze(-1), L)
    t = timeit(chol)
    print(f"cholesky_ex + solve:    {t:.3f} ms")
    t = timeit(lambda: model_cost_per_chunk(x2, c2, J, bb2, N, None, rows, 2, lay))
    print(f"model_cost kernel:      {t:.3f} ms")
    # elementwise block
    dp = torch.zeros(2, 512, device=dev)
    costv = torch.ones(2, device=dev)
    nu = torch.full((2,), 2.0, device=dev)
    active = torch.ones(2, dtype=torch.bool, device=dev)
    def elem():
        dpc = lm_mod._vec_to_jones(dp, 2, N)
        Jn = J + dpc
        denom = (dp * (mu[:, None] * dp + Jtr)).sum(dim=-1).clamp_min(1e-30)
        rho = (costv - costv) / denom
        acc = (rho > 0) & active
        stepn = dp.norm(dim=-1)
        pn = lm_mod._jones_norm(J, 2)
        Jx = torch.where(acc[:, None, None, None], Jn, J)
        return Jx
    t = timeit(elem)
    print(f"elementwise block:      {t:.3f} ms")
    # full lm_solve iteration cost
    prob = lm_mod.LMProblem(x2, c2, bb2, N, 2, rows, layout=lay)
    t = timeit(lambda: lm_mod.lm_solve(prob, J, maxiter=1), n=10)
    print(f"lm_solve(maxiter=1):    {t:.3f} ms")
    t = timeit(lambda: lm_mod.lm_solve(prob, J, maxiter=8), n=5)
    print(f"lm_solve(maxiter=8):    {t:.3f} ms")
    from sagecal_amd.ops.hip_host import chol_solve_damped
    mu2 = torch.ones(2, device=dev) * 0.1
    t = timeit(lambda: chol_solve_damped(JtJ, Jtr, mu2))
    print(f"custom chol_solve:      {t:.3f} ms")
    JtJ10 = JtJ.repeat(5, 1, 1); Jtr10 = Jtr.repeat(5, 1)
    mu10 = mu2.repeat(5)
    t = timeit(lambda: chol_solve_damped(JtJ10, Jtr10, mu10))
    print(f"custom chol batch10:    {t:.3f} ms")


if __name__ == '__main__':
    main()
