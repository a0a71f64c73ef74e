"""Correctness + latency of the multi-WG Cholesky path vs the fused
single-kernel path and torch.cholesky_solve (fp64 reference)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, numpy as np
import sagecal_amd.ops.hip.dirac_hip as ext
dev = 'cuda:0'
rng = np.random.default_rng(1)
for n, batch in ((512, 5), (512, 1), (256, 10), (64, 8), (512, 16)):
    Araw = torch.tensor(rng.standard_normal((batch, n, n)), dtype=torch.float32, device=dev)
    A = ((Araw @ Araw.transpose(-1, -2)) / n
         + 0.5 * torch.eye(n, device=dev).unsqueeze(0)).contiguous()
    b = torch.tensor(rng.standard_normal((batch, n)), dtype=torch.float32,
                     device=dev).contiguous()
    mu = torch.full((batch,), 0.1, device=dev)
    sc = torch.empty(batch, 2 * n * n, dtype=torch.float32, device=dev)
    # fp64 reference
    Ad = A.double() + 0.1 * torch.eye(n, device=dev, dtype=torch.float64)
    xref = torch.cholesky_solve(b.double().unsqueeze(-1),
                                torch.linalg.cholesky(Ad)).squeeze(-1)
    x1, i1 = ext.chol_solve(A, b, mu, sc, 3)
    x2, i2 = ext.chol_solve_mw(A, b, mu, sc, 4)
    e1 = (x1.double() - xref).abs().max() / xref.abs().max()
    e2 = (x2.double() - xref).abs().max() / xref.abs().max()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(30): ext.chol_solve(A, b, mu, sc, 3)
    torch.cuda.synchronize(); t1 = (time.perf_counter() - t0) / 30 * 1e3
    t0 = time.perf_counter()
    for _ in range(30): ext.chol_solve_mw(A, b, mu, sc, 4)
    torch.cuda.synchronize(); t2 = (time.perf_counter() - t0) / 30 * 1e3
    print(f"n={n:4d} B={batch:3d}: relerr single={e1:.2e} mw={e2:.2e}  "
          f"t single={t1:.3f}ms mw={t2:.3f}ms  info={int(i2.sum())}")
    assert e2 < 5e-4, "mw numerics off"
# stage ablation at the headline shape
n, batch = 512, 5
Araw = torch.tensor(rng.standard_normal((batch, n, n)), dtype=torch.float32, device=dev)
A = ((Araw @ Araw.transpose(-1, -2)) / n
     + 0.5 * torch.eye(n, device=dev).unsqueeze(0)).contiguous()
b = torch.tensor(rng.standard_normal((batch, n)), dtype=torch.float32, device=dev).contiguous()
mu = torch.full((batch,), 0.1, device=dev)
sc = torch.empty(batch, 2 * n * n, dtype=torch.float32, device=dev)
for st in (1, 2, 3, 4):
    ext.chol_solve_mw(A, b, mu, sc, st); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50): ext.chol_solve_mw(A, b, mu, sc, st)
    torch.cuda.synchronize()
    print(f"mw stage {st}: {(time.perf_counter()-t0)/50*1e3:.3f} ms")
print("OK")
