import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, numpy as np
import sagecal_amd.ops.hip.dirac_hip as ext
dev='cuda:0'
rng = np.random.default_rng(1)
n, batch = 512, 2
Araw = torch.tensor(rng.standard_normal((batch, n, n)), dtype=torch.float32, device=dev)
A = ((Araw @ Araw.transpose(-1,-2))/n + 0.5*torch.eye(n, device=dev).unsqueeze(0)).contiguous()
b = torch.tensor(rng.standard_normal((batch, n)), dtype=torch.float32, device=dev).contiguous()
mu = torch.full((batch,), 0.1, device=dev)
sc = torch.empty(batch, 2*n*n, dtype=torch.float32, device=dev)
def run(st):
    return ext.chol_solve(A, b, mu, sc, st)
for st in (0,1,2,3,4,5,6):
    run(st); torch.cuda.synchronize()
    t0=time.perf_counter()
    for _ in range(20): run(st)
    torch.cuda.synchronize()
    print(f"stage {st}: {(time.perf_counter()-t0)/20*1e3:.3f} ms")
# batch scaling
for bsz in (1,4,16):
    Ab = A[:1].repeat(bsz,1,1).contiguous(); bbv=b[:1].repeat(bsz,1).contiguous()
    mub = mu[:1].repeat(bsz); scb = torch.empty(bsz, 2*n*n, dtype=torch.float32, device=dev)
    ext.chol_solve(Ab,bbv,mub,scb,3); torch.cuda.synchronize()
    t0=time.perf_counter()
    for _ in range(20): ext.chol_solve(Ab,bbv,mub,scb,3)
    torch.cuda.synchronize()
    print(f"batch {bsz}: {(time.perf_counter()-t0)/20*1e3:.3f} ms")
