import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, numpy as np
from sagecal_amd.ops.hip_host import chol_solve_damped
dev='cuda:0'
rng = np.random.default_rng(1)
n, batch = 512, 2
Araw = torch.tensor(rng.standard_normal((batch, n, n)), dtype=torch.float32, device=dev)
A = (Araw @ Araw.transpose(-1,-2))/n + 0.5*torch.eye(n, device=dev).unsqueeze(0)
b = torch.tensor(rng.standard_normal((batch, n)), dtype=torch.float32, device=dev)
mu = torch.full((batch,), 0.1, device=dev)
for _ in range(3): chol_solve_damped(A, b, mu)
torch.cuda.synchronize()
t0=time.perf_counter()
for _ in range(20): chol_solve_damped(A, b, mu)
torch.cuda.synchronize()
print("per call ms:", (time.perf_counter()-t0)/20*1e3)
