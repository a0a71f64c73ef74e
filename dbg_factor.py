import torch, numpy as np
import sagecal_amd.ops.hip.dirac_hip as ext
dev='cuda:0'
rng = np.random.default_rng(2)
for n in (32, 64):
    Araw = torch.tensor(rng.standard_normal((1,n,n)), dtype=torch.float32, device=dev)
    A = ((Araw @ Araw.transpose(-1,-2))/n + 0.5*torch.eye(n,device=dev)).contiguous()
    b = torch.ones(1,n, device=dev)
    mu = torch.full((1,), 0.25, device=dev)
    sc = torch.zeros(1, 2*n*n, device=dev)
    dp, info = ext.chol_solve(A, b, mu, sc, 3)
    torch.cuda.synchronize()
    L = sc[0,:n*n].reshape(n,n).tril()
    Lref = torch.linalg.cholesky((A[0] + 0.25*torch.eye(n,device=dev)).double()).float()
    print(f"n={n} L err:", float((L-Lref).abs().max()), "info", int(info[0]))
    ref = torch.linalg.solve((A[0]+0.25*torch.eye(n,device=dev)).double(), b[0].double())
    print(f"n={n} dp err:", float((dp[0].double()-ref).abs().max()))
    LT = sc[0,n*n:].reshape(n,n).triu()
    print(f"n={n} LT err:", float((LT-Lref.T).abs().max()))
