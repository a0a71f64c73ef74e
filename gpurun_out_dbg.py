import torch, numpy as np
import sagecal_amd.ops.hip.dirac_hip as ext
dev='cuda:0'
N=3; Nbase=3; T=2; nseg=1
pairs = torch.tensor([[0,1],[0,2],[1,2]], dtype=torch.int32, device=dev)
B = nseg*T*Nbase
rng = np.random.default_rng(0)
coh = torch.tensor(rng.standard_normal((B,4))+1j*rng.standard_normal((B,4)), dtype=torch.complex64, device=dev)
J = torch.tensor(rng.standard_normal((1,N,4))+1j*rng.standard_normal((1,N,4)), dtype=torch.complex64, device=dev)
ct = torch.zeros(nseg*T, dtype=torch.int32, device=dev)
out = ext.apply_jones(None, coh.reshape(-1,4), J.reshape(-1,4), pairs.reshape(-1), ct, Nbase, T, N, nseg, 1, 0)
torch.cuda.synchronize()
# reference
def m22(t): return t.reshape(-1,2,2)
Jm = J.reshape(N,2,2)
Cm = m22(coh)
pr = pairs.long()
bb = pr.repeat(T,1)
V = Jm[bb[:,0]] @ Cm @ Jm[bb[:,1]].conj().transpose(-1,-2)
print("kernel:", out.cpu().numpy()[0])
print("ref   :", V.reshape(-1,4).cpu().numpy()[0])
err = (out.reshape(-1,2,2)-V).abs().max()
print("max err", float(err))
# with x / sub=1
x = torch.tensor(rng.standard_normal((B,4))+1j*rng.standard_normal((B,4)), dtype=torch.complex64, device=dev)
out2 = ext.apply_jones(x, coh.reshape(-1,4), J.reshape(-1,4), pairs.reshape(-1), ct, Nbase, T, N, nseg, 1, 1)
err2 = (out2.reshape(-1,2,2)-(x.reshape(-1,2,2)-V)).abs().max()
print("sub err", float(err2))
# model cost
c = ext.model_cost(x, coh.reshape(-1,4), J.reshape(-1,4), pairs.reshape(-1), ct, None, Nbase, T, N, nseg, 1)
cr = ((x.reshape(-1,2,2)-V).abs()**2).sum()
print("cost", float(c.sum()), float(cr))
