"""Influence-function diagnostics of the calibration.

Re-implements the role of /root/reference/src/lib/Radio/diagnostics.c +
influence_function.cu (calculate_diagnostics_gpu, diagnostics.c:1019):
quantify how strongly each visibility influences the solutions — the
statistical leverage of the calibration. The reference builds the 4N x 4N
data-influence Hessian (kernel_hessian, influence_function.cu:83) and
derivative blocks dJ/ddata (kernel_d_solutions :287) and replaces output
data with the influence eigen-structure (-i 1).

Here (same math, our machinery): with the Gauss-Newton normal matrix
H = JtJ from the fused assembly kernels and per-baseline design blocks
D_b (closed 2x2-complex forms), the solution sensitivity to visibility b
is dtheta/dx_b = H^-1 D_b^T, and the self-influence (leverage) of b is
  lev_b = tr(D_b H^-1 D_b^T)   in [0, 8]
(the diagonal block of the hat matrix). Residual influence uses
I - D H^-1 D^T.
"""
import torch

from ..ops import dispatch as ops
from ..ops import reference as R


def leverage(x, coh, J, bb, N, weights=None, chunk_rows=None, nchunk=1,
             layout=None, mu=1e-6):
    """Per-baseline leverage scores [B] for one cluster's solve."""
    JtJ, Jtr, _ = ops.jtj_jtr(x, coh, J, bb, N, weights, chunk_rows,
                              nchunk, layout)
    dev = x.device
    rdt = x.real.dtype
    n = 8 * N
    eye = torch.eye(n, dtype=rdt, device=dev).unsqueeze(0)
    Hinv = torch.linalg.inv(JtJ.to(torch.float64)
                            + mu * eye.to(torch.float64))
    if chunk_rows is None:
        chunk_rows = torch.zeros(x.shape[0], dtype=torch.long, device=dev)
    # per-baseline design blocks (as in reference.jtj_jtr derivation)
    Jp = J[chunk_rows, bb[:, 0]]
    Jq = J[chunk_rows, bb[:, 1]]
    G1 = coh @ Jq.conj().transpose(-1, -2)
    K = Jp @ coh
    D1 = _lin_blocks(G1)                      # [B, 8, 8] (station p cols)
    D2 = _anti_blocks(K)                      # [B, 8, 8] (station q cols)
    B_ = x.shape[0]
    lev = torch.zeros(B_, dtype=torch.float64, device=dev)
    p_idx = (chunk_rows * N + bb[:, 0])
    q_idx = (chunk_rows * N + bb[:, 1])
    Hc = Hinv[chunk_rows]                     # [B, 8N, 8N]
    # gather relevant 8x8 blocks of H^-1 (vectorized fancy indexing)
    ar8 = torch.arange(8, device=dev)
    bidx = torch.arange(B_, device=dev)[:, None, None]

    def hblk(i_idx, j_idx):
        ri = ((i_idx % N) * 8)[:, None] + ar8          # [B, 8]
        cj = ((j_idx % N) * 8)[:, None] + ar8
        return Hc[bidx, ri[:, :, None], cj[:, None, :]]
    Hpp = hblk(p_idx, p_idx)
    Hqq = hblk(q_idx, q_idx)
    Hpq = hblk(p_idx, q_idx)
    D1d = D1.double(); D2d = D2.double()
    T1 = D1d @ Hpp @ D1d.transpose(-1, -2)
    T2 = D2d @ Hqq @ D2d.transpose(-1, -2)
    T3 = D1d @ Hpq @ D2d.transpose(-1, -2)
    lev = (T1 + T2 + 2 * T3).diagonal(dim1=-2, dim2=-1).sum(-1)
    if weights is not None:
        lev = lev * weights.double()
    return lev


def _lin_blocks(G1):
    """Realified per-baseline design block of the complex-linear map
    dV = dJ1 G1 (I2 (x) realify(G1^T)): [B, 8, 8]."""
    B = G1.shape[0]
    M4 = R.realify(G1.transpose(-1, -2))
    out = torch.zeros(B, 8, 8, dtype=G1.real.dtype, device=G1.device)
    out[:, 0:4, 0:4] = M4
    out[:, 4:8, 4:8] = M4
    return out


def _anti_blocks(K):
    """Realified design block of the anti-linear map dV = K conj(dJ2)^T
    (row-permuted I2 (x) antirealify(K)): [B, 8, 8]."""
    B = K.shape[0]
    A4 = R.antirealify(K)
    out = torch.zeros(B, 8, 8, dtype=K.real.dtype, device=K.device)
    # V rows group as (V00,V10 | V01,V11) for J2 columns (00,01 | 10,11):
    # map back to vecR(V) row order [00,01,10,11]
    perm = torch.tensor([0, 1, 4, 5, 2, 3, 6, 7])
    tmp = torch.zeros_like(out)
    tmp[:, 0:4, 0:4] = A4
    tmp[:, 4:8, 4:8] = A4
    out = tmp[:, perm, :]
    return out


def influence_map(state, cohs, tile, bb, cluster=0, layout=None):
    """Replace-output-data diagnostics (-i 1): per-visibility leverage of
    the chosen cluster's solve, shaped like the data [B]."""
    ci = cluster
    T, Nbase = tile.tilesz, tile.Nbase
    Bn = tile.x.shape[0]
    rows = R.chunk_rows_for(ci, state.nchunks, T, Nbase, Bn, tile.x.device)
    return leverage(tile.x, cohs[ci], state.cluster_J(ci), bb, state.N,
                    chunk_rows=rows, nchunk=state.nchunks[ci],
                    layout=layout)
