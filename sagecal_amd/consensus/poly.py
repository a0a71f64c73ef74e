"""Frequency-polynomial consensus machinery.

Re-implements /root/reference/src/lib/Dirac/consensus_poly.c:
  - setup_polynomials (consensus_poly.c:39-140): basis types 0 (monomial in
    (f-f0)/f0), 1 (row-normalized monomial), 2 (Bernstein over [fmin,fmax]),
    3 (mixed (f-f0)/f0 and (f0/f-1) powers);
  - find_prod_inverse_full (consensus_poly.c:465): per-cluster
    inv(sum_f rho_cf B_f B_f^T), federated variant adds alpha*I;
  - update_global_z: Z_c = Bii_c (sum_f B_f x (Y_f + rho J_f));
  - update_rho_bb (consensus_poly.c:860-1011): Barzilai-Borwein adaptive
    penalty from the correlation of dual/primal deltas.
All dense ops are torch (batched over clusters) so they run on CPU or GPU.
"""
import math
import torch


def setup_polynomials(freqs, freq0, Npoly, ptype=0):
    """Basis matrix B [Nf, Npoly] (row f = basis evaluated at freqs[f])."""
    freqs = torch.as_tensor(freqs, dtype=torch.float64)
    Nf = freqs.shape[0]
    B = torch.zeros(Nf, Npoly, dtype=torch.float64)
    if ptype in (0, 1):
        frat = (freqs - freq0) / freq0
        B[:, 0] = 1.0
        for m in range(1, Npoly):
            B[:, m] = B[:, m - 1] * frat
        if ptype == 1:
            nrm = B.norm(dim=0).clamp_min(1e-300)
            B = B / nrm
    elif ptype == 2:
        fmax, fmin = freqs.max(), freqs.min()
        x = (freqs - fmin) / (fmax - fmin) if fmax > fmin \
            else torch.zeros_like(freqs)
        n = Npoly - 1
        for j in range(Npoly):
            binom = math.factorial(n) / (math.factorial(n - j)
                                         * math.factorial(j))
            B[:, j] = binom * x ** j * (1 - x) ** (n - j)
    elif ptype == 3:
        B[:, 0] = 1.0
        f1 = (freqs - freq0) / freq0
        f2 = freq0 / freqs - 1.0
        for m in range(1, Npoly):
            p = (m + 1) // 2
            B[:, m] = f1 ** p if m % 2 == 1 else f2 ** p
    else:
        raise ValueError(f"unknown poly type {ptype}")
    return B


def find_prod_inverse(B, rho, alpha=0.0):
    """Bii [M, Npoly, Npoly] = inv(sum_f rho[c,f] B_f B_f^T (+ alpha I)).

    B: [Nf, Npoly]; rho: [M, Nf] (per cluster per band penalty).
    Federated variant (find_prod_inverse_full_fed) via alpha > 0."""
    BBt = torch.einsum('fp,fq->fpq', B, B)          # [Nf, P, P]
    A = torch.einsum('mf,fpq->mpq', rho.to(B.dtype), BBt)
    if alpha > 0:
        A = A + alpha * torch.eye(B.shape[1], dtype=B.dtype)
    # pseudo-inverse for robustness at Npoly > Nf
    return torch.linalg.pinv(A)


def update_global_z(z_accum, Bii):
    """Z [M, Npoly, ...] from the reduced accumulator
    z_accum [M, Npoly, ...] = sum_f B_f (Y_f + rho_f J_f):
    Z_c = Bii_c z_c applied along the Npoly axis."""
    M, P = z_accum.shape[0], z_accum.shape[1]
    flat = z_accum.reshape(M, P, -1)
    if flat.is_complex():
        Bc = Bii.to(flat.real.dtype).to(flat.device)
        out = torch.complex(
            torch.bmm(Bc, flat.real), torch.bmm(Bc, flat.imag))
    else:
        out = torch.bmm(Bii.to(flat.dtype).to(flat.device), flat)
    return out.reshape(z_accum.shape)


def eval_poly_jones(Z, Bf):
    """B_f Z: Jones at one frequency from polynomial coefficients.
    Z: [M, Npoly, N, 2, 2] complex; Bf: [Npoly] reals."""
    w = torch.as_tensor(Bf, dtype=Z.real.dtype, device=Z.device)
    return torch.einsum('p,mpnij->mnij', w.to(Z.dtype), Z)


def update_rho_bb_ip(rho, rho_upper, ip11, ip12, ip22, eps=1e-12,
                     alphacorrmin=0.2):
    """Barzilai-Borwein penalty update per cluster from the inner
    products of the dual delta dY = Yhat - Yhat_prev and primal delta
    dJ = J - J_prev (rho_bb_threadfn, consensus_poly.c:860-926,
    oracle-verified): alphaMG if 2 alphaMG > alphaSD else
    alphaSD - alphaMG/2; rho := alphahat when the deltas correlate
    (corr > 0.2) and 1e-3 < alphahat < rho_upper."""
    rho_upper = torch.as_tensor(rho_upper, dtype=rho.dtype).expand_as(rho)
    corr = ip12 / torch.sqrt((ip11 * ip22).clamp_min(eps))
    alphaSD = ip11 / ip12.clamp_min(eps)
    alphaMG = ip12 / ip22.clamp_min(eps)
    alphahat = torch.where(2.0 * alphaMG > alphaSD, alphaMG,
                           alphaSD - 0.5 * alphaMG)
    ok = (ip12 > eps) & (ip11 > eps) & (ip22 > eps) \
        & (corr > alphacorrmin) & (alphahat > 1e-3) \
        & (alphahat < rho_upper) & torch.isfinite(alphahat)
    return torch.where(ok, alphahat, rho)


def update_rho_bb(rho, rho_upper, dY, dJ, eps=1e-12, alphacorrmin=0.2):
    """update_rho_bb_ip on flattened per-cluster deltas [M, K]."""
    ip12 = (dY * dJ).sum(dim=1)
    ip11 = (dY * dY).sum(dim=1)
    ip22 = (dJ * dJ).sum(dim=1)
    return update_rho_bb_ip(rho, rho_upper, ip11, ip12, ip22, eps,
                            alphacorrmin)
