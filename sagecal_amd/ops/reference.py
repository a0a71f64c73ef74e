"""PyTorch reference implementations of every hot op.

These are the golden models the HIP kernels (sagecal_amd/ops/hip) are tested
against, and the CPU production path. Math mirrors
/root/reference/src/lib/Radio/predict.c (coherency predict, smearing,
extended-source envelopes) and the solver math of
/root/reference/src/lib/Dirac/{clmfit.c,robustlm.c,mderiv.cu} — re-derived
in closed 2x2-complex form (no dense Jacobian: JtJ/Jtr are assembled from
per-baseline 2x2 Gram blocks; see jtj_jtr()).

Conventions:
  - visibilities/coherencies: complex tensors [..., 2, 2]
  - Jones parameter real-vector order per station: the interleaved real view
    of the row-major 2x2 complex matrix: [J00r,J00i,J01r,J01i,J10r,J10i,
    J11r,J11i] (solution-file output permutes to the reference's column-major
    order; see solutions.py).
  - baselines: time-major, row = t*Nbase + b, station pairs p<q.
  - u,v,w in seconds (already divided by c), as in Data::loadData.
"""
import math
import torch

TWO_PI = 2.0 * math.pi
OMEGA_EARTH = 7.2921150e-5
CLM_EPSILON = 1e-12


# ---------------------------------------------------------------------------
# Source packs: struct-of-arrays tensors for all clusters, concatenated.
# ---------------------------------------------------------------------------

class SourcePack:
    """All clusters' sources concatenated into contiguous tensors.

    cluster_off[c] .. cluster_off[c+1] are the source rows of cluster c.
    This is the layout the HIP predict kernel consumes directly (LDS-staged
    source tiles), replacing the reference's per-cluster pointer arrays
    (clus_source_t, Dirac_common.h:173).
    """

    FIELDS = ('ll', 'mm', 'nn1', 'sI', 'sQ', 'sU', 'sV',
              'sI0', 'sQ0', 'sU0', 'sV0', 'spec_idx', 'spec_idx1',
              'spec_idx2', 'f0', 'eX', 'eY', 'eP', 'cxi', 'sxi', 'cphi',
              'sphi')

    def __init__(self, clusters, dtype=torch.float64, device='cpu'):
        import numpy as np
        self.M = len(clusters)
        self.nchunk = torch.tensor([c.nchunk for c in clusters],
                                   dtype=torch.long, device=device)
        self.cluster_ids = [c.cluster_id for c in clusters]
        offs = [0]
        for c in clusters:
            offs.append(offs[-1] + c.nsrc)
        self.cluster_off = torch.tensor(offs, dtype=torch.long, device=device)
        cat = lambda k: torch.tensor(
            np.concatenate([getattr(c, k) for c in clusters]) if clusters
            else np.zeros(0), dtype=dtype, device=device)
        for f_ in self.FIELDS:
            key = f_ if f_ != 'nn1' else 'nn1'
            setattr(self, f_, cat(key))
        self.stype = torch.tensor(
            np.concatenate([c.stype for c in clusters]).astype(np.int32),
            dtype=torch.int32, device=device)
        self.use_proj = torch.tensor(
            np.concatenate([c.use_proj for c in clusters]).astype(np.int32),
            dtype=torch.int32, device=device)
        self.K = int(self.cluster_off[-1])
        # shapelet metadata: global source idx -> (n0, beta, modes)
        self.shapelets = {}
        for ci, c in enumerate(clusters):
            base = offs[ci]
            for (li, n0, beta, coeff) in c.shapelets:
                self.shapelets[base + li] = (int(n0), float(beta),
                                             np.asarray(coeff))

    def to(self, device, dtype=None):
        for f_ in self.FIELDS:
            t = getattr(self, f_)
            setattr(self, f_, t.to(device=device,
                                   dtype=dtype or t.dtype))
        self.stype = self.stype.to(device)
        self.use_proj = self.use_proj.to(device)
        self.cluster_off = self.cluster_off.to(device)
        self.nchunk = self.nchunk.to(device)
        return self


# ---------------------------------------------------------------------------
# Predict: per-cluster coherencies  (reference predict.c:120-263,
# predict_model.cu kernel_coherencies predict_model.cu:1059)
# ---------------------------------------------------------------------------

def _source_flux_at(pack, sel, freq, freq0):
    """Flux at `freq` via the log-polynomial spectral index
    (readsky.c:353-376 sign-preserving form). sel indexes sources."""
    if abs(freq - freq0) < 1.0:  # at the packed reference freq: precomputed
        return (pack.sI[sel], pack.sQ[sel], pack.sU[sel], pack.sV[sel])
    lf = torch.log(freq / pack.f0[sel])
    flog = (pack.spec_idx[sel] * lf + pack.spec_idx1[sel] * lf ** 2
            + pack.spec_idx2[sel] * lf ** 3)
    def scale(s0):
        mag = torch.exp(torch.log(torch.abs(s0).clamp_min(1e-300)) + flog)
        return torch.where(s0 == 0, torch.zeros_like(s0),
                           torch.sign(s0) * mag)
    return (scale(pack.sI0[sel]), scale(pack.sQ0[sel]),
            scale(pack.sU0[sel]), scale(pack.sV0[sel]))


def _extended_envelope(pack, sel, ul, vl, wl):
    """Extended-source uv-envelope at wavelength-scaled (u,v,w) [B, S].

    gaussian/disk/ring per predict.c:32-90; shapelet handled separately
    (shapelet.py). ul,vl,wl: [B,1] broadcast against source rows sel [S]."""
    stype = pack.stype[sel]
    if bool((stype == 0).all()):
        return None
    cxi, sxi = pack.cxi[sel], pack.sxi[sel]
    cphi, sphi = pack.cphi[sel], pack.sphi[sel]
    up_p = ul * cxi - vl * cphi * sxi + wl * sphi * sxi
    vp_p = ul * sxi + vl * cphi * cxi - wl * sphi * cxi
    # gaussian honors use_projection (gaussian_contrib, predict.c:36-44);
    # disk/ring ALWAYS apply the rotation (disk_contrib/ring_contrib,
    # predict.c:60-90 have no use_projection branch) — oracle-verified
    use_p = pack.use_proj[sel].to(torch.bool) | (stype >= 2)
    up = torch.where(use_p, up_p, ul.expand_as(up_p))
    vp = torch.where(use_p, vp_p, vl.expand_as(vp_p))
    eX, eY, eP = pack.eX[sel], pack.eY[sel], pack.eP[sel]
    env = torch.ones_like(up)
    g = stype == 1  # gaussian
    if bool(g.any()):
        cp, sp = torch.cos(eP), torch.sin(eP)
        ut = eX * (cp * up - sp * vp)
        vt = eY * (sp * up + cp * vp)
        ge = torch.exp(-2.0 * math.pi ** 2 * (ut * ut + vt * vt))
        env = torch.where(g, ge, env)
    d = stype == 2  # disk: j1(2*pi*r*|uv|)
    r = stype == 3  # ring: j0(2*pi*r*|uv|)
    if bool(d.any()) or bool(r.any()):
        b = torch.sqrt(up * up + vp * vp) * eX * TWO_PI
        if bool(r.any()):
            env = torch.where(r, _bessel_j0(b), env)
        if bool(d.any()):
            env = torch.where(d, _bessel_j1(b), env)
    # shapelets: gaussian projection handled above is skipped;
    # shapelet envelope computed in shapelet.py and substituted by caller.
    return env


def _bessel_j0(x):
    if hasattr(torch.special, 'bessel_j0'):
        return torch.special.bessel_j0(x)
    import scipy.special as sp
    return torch.from_numpy(sp.j0(x.cpu().numpy())).to(x)


def _bessel_j1(x):
    if hasattr(torch.special, 'bessel_j1'):
        return torch.special.bessel_j1(x)
    import scipy.special as sp
    return torch.from_numpy(sp.j1(x.cpu().numpy())).to(x)


def predict_coh(pack, u, v, w, freq, freq0, fdelta, tdelta, dec0,
                clusters=None, shapelet_env=None):
    """Coherencies per cluster at one frequency: [M, B, 2, 2] complex.

    u,v,w: [B] in seconds. freq: channel frequency. freq0: the pack's flux
    reference frequency. fdelta: channel bandwidth for freq smearing.
    tdelta: integration time for time smearing. dec0: phase-centre dec.

    Reference math: predict.c:163-243 (phase 2*pi*(u l + v m + w (n-1)),
    freq smearing |sinc(G*fdelta/2)|, time smearing erf formula, Stokes ->
    coherency [[I+Q, U+jV],[U-jV, I-Q]]).
    """
    B = u.shape[0]
    M = pack.M
    cdtype = torch.complex128 if u.dtype == torch.float64 else torch.complex64
    out = torch.zeros(M, B, 2, 2, dtype=cdtype, device=u.device)
    uc, vc, wc = u.unsqueeze(1), v.unsqueeze(1), w.unsqueeze(1)
    for ci in range(M):
        s0, s1 = int(pack.cluster_off[ci]), int(pack.cluster_off[ci + 1])
        sel = slice(s0, s1)
        ll, mm, nn1 = pack.ll[sel], pack.mm[sel], pack.nn1[sel]
        G = TWO_PI * (uc * ll + vc * mm + wc * nn1)          # [B,S]
        ph = G * freq
        phr, phi = torch.cos(ph), torch.sin(ph)
        # freq smearing
        smfac = G * (fdelta * 0.5)
        sm = torch.where(G.abs() > 0,
                         torch.abs(torch.sinc(smfac / math.pi)),
                         torch.ones_like(G))
        # time smearing (predict.c:94-107)
        if tdelta > 0:
            bl = torch.sqrt(uc * uc + vc * vc + wc * wc) * freq
            ds = math.sin(dec0) * mm
            r1 = torch.sqrt(ll * ll + ds * ds)
            prod = OMEGA_EARTH * tdelta * bl * r1
            smt = torch.where(prod > CLM_EPSILON,
                              1.0645 * torch.erf(0.8326 * prod) / prod.clamp_min(CLM_EPSILON),
                              torch.ones_like(prod))
            sm = sm * smt
        phr, phi = phr * sm, phi * sm
        env = _extended_envelope(pack, sel, uc * freq, vc * freq, wc * freq)
        if env is not None:
            phr, phi = phr * env, phi * env
        # shapelet sources: complex uv envelope (shapelet.py)
        if getattr(pack, 'shapelets', None):
            from .. import shapelet as shmod
            for gi, (n0, beta, modes) in pack.shapelets.items():
                if not (s0 <= gi < s1):
                    continue
                si = gi - s0
                envc = shmod.shapelet_contrib(
                    u * freq, v * freq, w * freq,
                    float(pack.eX[gi]), float(pack.eY[gi]),
                    float(pack.eP[gi]), float(pack.cxi[gi]),
                    float(pack.sxi[gi]), float(pack.cphi[gi]),
                    float(pack.sphi[gi]), bool(pack.use_proj[gi]),
                    beta, n0, modes)
                pc = torch.complex(phr[:, si], phi[:, si]) * envc
                phr[:, si] = pc.real
                phi[:, si] = pc.imag
        I, Q, U, V = _source_flux_at(pack, sel, float(freq), freq0)
        Ph = torch.complex(phr, phi)
        IIl = Ph * I
        QQl = Ph * Q
        UUl = Ph * U
        VVl = Ph * V
        out[ci, :, 0, 0] = (IIl + QQl).sum(dim=1)
        out[ci, :, 0, 1] = (UUl + 1j * VVl).sum(dim=1)
        out[ci, :, 1, 0] = (UUl - 1j * VVl).sum(dim=1)
        out[ci, :, 1, 1] = (IIl - QQl).sum(dim=1)
    return out


# ---------------------------------------------------------------------------
# Model application / residuals (reference residual.c, predict_model.cu
# kernel_residuals:1241)
# ---------------------------------------------------------------------------

def apply_jones(coh, J, bb, chunk_rows=None):
    """Model visibilities of one cluster: V_b = J_p C_b J_q^H.

    coh: [B,2,2] complex; J: [nchunk, N, 2, 2] complex; bb: [B,2] long.
    chunk_rows: [B] long chunk index per row (time-chunk hybrid mapping,
    reference mderiv.cu:82-93); None = single chunk 0."""
    if chunk_rows is None:
        Jp = J[0, bb[:, 0]]
        Jq = J[0, bb[:, 1]]
    else:
        Jp = J[chunk_rows, bb[:, 0]]
        Jq = J[chunk_rows, bb[:, 1]]
    return Jp @ coh @ Jq.conj().transpose(-1, -2)



def chunk_rows_for(ci, nchunks, T, Nbase, B, device):
    """Chunk index per baseline row for cluster ci given its nchunk count.

    Time-major rows: row // Nbase = timeslot; chunks split T slots evenly
    with remainder to the last chunk (reference lmfit.c:893-967 chunk loop)."""
    nc = nchunks[ci] if not torch.is_tensor(nchunks) else int(nchunks[ci])
    if nc <= 1:
        return None
    t_idx = torch.arange(B, device=device) // Nbase
    tpc = (T + nc - 1) // nc
    return (t_idx // tpc).clamp_max(nc - 1)


# ---------------------------------------------------------------------------
# JtJ / Jtr assembly — the LM core, closed 2x2-complex form.
# ---------------------------------------------------------------------------

def realify(Mc):
    """Complex [..,2,2] -> real [..,4,4]: each entry a -> [[ar,-ai],[ai,ar]]
    (complex-linear map realification)."""
    sh = Mc.shape[:-2]
    out = torch.zeros(*sh, 4, 4, dtype=Mc.real.dtype, device=Mc.device)
    ar, ai = Mc.real, Mc.imag
    out[..., 0::2, 0::2] = ar
    out[..., 0::2, 1::2] = -ai
    out[..., 1::2, 0::2] = ai
    out[..., 1::2, 1::2] = ar
    return out


def antirealify(Mc):
    """Complex [..,2,2] -> real [..,4,4] of the ANTI-linear map
    x -> M conj(x): entry a -> [[ar, ai],[ai, -ar]]."""
    sh = Mc.shape[:-2]
    out = torch.zeros(*sh, 4, 4, dtype=Mc.real.dtype, device=Mc.device)
    ar, ai = Mc.real, Mc.imag
    out[..., 0::2, 0::2] = ar
    out[..., 0::2, 1::2] = ai
    out[..., 1::2, 0::2] = ai
    out[..., 1::2, 1::2] = -ar
    return out


def vecR(Mc):
    """Complex [..,2,2] -> real [..,8]: row-major interleaved re/im."""
    return torch.view_as_real(Mc).reshape(*Mc.shape[:-2], 8)


def matC(v):
    """Inverse of vecR."""
    return torch.view_as_complex(v.reshape(*v.shape[:-1], 2, 2, 2).contiguous())


def jtj_jtr(x, coh, J, bb, N, weights=None, chunk_rows=None, nchunk=1):
    """Assemble Gauss-Newton JtJ [nchunk,8N,8N], Jtr [nchunk,8N] and cost for
    ONE cluster, WITHOUT forming the dense Jacobian.

    Per baseline b=(p,q), with G1 = C J_q^H, K = J_p C, r = x - J_p C J_q^H:
      grad_p (complex 2x2)  = r G1^H          -> Jtr rows of station p
      grad_q (complex 2x2)  = r^H K           -> Jtr rows of station q
      H[p,p] += realify(conj(G1 G1^H)) ⊗ I2-block structure
      H[q,q] += realify(conj(K^H K))
      H[p,q] += antirealify-structured cross conj(G1_{jk}) K_{ib}
    (derivation: V=J1 C J2^H is complex-linear in J1, anti-linear in J2;
    realified Gauss-Newton blocks follow. Replaces the reference's dense
    M x 8N Jacobian + cuBLAS J^T J (mderiv.cu:1069, clmfit_cuda.c:364).)

    Verified against torch.autograd in tests/test_lm.py.
    """
    dev = x.device
    rdt = x.real.dtype
    if chunk_rows is None:
        chunk_rows = torch.zeros(x.shape[0], dtype=torch.long, device=dev)
    Jp = J[chunk_rows, bb[:, 0]]
    Jq = J[chunk_rows, bb[:, 1]]
    G1 = coh @ Jq.conj().transpose(-1, -2)          # [B,2,2]
    K = Jp @ coh
    V = Jp @ G1
    r = x - V
    if weights is not None:
        wt = weights
    else:
        wt = torch.ones(x.shape[0], dtype=rdt, device=dev)
    cost = (wt * (r.abs() ** 2).sum(dim=(-1, -2))).sum()
    w2 = wt[:, None, None]

    # gradients (2x2 complex per station)
    gp = (w2 * (r @ G1.conj().transpose(-1, -2)))     # [B,2,2]
    gq = (w2 * (r.conj().transpose(-1, -2) @ K))
    # diag blocks (2x2 complex Gram); expand to 4x4 real, then to 8x8 via I2⊗
    Q1 = (w2 * (G1 @ G1.conj().transpose(-1, -2))).conj()   # conj(G1 G1^H)
    Q2 = (w2 * (K.conj().transpose(-1, -2) @ K)).conj()     # conj(K^H K)
    # cross block: C12[(i,j),(k,b)] = conj(G1[j,k]) * K[i,b]  (4x4 complex)
    C12 = (w2.unsqueeze(-1).unsqueeze(-1)
           * G1.conj().permute(0, 1, 2)[:, None, :, :, None]
           * K[:, :, None, None, :])                 # [B, i, j, k, b]

    Mt = nchunk
    JtJ = torch.zeros(Mt, 8 * N, 8 * N, dtype=rdt, device=dev)
    Jtr = torch.zeros(Mt, 8 * N, dtype=rdt, device=dev)

    # scatter per chunk
    p_idx = bb[:, 0]
    q_idx = bb[:, 1]
    lin_p = chunk_rows * N + p_idx
    lin_q = chunk_rows * N + q_idx

    # Jtr: vecR of [gp rows] into 8 slots of station p
    Jtr_flat = Jtr.view(Mt * N, 8)
    Jtr_flat.index_add_(0, lin_p, vecR(gp))
    Jtr_flat.index_add_(0, lin_q, vecR(gq))

    # JtJ diag blocks: I2 ⊗ realify(Q) at (p,p)
    R1 = realify(Q1)   # [B,4,4]
    R2 = realify(Q2)
    diag_pp = torch.zeros(x.shape[0], 8, 8, dtype=rdt, device=dev)
    diag_pp[:, 0:4, 0:4] = R1
    diag_pp[:, 4:8, 4:8] = R1
    diag_qq = torch.zeros(x.shape[0], 8, 8, dtype=rdt, device=dev)
    diag_qq[:, 0:4, 0:4] = R2
    diag_qq[:, 4:8, 4:8] = R2
    # cross 8x8: rows = j1 params (i,j)->(2i+j)*2+(re/im), cols = j2 (k,b)
    cross = antirealify_cross(C12)

    # accumulate into block matrix via index_add on flattened blocks
    blk = torch.zeros(Mt * N * N, 8, 8, dtype=rdt, device=dev)
    blk.index_add_(0, chunk_rows * N * N + p_idx * N + p_idx, diag_pp)
    blk.index_add_(0, chunk_rows * N * N + q_idx * N + q_idx, diag_qq)
    blk.index_add_(0, chunk_rows * N * N + p_idx * N + q_idx, cross)
    blk.index_add_(0, chunk_rows * N * N + q_idx * N + p_idx,
                   cross.transpose(-1, -2))
    JtJ = (blk.view(Mt, N, N, 8, 8).permute(0, 1, 3, 2, 4)
           .reshape(Mt, 8 * N, 8 * N))
    return JtJ, Jtr, cost


def antirealify_cross(C12):
    """C12: [B, i, j, k, b] complex with entries conj(G1_{jk}) K_{ib}.
    Build real 8x8 cross blocks: row index (i,j,re/im of J1 entry (i,j)),
    col index (k,b,re/im of J2 entry (k,b)); each complex scalar c acting
    anti-linearly: [[cr, ci],[ci, -cr]]."""
    B = C12.shape[0]
    # reorder to [B, (i,j), (k,b)]
    c = C12.permute(0, 1, 2, 3, 4).reshape(B, 4, 4)
    out = torch.zeros(B, 8, 8, dtype=C12.real.dtype, device=C12.device)
    cr, ci_ = c.real, c.imag
    out[:, 0::2, 0::2] = cr
    out[:, 0::2, 1::2] = ci_
    out[:, 1::2, 0::2] = ci_
    out[:, 1::2, 1::2] = -cr
    return out


def model_and_cost(x, coh, J, bb, weights=None, chunk_rows=None):
    """Weighted cost sum w_b ||x_b - J_p C J_q^H||^2 (kernel_fcost
    mderiv.cu:822)."""
    V = apply_jones(coh, J, bb, chunk_rows)
    r = x - V
    e2 = (r.abs() ** 2).sum(dim=(-1, -2))
    if weights is not None:
        e2 = e2 * weights
    return e2.sum(), r


def robust_cost(x, coh, J, bb, nu, chunk_rows=None):
    """Student's-t cost sum ln(1 + ||e||^2/nu) (kernel_fcost_robust
    mderiv.cu:707)."""
    V = apply_jones(coh, J, bb, chunk_rows)
    r = x - V
    e2 = (r.abs() ** 2).sum(dim=(-1, -2))
    return torch.log1p(e2 / nu).sum(), r


# ---------------------------------------------------------------------------
# Robust (Student's-t) weight and nu updates (reference robust.cu:454-520,
# updatenu.c:263-338)
# ---------------------------------------------------------------------------

def update_weights(r, nu, p=8):
    """w_b = (nu + p) / (nu + ||e_b||^2), p=8 for full 2x2 complex residual
    (robust.cu kernel_updateweights)."""
    e2 = (r.abs() ** 2).sum(dim=(-1, -2))
    return (nu + p) / (nu + e2)


def update_nu_aecm(w, nu_old, nulow=2.0, nuhigh=30.0, Nd=30, p=8):
    """AECM grid search for nu (updatenu.c:263-338):
    minimize |q(nu)| over grid, q = -psi(nu/2)+log(nu/2) + mean(log w - w)
    + psi((nu_old+p)/2) - log((nu_old+p)/2) + 1."""
    logsumw = (torch.log(w) - w).mean()
    dgm = torch.special.digamma(torch.tensor((nu_old + p) * 0.5,
                                             dtype=w.dtype, device=w.device))
    dgm = dgm - math.log((nu_old + p) * 0.5)
    grid = torch.linspace(nulow, nuhigh, Nd, dtype=w.dtype, device=w.device)
    q = (-torch.special.digamma(grid * 0.5) + torch.log(grid * 0.5)
         + logsumw + dgm + 1.0)
    idx = torch.argmin(torch.abs(q))
    return float(grid[idx])


# ---------------------------------------------------------------------------
# Full-parameter gradient for LBFGS (reference mderiv.cu kernel_deriv:29,
# robust.cu kernel_deriv_robust)
# ---------------------------------------------------------------------------

def lbfgs_cost_grad(x, cohs, J_packed, chunk_off, nchunks, bb, T, Nbase,
                    robust_nu=None, weights=None):
    """Cost and gradient over the FULL parameter vector (all clusters,
    all chunks): returns (cost, grad [Mt*N*8]).

    Gaussian: cost = sum ||r||^2, grad_p^(c) = -2 * realpack(r G1^H) summed
    over baselines (the reference's kernel_deriv computes the same value
    thread-per-parameter; here per-baseline 2x2 products + index_add — the
    memory-efficient layout noted in SURVEY.md §7 wave 2).
    Robust: cost = sum ln(1+||r||^2/nu), per-baseline scale (nu+8)/(nu+||r||^2)
    ... actually d/dtheta ln(1+e/nu) = (1/(nu+e)) de — constant 8 appears in
    the reference's weight; here grad uses 1/(nu+e2) * nu-normalized form
    matching robust_lbfgs.c:94-131."""
    dev = x.device
    M = cohs.shape[0]
    N = int(J_packed.shape[1])
    B = x.shape[0]
    Mt = J_packed.shape[0]
    # total model
    Vtot = torch.zeros_like(x)
    per_cluster = []
    for ci in range(M):
        rows = chunk_rows_for(ci, nchunks, T, Nbase, B, dev)
        rows_idx = rows if rows is not None else torch.zeros(
            B, dtype=torch.long, device=dev)
        Jc = J_packed[chunk_off[ci]:chunk_off[ci] + int(nchunks[ci])]
        Vc = apply_jones(cohs[ci], Jc, bb, rows)
        Vtot = Vtot + Vc
        per_cluster.append(rows_idx)
    r = x - Vtot
    e2 = (r.abs() ** 2).sum(dim=(-1, -2))
    if robust_nu is not None:
        cost = torch.log1p(e2 / robust_nu).sum()
        scale = 1.0 / (robust_nu + e2)           # robust_lbfgs.c:94-131
    else:
        cost = e2.sum()
        scale = torch.ones_like(e2)
    if weights is not None:
        cost = cost  # weights already folded into x upstream if used
    grad = torch.zeros(Mt * N, 8, dtype=x.real.dtype, device=dev)
    rs = r * scale[:, None, None]
    for ci in range(M):
        rows_idx = per_cluster[ci]
        Jc_rows = chunk_off[ci] + rows_idx
        Jp = J_packed[Jc_rows, bb[:, 0]]
        Jq = J_packed[Jc_rows, bb[:, 1]]
        G1 = cohs[ci] @ Jq.conj().transpose(-1, -2)
        K = Jp @ cohs[ci]
        gp = rs @ G1.conj().transpose(-1, -2)
        gq = rs.conj().transpose(-1, -2) @ K
        grad.index_add_(0, Jc_rows * N + bb[:, 0], -2.0 * vecR(gp))
        grad.index_add_(0, Jc_rows * N + bb[:, 1], -2.0 * vecR(gq))
    return cost, grad.reshape(-1)


# ---------------------------------------------------------------------------
# Joint (all-cluster) Gauss-Newton assembly.
# The reference polishes jointly only with LBFGS (lmfit.c:1019-1037); a full
# joint LM with cross-cluster JtJ blocks converges quadratically and is
# GEMM/batched-solve shaped — ideal for MI355X. Used as the final refinement
# after SAGE EM sweeps (solvers/sage.py).
# ---------------------------------------------------------------------------

def joint_jtj_jtr(x, cohs, J_packed, chunk_off, nchunks, bb, T, Nbase,
                  weights=None):
    """JtJ [P,P], Jtr [P] over the FULL parameter vector, P = 8*Mt*N.

    Per baseline b and cluster ci: G1_ci = C_ci Jq^H, K_ci = Jp C_ci.
    Blocks between (station s1 of ci) and (station s2 of cj) are nonzero
    only when {s1,s2} ⊆ {p(b),q(b)}:
      lin-lin   (p,ci)x(p,cj): I2 ⊗ realify(conj(G1_ci G1_cj^H))
      anti-anti (q,ci)x(q,cj): I2 ⊗ realify(conj(K_ci^H K_cj))
      lin-anti  (p,ci)x(q,cj): antirealify 4x4-complex with entries
                conj(G1_ci[j,k]) K_cj[i,b]
    (and symmetric transposes). Derivation as in jtj_jtr().
    """
    dev = x.device
    rdt = x.real.dtype
    cdt = x.dtype
    M = cohs.shape[0]
    N = J_packed.shape[1]
    Mt = J_packed.shape[0]
    B = x.shape[0]
    MtN = Mt * N
    wt = weights if weights is not None else torch.ones(B, dtype=rdt,
                                                        device=dev)

    # per-cluster primitives and residual
    rows_l, G1_l, K_l, lin_p, lin_q = [], [], [], [], []
    Vtot = torch.zeros_like(x)
    for ci in range(M):
        rows = chunk_rows_for(ci, nchunks, T, Nbase, B, dev)
        if rows is None:
            rows = torch.zeros(B, dtype=torch.long, device=dev)
        rows = rows + chunk_off[ci]
        Jp = J_packed[rows, bb[:, 0]]
        Jq = J_packed[rows, bb[:, 1]]
        G1 = cohs[ci] @ Jq.conj().transpose(-1, -2)
        K = Jp @ cohs[ci]
        Vtot = Vtot + Jp @ G1
        rows_l.append(rows)
        G1_l.append(G1)
        K_l.append(K)
        lin_p.append(rows * N + bb[:, 0])
        lin_q.append(rows * N + bb[:, 1])
    r = x - Vtot
    cost = (wt * (r.abs() ** 2).sum(dim=(-1, -2))).sum()
    w2 = wt[:, None, None]

    Jtr = torch.zeros(MtN, 8, dtype=rdt, device=dev)
    # complex accumulators: "same-kind" blocks [MtN,MtN] of 2x2 complex,
    # and "cross-kind" blocks of 4x4 complex (i,j)x(k,b)
    Hlin = torch.zeros(MtN * MtN, 2, 2, dtype=cdt, device=dev)
    Hcross = torch.zeros(MtN * MtN, 4, 4, dtype=cdt, device=dev)
    touched_lin = []
    touched_cross = []

    def addc(dst, idx, val):
        torch.view_as_real(dst).reshape(dst.shape[0], -1).index_add_(
            0, idx,
            torch.view_as_real(val.resolve_conj()).reshape(val.shape[0], -1))

    for ci in range(M):
        gp = w2 * (r @ G1_l[ci].conj().transpose(-1, -2))
        gq = w2 * (r.conj().transpose(-1, -2) @ K_l[ci])
        Jtr.index_add_(0, lin_p[ci], vecR(gp))
        Jtr.index_add_(0, lin_q[ci], vecR(gq))
        for cj in range(M):
            Epp = (w2 * (G1_l[ci] @ G1_l[cj].conj().transpose(-1, -2))).conj()
            Eqq = (w2 * (K_l[ci].conj().transpose(-1, -2) @ K_l[cj])).conj()
            addc(Hlin, lin_p[ci] * MtN + lin_p[cj], Epp)
            addc(Hlin, lin_q[ci] * MtN + lin_q[cj], Eqq)
            # cross (p of ci) x (q of cj): [i,j,k,b] = conj(G1_ci[j,k]) K_cj[i,b]
            Cx = (w2.unsqueeze(-1).unsqueeze(-1)
                  * G1_l[ci].conj()[:, None, :, :, None]
                  * K_l[cj][:, :, None, None, :]).reshape(B, 4, 4)
            addc(Hcross, lin_p[ci] * MtN + lin_q[cj], Cx)
            touched_cross.append((ci, cj))

    # expand to real H
    H = torch.zeros(MtN, 8, MtN, 8, dtype=rdt, device=dev)
    Hlin = Hlin.view(MtN, MtN, 2, 2)
    R4 = realify(Hlin).permute(0, 2, 1, 3)   # [MtN,4,MtN,4]
    H[:, 0:4, :, 0:4] += R4
    H[:, 4:8, :, 4:8] += R4
    Hc = Hcross.view(MtN, MtN, 4, 4)
    X = torch.zeros(MtN, MtN, 8, 8, dtype=rdt, device=dev)
    cr, ci_ = Hc.real, Hc.imag
    X[..., 0::2, 0::2] = cr
    X[..., 0::2, 1::2] = ci_
    X[..., 1::2, 0::2] = ci_
    X[..., 1::2, 1::2] = -cr
    H += X.permute(0, 2, 1, 3)
    H += X.permute(1, 3, 0, 2)       # symmetric transpose of cross blocks
    H = H.reshape(8 * MtN, 8 * MtN)
    return H, Jtr.reshape(-1), cost
