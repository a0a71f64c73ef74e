"""Shapelet source models: uv-domain basis evaluation and modes-file I/O.

Re-implements /root/reference/src/lib/Radio/shapelet.c:
  - calculate_uv_mode_vectors_scalar (:49-135): per (n1,n2) mode the value
    phi_n1(-u beta) phi_n2(v beta) with phi_n(x) = H_n(x) exp(-x^2/2) /
    sqrt(2^{n+1} n!) (physicists' Hermite), real for even n1+n2 with sign
    (-1)^{(n1+n2)/2}, imaginary for odd with sign (-1)^{(n1+n2-1)/2};
  - shapelet_contrib (:141-188): projection rotation (NEGATED u,v relative
    to the gaussian projection), 1/eX,1/eY scaling, eP rotation, summed
    modes x 2*pi*a*b;
  - read_shapelet_modes (readsky.c:149-190): `<name>.fits.modes` text file
    (RA/Dec line, then n0 beta, then n0^2 `idx value` rows).

Vectorized torch (runs on CPU or GPU tensors); the HIP predict kernel
leaves shapelet sources to this path (their fluxes are zeroed in the
kernel pack and their contribution is added by the host — sources per
cluster are few while baselines are many, so the torch pass is cheap).
The device-side malloc + recursive Hermite of the reference's CUDA path
(predict_model.cu:708-784) is replaced by closed-loop recurrences on
tensors (SURVEY.md §7 "shapelets" hard-part note).
"""
import math

import numpy as np
import torch


def hermite_phi(x, n0):
    """phi_n(x) = H_n(x) exp(-x^2/2) / sqrt(2^{n+1} n!) for n < n0.
    x: [...]; returns [..., n0]."""
    out = torch.empty(*x.shape, n0, dtype=x.dtype, device=x.device)
    e = torch.exp(-0.5 * x * x)
    H_nm1 = None
    H_nm2 = None
    fact = 1.0
    for n in range(n0):
        if n == 0:
            H = torch.ones_like(x)
        elif n == 1:
            H = 2 * x
        else:
            H = 2 * x * H_nm1 - 2 * (n - 1) * H_nm2
        norm = math.sqrt(2.0 ** (n + 1) * fact)
        out[..., n] = H * e / norm
        H_nm2, H_nm1 = H_nm1, H
        fact *= (n + 1)
    return out


def uv_mode_vectors(u, v, beta, n0):
    """Complex mode values [B, n0, n0] at (u, v): mode (n1, n2) =
    sign * phi_n1(u b) phi_n2(v b), x (i) when n1+n2 odd."""
    pu = hermite_phi(u * beta, n0)       # [B, n0]
    pv = hermite_phi(v * beta, n0)
    prod = pu[:, None, :] * pv[:, :, None]   # [B, n2, n1]
    n1 = torch.arange(n0, device=u.device)
    nsum = n1[None, :] + n1[:, None]         # [n2, n1]
    odd = (nsum % 2) == 1
    sign = torch.where(odd, (-1.0) ** ((nsum - 1) // 2),
                       (-1.0) ** (nsum // 2)).to(u.dtype)
    val = prod * sign
    zero = torch.zeros_like(val)
    re = torch.where(odd, zero, val)
    im = torch.where(odd, val, zero)
    return torch.complex(re, im)          # [B, n2, n1] complex


def shapelet_contrib(u, v, w, eX, eY, eP, cxi, sxi, cphi, sphi, use_proj,
                     beta, n0, modes):
    """Complex envelope [B] for one shapelet source.

    u,v,w in wavelengths; modes: [n0*n0] (column-major (n2*n0+n1) as in
    the reference's mode files)."""
    if use_proj:
        up = -u * cxi + v * cphi * sxi - w * sphi * sxi
        vp = -u * sxi - v * cphi * cxi + w * sphi * cxi
    else:
        up, vp = u, v
    a, b = 1.0 / eX, 1.0 / eY
    cp, sp = math.cos(eP), math.sin(eP)
    ut = a * (cp * up - sp * vp)
    vt = b * (sp * up + cp * vp)
    basis = uv_mode_vectors(-ut, vt, beta, n0)     # [B, n2, n1]
    m = torch.as_tensor(modes, dtype=basis.real.dtype,
                        device=basis.device).reshape(n0, n0)
    out = (basis * m[None]).sum(dim=(-1, -2))
    return 2.0 * math.pi * a * b * out


def read_modes_file(path):
    """Parse `<name>.fits.modes` (readsky.c:149-190). Returns
    (n0, beta, modes[n0*n0])."""
    with open(path) as f:
        toks = f.read().split()
    # RA h m s, Dec d m s (ignored)
    idx = 6
    n0 = int(toks[idx]); beta = float(toks[idx + 1])
    idx += 2
    M = n0 * n0
    modes = np.zeros(M)
    for ci in range(M):
        modes[ci] = float(toks[idx + 2 * ci + 1])
    return n0, beta, modes


def write_modes_file(path, ra, dec, n0, beta, modes):
    """Writer (for tests / buildsky shapelet decomposition output)."""
    with open(path, 'w') as f:
        f.write("0 0 0.0 0 0 0.0\n")
        f.write(f"{n0} {beta:.9e}\n")
        for i, m in enumerate(np.asarray(modes).reshape(-1)):
            f.write(f"{i} {m:.9e}\n")


def image_basis(l, m, n0, beta):
    """Image-plane basis dual to uv_mode_vectors (verified to machine
    precision against the 2-D DFT): mode (n1, n2) image function
    (4 pi^2 / beta^2) phi_n1(2 pi l / beta) phi_n2(2 pi m / beta)
    under V(u,v) = int f(l,m) e^{2 pi i (-u l + v m)} dl dm (the
    reference's "decompose f(-l,m)" convention, shapelet.c:171).
    Returns [B, n0*n0] (column-major modes: n2*n0+n1)."""
    C = 4.0 * math.pi ** 2 / beta ** 2
    lb = torch.as_tensor(l, dtype=torch.float64) * (2 * math.pi / beta)
    mb = torch.as_tensor(m, dtype=torch.float64) * (2 * math.pi / beta)
    pl = hermite_phi(lb, n0)
    pm = hermite_phi(mb, n0)
    return (C * pl[:, None, :] * pm[:, :, None]).reshape(len(lb),
                                                         n0 * n0)


def decompose_image(l, m, flux, n0, beta):
    """Least-squares shapelet decomposition of image-plane fluxes (the
    role of buildsky's shapelet fitting / shapelet_modes, shapelet.c API
    Dirac_radio.h:417), returning modes that shapelet_contrib renders
    back to the same image (Fourier-dual bases)."""
    Bmat = image_basis(l, m, n0, beta)
    y = torch.as_tensor(flux, dtype=torch.float64)
    sol = torch.linalg.lstsq(Bmat, y.unsqueeze(-1)).solution.squeeze(-1)
    return sol.numpy()


# ---------------------------------------------------------------------------
# Shapelet products (shapelet.c shapelet_product* / diffuse_predict.c):
# the pointwise product of two Gauss-Hermite series with scales alpha,
# beta is EXACTLY a series with scale gamma, 1/gamma^2 = 1/alpha^2 +
# 1/beta^2, of order i+j — the machinery that applies a station-dependent
# Jones-valued spatial model Z(l,m) to a diffuse shapelet sky:
# Jp(l,m) C(l,m) Jq(l,m)^H per baseline, all three shapelet series.
# ---------------------------------------------------------------------------

def product_scale(alpha, beta):
    """gamma with 1/gamma^2 = 1/alpha^2 + 1/beta^2."""
    return 1.0 / math.sqrt(1.0 / alpha ** 2 + 1.0 / beta ** 2)


def shapelet_product_tensor(L, M_, N_, alpha, beta, gamma):
    """1-D product-projection tensor T [N_, L, M_]:
    phi_i(x/alpha) phi_j(x/beta) = sum_k T[k, i, j] phi_k(x/gamma)
    (exact when N_ > L-1 + M_-1 and gamma = product_scale(alpha, beta)).
    Computed by Gauss-Hermite quadrature (integrand is a polynomial times
    the combined Gaussian, so the quadrature is exact)."""
    import numpy.polynomial.hermite as H
    # int phi_i^a phi_j^b phi_k^g dx ; phi normalization per hermite_phi
    # orthonormality: int phi_m^g phi_n^g dx = (gamma sqrt(pi)/2) delta
    a2 = 0.5 / alpha ** 2
    b2 = 0.5 / beta ** 2
    g2 = 0.5 / gamma ** 2
    s = math.sqrt(a2 + b2 + g2)
    deg = L + M_ + N_ + 2
    t, wq = H.hermgauss(deg)
    x = t / s
    pa = hermite_phi(torch.tensor(x / alpha), L).numpy()
    pb = hermite_phi(torch.tensor(x / beta), M_).numpy()
    pg = hermite_phi(torch.tensor(x / gamma), N_).numpy()
    # strip the Gaussians (hermgauss supplies exp(-t^2) itself)
    pa = pa * np.exp(0.5 * a2 * 2 * (x ** 2))[:, None]
    pb = pb * np.exp(0.5 * b2 * 2 * (x ** 2))[:, None]
    pg = pg * np.exp(0.5 * g2 * 2 * (x ** 2))[:, None]
    scale = 1.0 / s
    T = np.einsum('q,qi,qj,qk->kij', wq * scale, pa, pb, pg)
    return torch.tensor(T / (gamma * math.sqrt(math.pi) / 2.0))


def shapelet_product_jones(F, G, alpha, beta, n_out=None, conj_g=False):
    """Jones-valued 2-D series product: F [Lf*Lf, 2, 2] (scale alpha,
    column-major modes j2*Lf+j1) times G [Lg*Lg, 2, 2] (scale beta) ->
    H [n_out*n_out, 2, 2] at scale product_scale(alpha, beta)
    (shapelet_product_jones, shapelet.c API Dirac_radio.h:420-430).
    conj_g: use G^H (per-mode conjugate transpose — valid when G's
    image-plane series is Hermitian-symmetric in the Jones sense)."""
    Lf = int(math.isqrt(F.shape[0]))
    Lg = int(math.isqrt(G.shape[0]))
    if n_out is None:
        n_out = Lf + Lg - 1
    gamma = product_scale(alpha, beta)
    T = shapelet_product_tensor(Lf, Lg, n_out, alpha, beta, gamma).to(
        torch.complex128)
    Fm = F.reshape(Lf, Lf, 2, 2).to(torch.complex128)
    Gm = G.reshape(Lg, Lg, 2, 2).to(torch.complex128)
    if conj_g:
        Gm = Gm.conj().transpose(-1, -2)
    # modes indexed [n2][n1]: contract each axis with its own T
    Hm = torch.einsum('kac,lbd,abij,cdjm->klim', T, T, Fm, Gm)
    return Hm.reshape(n_out * n_out, 2, 2), gamma


def eval_uv_jones(u, v, modes, beta, a=1.0, b=1.0):
    """uv-plane evaluation of a COMPLEX Jones-valued mode series:
    [B, 2, 2] = sum_k modes[k] . basis_k(-u, v) * 2 pi a b (the complex
    generalization of shapelet_contrib's mode sum; u, v in wavelengths)."""
    n0 = int(math.isqrt(modes.shape[0]))
    basis = uv_mode_vectors(-u, v, beta, n0)       # [B, n2, n1]
    m = modes.reshape(n0, n0, 2, 2).to(torch.complex128)
    out = torch.einsum('bkl,klij->bij', basis.to(torch.complex128), m)
    return 2.0 * math.pi * a * b * out


def recalculate_diffuse_coherencies(u, v, w, bb, Z, beta_z, Cm, beta_c,
                                    ll, mm, nn1, freq, fdelta):
    """Per-baseline coherencies of a diffuse shapelet cluster with a
    per-station Jones-valued spatial model applied
    (recalculate_diffuse_coherencies, diffuse_predict.c:295):
      V_pq = [ Zp(l,m) C(l,m) Zq(l,m)^H ]^(uv)(u_pq, v_pq) * phase * smear
    u,v,w: [B] seconds; bb: [B,2] station pairs; Z: [N, Gz*Gz, 2, 2]
    per-station spatial series (scale beta_z); Cm: [Gc*Gc, 2, 2] sky
    coherency series (scale beta_c); (ll, mm, nn1): source direction;
    freq, fdelta: Hz. Returns [B, 2, 2] complex64/128."""
    N = Z.shape[0]
    # stage 1: C_Zq = C x Zq^H per station
    CZ = []
    for q in range(N):
        h, g1 = shapelet_product_jones(Cm, Z[q], beta_c, beta_z,
                                       conj_g=True)
        CZ.append(h)
    CZ = torch.stack(CZ)
    # stage 2: H_pq = Zp x C_Zq  per needed pair (unique pairs of bb)
    pairs = {}
    for p, q in {(int(p), int(q)) for p, q in bb.tolist()}:
        h, g2 = shapelet_product_jones(Z[p], CZ[q], beta_z, g1)
        pairs[(p, q)] = h
    uf = u * freq
    vf = v * freq
    G = 2.0 * math.pi * (u * ll + v * mm + w * nn1)
    ph = torch.remainder(G * freq, 2.0 * math.pi)
    phc = torch.complex(torch.cos(ph), torch.sin(ph))
    smf = G * (fdelta * 0.5)
    sm = torch.where(smf.abs() > 1e-12, (torch.sin(smf) / smf).abs(),
                     torch.ones_like(smf))
    out = torch.zeros(len(u), 2, 2, dtype=torch.complex128)
    for (p, q), h in pairs.items():
        sel = (bb[:, 0] == p) & (bb[:, 1] == q)
        if bool(sel.any()):
            out[sel] = eval_uv_jones(uf[sel], vf[sel], h, g2)
    return out * (phc * sm)[:, None, None]
