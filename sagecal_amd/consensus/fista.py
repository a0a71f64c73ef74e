"""Spatial-model regularization: elastic-net FISTA + accelerated projected
gradient.

Re-implements /root/reference/src/lib/Dirac/fista.c:
  - update_spatialreg_fista (Dirac.h:1542-1556):
      Z = argmin sum_k ||Z_k - Z Phi_k||^2 + lambda ||Z||^2 + mu ||Z||_1
    where Z_k are per-direction consensus solutions (2 Npoly N x 2) and
    Phi_k the spatial basis evaluated at direction k's sky position
    (2G x 2); solved by FISTA (Beck & Teboulle 2009) with soft threshold;
  - the diffuse-model constraint variant (+ Psi^H (Z - Zdiff) +
    gamma/2 ||Z - Zdiff||^2, Dirac.h:1572);
  - accel_proj_grad (Dirac.h:1587) generic accelerated gradient.
"""
import math

import numpy as np
import torch


def spatial_basis(ll, mm, G0, beta=1.0, kind='shapelet'):
    """Spatial basis Phi [M, G] at cluster direction cosines: shapelet
    (Gauss-Hermite, sagecal -X model order) or spherical-harmonic modes
    (sagecal_master.cpp:293-423 builds these from cluster centroids)."""
    from .. import shapelet as shmod
    M = len(ll)
    if kind == 'shapelet':
        bas = shmod.image_basis(np.asarray(ll), np.asarray(mm), G0, beta)
        return bas.to(torch.float64)          # [M, G0*G0]
    raise ValueError(kind)


def soft_threshold(Z, mu):
    mag = Z.abs()
    scale = (1.0 - mu / mag.clamp_min(1e-30)).clamp_min(0.0)
    return Z * scale


def update_spatialreg_fista(Zbar, Phi, lam=0.01, mu=1e-4, maxiter=40,
                            Zdiff=None, Psi=None, gamma=0.0):
    """Solve Z = argmin sum_k ||Zbar_k - Z Phi_k||^2 + lam ||Z||^2
    + mu ||Z||_1 [+ Re<Psi, Z - Zdiff> + gamma/2 ||Z - Zdiff||^2].

    Zbar: [M, P] complex (per-direction stacked solution vectors);
    Phi:  [M, G] real basis rows. Returns Z [P, G] complex.
    """
    M, P = Zbar.shape
    G = Phi.shape[1]
    Phi = Phi.to(torch.float64)
    A = Phi.T @ Phi + lam * torch.eye(G, dtype=torch.float64)  # [G,G]
    if gamma > 0:
        A = A + 0.5 * gamma * torch.eye(G, dtype=torch.float64)
    # Lipschitz constant of the smooth part
    L = 2.0 * float(torch.linalg.eigvalsh(A).max())
    R = Zbar.T @ Phi.to(Zbar.dtype)            # [P, G]
    Z = torch.zeros(P, G, dtype=Zbar.dtype)
    Y = Z.clone()
    t = 1.0
    for it in range(maxiter):
        # grad of smooth part: 2 (Y A - R) (+ Psi + gamma (Y - Zdiff))
        gradsm = 2.0 * (Y @ A.to(Y.dtype) - R)
        if gamma > 0 and Zdiff is not None:
            gradsm = gradsm + (Psi if Psi is not None else 0) \
                + gamma * (Y - Zdiff)
        Znew = soft_threshold(Y - gradsm / L, mu / L)
        tnew = 0.5 * (1 + math.sqrt(1 + 4 * t * t))
        Y = Znew + ((t - 1) / tnew) * (Znew - Z)
        Z, t = Znew, tnew
    return Z


def accel_proj_grad(cost, grad, p0, itmax=100, lr=None, proj=None):
    """Generic accelerated (projected) gradient (accel_proj_grad,
    Dirac.h:1587; Nesterov/Beck-Teboulle/O'Donoghue restart)."""
    p = p0.clone()
    y = p.clone()
    t = 1.0
    if lr is None:
        g0 = grad(p)
        lr = 1.0 / (float(g0.norm()) + 1.0)
    c_prev = float(cost(p))
    for it in range(itmax):
        g = grad(y)
        pn = y - lr * g
        if proj is not None:
            pn = proj(pn)
        cn = float(cost(pn))
        if cn > c_prev:          # adaptive restart + step shrink
            lr *= 0.5
            y = p.clone()
            t = 1.0
            continue
        tn = 0.5 * (1 + math.sqrt(1 + 4 * t * t))
        y = pn + ((t - 1) / tn) * (pn - p)
        p, t, c_prev = pn, tn, cn
    return p
