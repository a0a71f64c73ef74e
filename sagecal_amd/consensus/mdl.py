"""Minimum-description-length polynomial order selection.

Re-implements /root/reference/src/lib/Dirac/mdl.c
minimum_description_length (Dirac.h:593-610): for each candidate Npoly in
[Kstart, Kfinish], fit the consensus polynomial to the per-band (rho J)
solutions, compute the residual sum of squares, and score
  AIC = F log(RSS/F) + 2 Npoly
  MDL = F/2 log(RSS/F) + Npoly/2 log(F)        (mdl.c:233-235)
Returns the Npoly minimizing MDL (reference prints both).
"""
import math

import torch

from . import poly


def minimum_description_length(J_bands, rho, freqs, freq0, weight=None,
                               polytype=0, Kstart=1, Kfinish=5):
    """J_bands: [F, M, K] complex (per-band, per-cluster flattened rho*J
    solutions); rho: [M]; weight: [F] (flag-ratio weights)."""
    F, M, K = J_bands.shape
    if weight is None:
        weight = torch.ones(F, dtype=torch.float64)
    scores = {}
    for Npoly in range(Kstart, min(Kfinish, F) + 1):
        ptype = 1 if Npoly == 1 else polytype
        B = poly.setup_polynomials(freqs, freq0, Npoly, ptype)
        rho_mf = rho[:, None].expand(-1, F) * weight[None, :]
        Bii = poly.find_prod_inverse(B, rho_mf)
        # z accumulator: sum_f w_f B_f (x) (rho J_f)
        acc = torch.zeros(M, Npoly, K, dtype=J_bands.dtype)
        for f in range(F):
            contrib = weight[f] * J_bands[f]
            for p in range(Npoly):
                acc[:, p] += float(B[f, p]) * contrib
        Z = poly.update_global_z(acc, Bii)
        rss = 0.0
        for f in range(F):
            BZ = torch.einsum('p,mpk->mk', B[f].to(Z.dtype), Z)
            r = J_bands[f] - float(weight[f]) * BZ
            rss += float((r.abs() ** 2).sum())
        rss = max(rss, 1e-300)
        aic = F * math.log(rss / F) + 2.0 * Npoly
        mdl = 0.5 * F * math.log(rss / F) + 0.5 * Npoly * math.log(F)
        scores[Npoly] = (mdl, aic)
    best = min(scores, key=lambda k: scores[k][0])
    return best, scores
