"""Manifold averaging of Jones solutions across frequencies.

Re-implements /root/reference/src/lib/Dirac/manifold_average.c:
calculate_manifold_average (:204) with the Procrustes projection
(project_procrustes_block :346): solutions at different bands differ by a
per-band unitary ambiguity J_f ~ J U_f; the average first aligns every band
to a reference via the polar/Procrustes factor from the SVD of J_f^H J_ref,
then means. calculate_manifold_average_projectback (:808) returns each
band's aligned-average projected back to its own frame.
"""
import torch


def manifold_average(J_bands, niter=2):
    """Average [F, N, 2, 2] complex Jones over bands up to unitary
    ambiguity. Returns (Javg [N,2,2], U [F,2,2] per-band aligners).

    Algorithm (manifold_average.c:204-345): start from band 0; for each
    band find the 2x2 unitary U_f = polar(J_f^H Jref) stacked over
    stations; average aligned solutions; iterate."""
    F, N = J_bands.shape[0], J_bands.shape[1]
    Javg = J_bands[0].clone()
    U = torch.eye(2, dtype=J_bands.dtype,
                  device=J_bands.device).expand(F, 2, 2).clone()
    for _ in range(niter):
        aligned = []
        for f in range(F):
            # stack stations: solve one common 2x2 unitary per band
            A = sum(J_bands[f, s].conj().T @ Javg[s] for s in range(N))
            Uf = polar_unitary(A)
            U[f] = Uf
            aligned.append(J_bands[f] @ Uf)
        Javg = torch.stack(aligned).mean(dim=0)
    return Javg, U


def polar_unitary(A):
    """Closest unitary to 2x2 complex A: U = W V^H with A = W S V^H."""
    W, S, Vh = torch.linalg.svd(A)
    return W @ Vh


def manifold_average_projectback(J_bands, niter=2):
    """Each band replaced by the aligned average projected back to the
    band's own frame (calculate_manifold_average_projectback:808):
    J_f <- Javg U_f^H."""
    Javg, U = manifold_average(J_bands, niter)
    out = torch.stack([Javg @ U[f].conj().T
                       for f in range(J_bands.shape[0])])
    return out, Javg
