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


def extract_phases(J, niter=2):
    """Phase-only reduction of solutions with a common unitary ambiguity
    (extract_phases, manifold_average.c:400): resolve the ambiguity by
    Jacobi joint diagonalization (Cardoso-Souloumiac sweeps: for each
    off-diagonal position, the Givens rotation maximizing the summed
    diagonal energy comes from the top eigenvector of a real 3x3 H =
    sum Re(h h^H), h = conj([a00-a11, a01+a10, i(a10-a01)])), then keep
    only exp(i*arg) of the diagonal.

    J: [N, 2, 2] complex -> [N, 2, 2] with unit-modulus diagonal, zero
    off-diagonals."""
    Jw = J.clone().to(torch.complex128)
    N = Jw.shape[0]
    for _ in range(niter):
        for which in (0, 1):
            a00, a01 = Jw[:, 0, 0], Jw[:, 0, 1]
            a10, a11 = Jw[:, 1, 0], Jw[:, 1, 1]
            if which == 0:
                h = torch.stack([a00 - a11, a01 + a10,
                                 1j * (a10 - a01)], dim=1).conj()
            else:
                h = torch.stack([a11 - a00, a10 + a01,
                                 1j * (a01 - a10)], dim=1).conj()
            H = (h.unsqueeze(2) * h.conj().unsqueeze(1)).sum(dim=0).real
            w, V = torch.linalg.eigh(H)
            z = V[:, -1]
            if float(z[0]) >= 0.0:
                c = torch.sqrt(0.5 + 0.5 * z[0]).to(torch.complex128)
                s = 0.5 * (z[1] - 1j * z[2]) / c
            else:
                c = torch.sqrt(0.5 - 0.5 * z[0]).to(torch.complex128)
                s = 0.5 * (-z[1] + 1j * z[2]) / c
            G = torch.stack([torch.stack([c, -s]),
                             torch.stack([s.conj(), c.conj()])])
            Jw = Jw @ G.conj().T
    out = torch.zeros_like(Jw)
    for d in (0, 1):
        ph = torch.angle(Jw[:, d, d])
        out[:, d, d] = torch.complex(torch.cos(ph), torch.sin(ph))
    return out.to(J.dtype)
