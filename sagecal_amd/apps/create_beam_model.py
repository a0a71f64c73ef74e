"""create_beam_model — fit measured element patterns into coefficients.

Re-implements scripts/beam_models/create_header.py: given sampled
voltage patterns on a (theta, phi, frequency) grid — theta.npy, phi.npy,
frequency.npy (1-D) and etheta.npy / ephi.npy complex arrays of shape
[n_freq, n_theta, n_phi] — least-squares fit each frequency onto the
spherical-harmonic mode stack and write an `ElementCoeffs` .npz that
`beams.element_beam` consumes (the reference writes a C header instead;
the math contract is the same). The Y dipole is the X pattern rotated
90 deg in azimuth (crossed-dipole convention).
"""
import argparse
import os
import sys

import numpy as np

from .. import beams


def fit_patterns(theta, phi, freqs, etheta, ephi, n0):
    """Least-squares spherical-harmonic fit per frequency.
    theta/phi in RADIANS (grid axes); patterns [Nf, n_th, n_ph]."""
    th, ph = np.meshgrid(theta, phi, indexing='ij')
    B = beams.sharmonic_basis(th.ravel(), ph.ravel(), n0)
    Bp = np.linalg.pinv(B)
    nm = beams.n_modes(n0)
    Nf = len(freqs)
    ct_x = np.zeros((Nf, nm), dtype=complex)
    cp_x = np.zeros((Nf, nm), dtype=complex)
    ct_y = np.zeros((Nf, nm), dtype=complex)
    cp_y = np.zeros((Nf, nm), dtype=complex)
    # Y dipole: X rotated 90 deg in phi (evaluate on the rolled grid)
    shift = int(round((np.pi / 2) / (phi[1] - phi[0]))) \
        if len(phi) > 1 else 0
    for fi in range(Nf):
        ct_x[fi] = Bp @ etheta[fi].ravel()
        cp_x[fi] = Bp @ ephi[fi].ravel()
        et_y = np.roll(etheta[fi], -shift, axis=1)
        ep_y = np.roll(ephi[fi], -shift, axis=1)
        ct_y[fi] = Bp @ et_y.ravel()
        cp_y[fi] = Bp @ ep_y.ravel()
    return beams.ElementCoeffs(n0, freqs, ct_x, cp_x, ct_y, cp_y)


def main(argv=None):
    ap = argparse.ArgumentParser(prog='create_beam_model')
    ap.add_argument('-d', dest='datadir', default='.',
                    help='directory holding theta/phi/frequency/etheta/'
                         'ephi .npy files')
    ap.add_argument('--order', dest='n0', type=int, default=7)
    ap.add_argument('--degrees', action='store_true',
                    help='theta/phi files are in degrees (the reference '
                         'convention)')
    ap.add_argument('-o', dest='output', default='elementcoeff.npz')
    args = ap.parse_args(argv)
    ld = lambda n: np.load(os.path.join(args.datadir, n))
    theta, phi = ld('theta.npy'), ld('phi.npy')
    if args.degrees:
        theta = np.deg2rad(theta)
        phi = np.deg2rad(phi)
    co = fit_patterns(theta, phi, ld('frequency.npy'),
                      ld('etheta.npy'), ld('ephi.npy'), args.n0)
    co.save(args.output)
    print(f"create_beam_model: order {args.n0}, "
          f"{len(co.freqs)} freqs -> {args.output}")
    return 0


if __name__ == '__main__':
    sys.exit(main())
