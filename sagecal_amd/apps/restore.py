"""restore — render a sky model (optionally with solutions) into a FITS
image.

Re-implements /root/reference/src/restore (restore.c:892, readsky.c,
shapelet_lm.c): each sky-model source is painted onto the image grid —
points as delta pixels (optionally convolved with a restoring beam),
gaussians as elliptical profiles, shapelets via the image-plane
Gauss-Hermite basis (shapelet.py image_basis) — with fluxes optionally
corrected by the mean solution amplitude of their cluster (the -s
solutions option of the reference).
"""
import argparse
import math
import sys

import numpy as np

from ..utils import fits as fitsio
from .. import sky as skymod
from .. import shapelet as shmod


def convolve_with_psf(img, bmaj_pix, bmin_pix, bpa=0.0):
    """FFT-convolve an image with an elliptical-Gaussian restoring beam
    (the role of convolve_with_psf, /root/reference/src/restore/fft.c:201
    — FFT of the model times FFT of the PSF, normalized to unit peak so
    point fluxes stay peak-calibrated like the reference's delta-FFT
    normalization). bmaj/bmin are FWHM in pixels, bpa in radians."""
    ny, nx = img.shape
    yy, xx = np.mgrid[0:ny, 0:nx]
    cy, cx = ny // 2, nx // 2
    sy = max(bmaj_pix, 1e-3) / FWHM_SIG
    sx = max(bmin_pix, 1e-3) / FWHM_SIG
    cp, sp = math.cos(bpa), math.sin(bpa)
    u = cp * (yy - cy) - sp * (xx - cx)
    v = sp * (yy - cy) + cp * (xx - cx)
    psf = np.exp(-0.5 * ((u / sy) ** 2 + (v / sx) ** 2))
    F = np.fft.rfft2(np.fft.ifftshift(psf))
    return np.fft.irfft2(np.fft.rfft2(img) * F, s=img.shape).real


FWHM_SIG = 2.0 * math.sqrt(2.0 * math.log(2.0))


def render_shapelet_fft(xgrid_l, ygrid_m, n0, beta, modes, flux,
                        bmaj_pix, bmin_pix, bpa=0.0):
    """Shapelet source rendered via the FFT path (calculate_mode_vectors
    + convolve_with_psf, fft.c): evaluate the image-plane Gauss-Hermite
    basis on the full grid, sum modes, convolve with the restoring PSF
    in the Fourier domain."""
    ny, nx = xgrid_l.shape
    bas = shmod.image_basis(xgrid_l.ravel(), ygrid_m.ravel(), n0,
                            beta).numpy()
    img = flux * (bas @ modes).reshape(ny, nx)
    if bmaj_pix > 0:
        img = convolve_with_psf(img, bmaj_pix, bmin_pix or bmaj_pix, bpa)
    return img


def render(clusters, hdr, shape, gains=None, beam_fwhm_pix=0.0):
    ny, nx = shape
    img = np.zeros((ny, nx))
    yy, xx = np.mgrid[0:ny, 0:nx]
    d2r = math.pi / 180.0
    pscale = abs(hdr['CDELT1']) * d2r
    for ci, c in enumerate(clusters):
        gain = 1.0
        if gains is not None:
            gain = float(gains[ci])
        for si in range(c.nsrc):
            # source (l, m) -> pixel
            ra0 = hdr['CRVAL1'] * d2r
            dec0 = hdr['CRVAL2'] * d2r
            # recover ra/dec from cosines
            ll, mm = c.ll[si], c.mm[si]
            nn = c.nn1[si] + 1.0
            dec = math.asin(np.clip(mm * math.cos(dec0)
                                    + nn * math.sin(dec0), -1, 1))
            ra = ra0 + math.atan2(ll, nn * math.cos(dec0)
                                  - mm * math.sin(dec0))
            x, y = fitsio.radec_to_pix(hdr, ra, dec)
            flux = c.sI[si] * gain
            st = int(c.stype[si])
            if st == 1:     # gaussian
                sx = max(c.eX[si] / pscale, 0.5)
                sy = max(c.eY[si] / pscale, 0.5)
                cp, sp = math.cos(c.eP[si]), math.sin(c.eP[si])
                dx = xx - x
                dy = yy - y
                u = cp * dx - sp * dy
                v = sp * dx + cp * dy
                g = np.exp(-0.5 * ((u / sx) ** 2 + (v / sy) ** 2))
                img += flux * g / g.sum()
            elif st == 4 and c.shapelets:    # shapelet
                match = [sh for sh in c.shapelets if sh[0] == si]
                if match:
                    _, n0, beta, modes = match[0]
                    lpix = (xx - x) * pscale * np.sign(hdr['CDELT1'])
                    mpix = (yy - y) * pscale * np.sign(hdr['CDELT2'])
                    if beam_fwhm_pix > 0:
                        # FFT path (fft.c): basis + PSF convolution in
                        # the Fourier domain; beam applied here so the
                        # final image-wide convolution can skip shapelets
                        img += render_shapelet_fft(
                            lpix, mpix, n0, beta, modes,
                            flux * pscale * pscale, beam_fwhm_pix,
                            beam_fwhm_pix)
                        continue
                    bas = shmod.image_basis(lpix.ravel(), mpix.ravel(),
                                            n0, beta).numpy()
                    img += flux * (bas @ modes).reshape(ny, nx) \
                        * pscale * pscale
            else:
                ix, iy = int(round(x)), int(round(y))
                if 0 <= ix < nx and 0 <= iy < ny:
                    img[iy, ix] += flux
    if beam_fwhm_pix > 0:
        from scipy import ndimage
        sig = beam_fwhm_pix / (2 * math.sqrt(2 * math.log(2)))
        img = ndimage.gaussian_filter(img, sig)
    return img


def main(argv=None):
    ap = argparse.ArgumentParser(prog='restore')
    ap.add_argument('-f', dest='fits', required=True,
                    help='template FITS image (defines grid/WCS)')
    ap.add_argument('-s', dest='sky', required=True)
    ap.add_argument('-c', dest='cluster', required=True)
    ap.add_argument('-p', dest='solfile', help='solutions file: scale '
                    'cluster fluxes by mean |J|^2')
    ap.add_argument('-o', dest='out', required=True, help='output FITS')
    ap.add_argument('-F', dest='format', type=int, default=0)
    ap.add_argument('-b', dest='beam', type=float, default=0.0,
                    help='restoring beam FWHM in pixels')
    ap.add_argument('-a', dest='add', type=int, default=0,
                    help='1: add to template image')
    args = ap.parse_args(argv)
    img0, hdr = fitsio.read_fits_image(args.fits)
    d2r = math.pi / 180.0
    clusters = skymod.read_sky_cluster(
        args.sky, args.cluster, hdr['CRVAL1'] * d2r, hdr['CRVAL2'] * d2r,
        hdr.get('RESTFRQ', 150e6), fmt=args.format)
    gains = None
    if args.solfile:
        from .. import solutions
        hdr_s, tiles = solutions.read_solutions(args.solfile)
        if tiles:
            nch = [c.nchunk for c in clusters]
            J = solutions.reorder_read_tile(tiles[0], nch)
            gains = []
            off = 0
            for c in clusters:
                Jc = J[off:off + c.nchunk]
                # apparent unpolarized flux under J: I' = tr(J J^H)/2 I
                # (restore.c withsol scaling)
                gains.append(float((Jc.abs() ** 2)
                                   .sum(dim=(-1, -2)).mean() / 2.0))
                off += c.nchunk
    img = render(clusters, hdr, img0.shape, gains, args.beam)
    if args.add:
        img = img + img0
    fitsio.write_fits_image(args.out, img, crval=(hdr['CRVAL1'],
                                                  hdr['CRVAL2']),
                            cdelt=(hdr['CDELT1'], hdr['CDELT2']),
                            crpix=(hdr['CRPIX1'], hdr['CRPIX2']),
                            freq=hdr.get('RESTFRQ', 150e6))
    print(f"restore: wrote {args.out}")
    return 0


if __name__ == '__main__':
    sys.exit(main())
