"""buildsky — construct a sky model + cluster file from a FITS image.

Re-implements the workflow of /root/reference/src/buildsky (main.c:61,
buildsky.c, fitpixels.c, cluster.c): island detection above a threshold
(or an external Duchamp-style mask), per-island MULTI-component Gaussian
fitting with model-order selection (fit k = 1..maxfits components by LM,
pick the order minimizing AIC/MDL — fitpixels.c/fitmultipixels.c
semantics), PSF-aware point/extended classification, merging of
components closer than rd*(bmaj+bmin)/2 (main.c -c), and weighted K-means
clustering of the fitted sources into calibration directions (cluster.c)
— producing the LSM sky file and cluster file the calibrator consumes
(the reference's create_clusters.py role included).
"""
import argparse
import os
import sys

import numpy as np

from ..utils import fits as fitsio

FWHM = 2.0 * np.sqrt(2.0 * np.log(2.0))


def find_islands(img, threshold=None, mask=None, nsigma=5.0):
    """Island pixel lists (y, x).

    With a Duchamp mask: pixels are grouped by their mask VALUE — each
    distinct nonzero value is one Duchamp object — matching the
    reference's id-keyed island table (buildsky.c read of the mask file);
    connectivity relabeling would split disconnected objects and merge
    touching ones. Without a mask: connected components above the
    threshold (default 5-sigma via MAD)."""
    from scipy import ndimage
    islands = []
    if mask is not None:
        mi = np.rint(np.asarray(mask)).astype(np.int64)
        for val in np.unique(mi):
            if val == 0:
                continue
            ys, xs = np.nonzero(mi == val)
            if len(ys) >= 1:
                islands.append((ys, xs))
        return islands
    if threshold is None:
        med = np.median(img)
        mad = np.median(np.abs(img - med)) * 1.4826 + 1e-12
        threshold = med + nsigma * mad
    sel = img > threshold
    lab, nl = ndimage.label(sel)
    for i in range(1, nl + 1):
        ys, xs = np.nonzero(lab == i)
        if len(ys) >= 1:
            islands.append((ys, xs))
    return islands


def convex_hull(ys, xs):
    """Convex hull of island pixels as an [Nh, 2] (y, x) vertex array
    (the role of construct_hull, /root/reference/src/buildsky/hull.c:1-521
    — computed with scipy instead of the reference's Graham stack)."""
    pts = np.stack([ys, xs], axis=1).astype(float)
    if len(pts) < 3 or np.ptp(pts[:, 0]) < 1e-9 or             np.ptp(pts[:, 1]) < 1e-9:
        return None
    from scipy.spatial import ConvexHull, QhullError
    try:
        h = ConvexHull(pts)
    except QhullError:
        return None
    return pts[h.vertices]


def outside_hull_distance(hull, y, x):
    """0 inside the hull; distance beyond the nearest edge outside
    (smooth penalty form of inside_hull, fitpixels.c:533-535)."""
    if hull is None:
        return 0.0
    # signed distances to edges; orientation normalized so the hull
    # centroid (always interior for a convex polygon) is "inside"
    nh = len(hull)
    cy, cx = hull[:, 0].mean(), hull[:, 1].mean()
    d = -np.inf
    for i in range(nh):
        y0, x0 = hull[i]
        y1, x1 = hull[(i + 1) % nh]
        ex, ey = x1 - x0, y1 - y0
        nrm = np.hypot(ex, ey) + 1e-30
        cr_c = (ex * (cy - y0) - ey * (cx - x0))
        sgn = 1.0 if cr_c > 0 else -1.0
        cr = sgn * (ex * (y - y0) - ey * (x - x0)) / nrm
        d = max(d, -cr)
    return max(d, 0.0)


def _moments(f, ys, xs):
    ftot = f.sum()
    cy = (f * ys).sum() / ftot
    cx = (f * xs).sum() / ftot
    vy = (f * (ys - cy) ** 2).sum() / ftot
    vx = (f * (xs - cx) ** 2).sum() / ftot
    vxy = (f * (xs - cx) * (ys - cy)).sum() / ftot
    return ftot, cy, cx, vy, vx, vxy


def _gauss_eval(theta, ys, xs, k):
    """Sum of k elliptical Gaussians; theta = k x [A, cy, cx, sy, sx, pa]."""
    out = np.zeros(len(ys))
    for i in range(k):
        A, cy, cx, sy, sx, pa = theta[6 * i:6 * i + 6]
        sy, sx = abs(sy) + 1e-3, abs(sx) + 1e-3
        cp, sp = np.cos(pa), np.sin(pa)
        dy, dx = ys - cy, xs - cx
        yr = cp * dy - sp * dx
        xr = sp * dy + cp * dx
        out += A * np.exp(-0.5 * ((yr / sy) ** 2 + (xr / sx) ** 2))
    return out


def fit_island_multi(img, ys, xs, maxfits=10, criterion='aic'):
    """Fit 1..maxfits elliptical-Gaussian components to one island by LM
    (scipy least_squares plays the role of clmfit_nocuda.c) and select
    the model order by AIC (2p + n ln(rss/n)) or MDL ((p/2) ln n + ...)
    — fitmultipixels.c model selection. Component centres are constrained
    to the island's convex hull by a smooth penalty (the reference's
    inside_hull INFINITY_L penalty, fitpixels.c:533-535 + hull.c).
    Returns list of component dicts in PIXEL units (flux, cy, cx, sy,
    sx, pa)."""
    from scipy.optimize import least_squares
    hull = convex_hull(ys, xs)

    def _hull_pen(th, k):
        return [100.0 * outside_hull_distance(hull, th[6 * i + 1],
                                              th[6 * i + 2])
                for i in range(k)]
    f = img[ys, xs].astype(float)
    ysf, xsf = ys.astype(float), xs.astype(float)
    n = len(f)
    if n < 7:
        # too few pixels for an LM fit: moment estimate (point-like)
        ftot, cy, cx, vy, vx, _ = _moments(np.abs(f) + 1e-12, ysf, xsf)
        s0 = float(np.sqrt(max(0.5 * (vy + vx), 1e-4)))
        return [dict(flux=float(f.sum()), cy=float(cy), cx=float(cx),
                     sy=s0, sx=s0, pa=0.0)]
    best = None
    best_score = np.inf
    theta = []
    resid = f.copy()
    rss = None
    kmax = max(1, min(maxfits, n // 7))
    for k in range(1, kmax + 1):
        # seed component k at the residual peak
        j = int(np.argmax(np.abs(resid)))
        ftot, cy, cx, vy, vx, _ = _moments(np.abs(resid) + 1e-12, ysf, xsf)
        s0 = max(0.7, np.sqrt(max(vy + vx, 1e-2) / (2.0 * k)))
        theta = list(theta) + [float(resid[j]), float(ysf[j]), float(xsf[j]),
                               s0, s0, 0.0]
        sol = least_squares(
            lambda th: np.concatenate(
                [_gauss_eval(th, ysf, xsf, k) - f, _hull_pen(th, k)]),
            theta, method='lm', max_nfev=200 * k)
        theta = list(sol.x)
        resid = f - _gauss_eval(sol.x, ysf, xsf, k)
        prev_rss = rss if k > 1 else None
        rss = float((resid ** 2).sum()) + 1e-300
        p = 6 * k
        if criterion == 'mdl':
            score = 0.5 * p * np.log(n) + 0.5 * n * np.log(rss / n)
        else:
            score = 2.0 * p + n * np.log(rss / n)
        # accept a higher order only on a criterion win AND a material
        # (>10%) rss drop AND physically sensible components — pure-noise
        # components are sub-pixel spikes of negligible flux and can fool
        # AIC/rss gates on small islands
        fsum = float(np.abs(f).sum())
        sane = all(
            max(abs(theta[6 * i + 3]), abs(theta[6 * i + 4])) >= 0.5
            and abs(2.0 * np.pi * theta[6 * i]
                    * (abs(theta[6 * i + 3]) + 1e-3)
                    * (abs(theta[6 * i + 4]) + 1e-3)) >= 0.02 * fsum
            for i in range(k))
        if (score < best_score - 1e-9 and sane
                and (prev_rss is None or rss < 0.9 * prev_rss)):
            best_score = score
            best = list(theta)
        else:
            break       # adding a component no longer pays: stop early
    comps = []
    for i in range(len(best) // 6):
        A, cy, cx, sy, sx, pa = best[6 * i:6 * i + 6]
        sy, sx = abs(sy) + 1e-3, abs(sx) + 1e-3
        comps.append(dict(flux=float(2.0 * np.pi * A * sy * sx),
                          cy=float(cy), cx=float(cx), sy=float(sy),
                          sx=float(sx), pa=float(pa % np.pi)))
    return comps


def fit_island(img, hdr, ys, xs, maxfits=10, criterion='aic',
               psf_pix=0.0):
    """Fit one island into one or more sky-model sources. Components whose
    fitted extent is at or below the PSF width (or sub-pixel when no PSF
    is given) are classified as points (fitpixels.c point-vs-gaussian)."""
    d2r = np.pi / 180.0
    pscale = abs(hdr['CDELT1']) * d2r           # rad/pixel
    out = []
    for c in fit_island_multi(img, ys, xs, maxfits, criterion):
        ra, dec = fitsio.pix_to_radec(hdr, c['cx'], c['cy'])
        sig_maj = max(c['sy'], c['sx'])
        sig_min = min(c['sy'], c['sx'])
        unresolved = sig_maj <= max(0.7, 1.05 * psf_pix / FWHM)
        if unresolved:
            out.append(dict(stype='P', ra=float(ra), dec=float(dec),
                            flux=c['flux'], eX=0.0, eY=0.0, eP=0.0))
        else:
            out.append(dict(stype='G', ra=float(ra), dec=float(dec),
                            flux=c['flux'],
                            eX=float(sig_maj * pscale * FWHM),
                            eY=float(sig_min * pscale * FWHM),
                            eP=float(c['pa'])))
    return out


def merge_close(srcs, rd, beam_rad):
    """Merge sources closer than rd*(bmaj+bmin)/2 radians (main.c -c):
    flux-weighted position, summed flux, widest extent kept."""
    if rd <= 0 or beam_rad <= 0 or len(srcs) < 2:
        return srcs
    srcs = sorted(srcs, key=lambda s: -s['flux'])
    out = []
    used = [False] * len(srcs)
    lim = rd * beam_rad
    for i, s in enumerate(srcs):
        if used[i]:
            continue
        grp = [s]
        used[i] = True
        for j in range(i + 1, len(srcs)):
            if used[j]:
                continue
            t = srcs[j]
            d = np.hypot((t['ra'] - s['ra']) * np.cos(s['dec']),
                         t['dec'] - s['dec'])
            if d < lim:
                grp.append(t)
                used[j] = True
        ft = sum(g['flux'] for g in grp)
        m = dict(grp[0])
        m['flux'] = ft
        m['ra'] = sum(g['flux'] * g['ra'] for g in grp) / ft
        m['dec'] = sum(g['flux'] * g['dec'] for g in grp) / ft
        m['eX'] = max(g['eX'] for g in grp)
        m['eY'] = max(g['eY'] for g in grp)
        out.append(m)
    return out


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


def fit_spectra(imgs, freqs, hdr, islands, maxfits, criterion, psf_pix,
                freq0=None):
    """Multi-frequency island fitting (buildmultisky.c): positions and
    shapes come from a joint fit of the MEAN image; per-plane fluxes are
    then linear least squares with the shapes FIXED, and each component
    gets a spectral index from a log-log fit. Returns sources with
    'si' filled and flux referenced to freq0 (default: mean freq)."""
    freqs = np.asarray(freqs, dtype=float)
    f0 = float(freq0 or freqs.mean())
    mean_img = np.mean(imgs, axis=0)
    out = []
    for ys, xs in islands:
        comps = fit_island_multi(mean_img, ys, xs, maxfits, criterion)
        if not comps:
            continue
        # design matrix of unit-amplitude components on this island
        ysf, xsf = ys.astype(float), xs.astype(float)
        cols = []
        for c in comps:
            th = [1.0, c['cy'], c['cx'], c['sy'], c['sx'], c['pa']]
            cols.append(_gauss_eval(th, ysf, xsf, 1))
        A = np.stack(cols, axis=1)                    # [npix, k]
        AtA = A.T @ A + 1e-12 * np.eye(A.shape[1])
        fluxes = []
        for img in imgs:                               # per-plane amps
            amp = np.linalg.solve(AtA, A.T @ img[ys, xs])
            fluxes.append(amp * np.array(
                [2.0 * np.pi * c['sy'] * c['sx'] for c in comps]))
        fluxes = np.stack(fluxes)                      # [F, k]
        lf = np.log(freqs / f0)
        for k, c in enumerate(comps):
            fk = fluxes[:, k]
            pos = fk > 1e-12
            if pos.sum() >= 2 and len(freqs) >= 2:
                # weighted linear fit of log flux vs log freq
                w = fk[pos]
                X = np.stack([np.ones(pos.sum()), lf[pos]], axis=1)
                sol, *_ = np.linalg.lstsq(X * w[:, None],
                                          np.log(fk[pos]) * w, rcond=None)
                s0, si = float(np.exp(sol[0])), float(sol[1])
            else:
                s0, si = float(fk.mean()), 0.0
            ra, dec = fitsio.pix_to_radec(hdr, c['cx'], c['cy'])
            d2r = np.pi / 180.0
            pscale = abs(hdr['CDELT1']) * d2r
            sig_maj, sig_min = max(c['sy'], c['sx']), min(c['sy'], c['sx'])
            unresolved = sig_maj <= max(0.7, 1.05 * psf_pix / FWHM)
            out.append(dict(
                stype='P' if unresolved else 'G', ra=float(ra),
                dec=float(dec), flux=s0, si=si,
                eX=0.0 if unresolved else float(sig_maj * pscale * FWHM),
                eY=0.0 if unresolved else float(sig_min * pscale * FWHM),
                eP=0.0 if unresolved else float(c['pa'])))
    return out, f0


def main(argv=None):
    ap = argparse.ArgumentParser(prog='buildsky')
    ap.add_argument('-f', dest='fits', help='FITS image')
    ap.add_argument('-d', dest='fitsdir',
                    help='directory of per-frequency FITS planes '
                         '(multi-frequency fit with spectral indices, '
                         'buildmultisky.c)')
    ap.add_argument('-m', dest='mask', help='mask FITS (Duchamp style)')
    ap.add_argument('-t', dest='threshold', type=float,
                    help='island threshold (default: 5 sigma)')
    ap.add_argument('-Q', dest='nclusters', type=int, default=4,
                    help='number of direction clusters (reference -k)')
    ap.add_argument('-l', dest='maxfits', type=int, default=10,
                    help='max components attempted per island')
    ap.add_argument('-o', dest='model_order', type=int, default=1)
    ap.add_argument('-C', dest='criterion', choices=['aic', 'mdl'],
                    default='aic', help='model-order criterion')
    ap.add_argument('-a', dest='bmaj', type=float, default=0.0,
                    help='PSF major axis (arcsec)')
    ap.add_argument('-b', dest='bmin', type=float, default=0.0,
                    help='PSF minor axis (arcsec)')
    ap.add_argument('-p', dest='bpa', type=float, default=0.0,
                    help='PSF position angle (deg)')
    ap.add_argument('-M', dest='merge_rd', type=float, default=0.0,
                    help='merge components closer than rd*(bmaj+bmin)/2 '
                         '(reference -c; renamed: -c is the cluster file '
                         'here)')
    ap.add_argument('-w', dest='sidelobe_cut', type=float, default=0.0,
                    help='drop islands whose peak is below this')
    ap.add_argument('-N', dest='negative', action='store_true',
                    help='fit negative flux instead of positive')
    ap.add_argument('-q', dest='rescale', type=int, default=0,
                    help='1: scale model fluxes to island total flux')
    ap.add_argument('-s', dest='outsky', help='output sky file')
    ap.add_argument('-c', dest='outcluster', help='output cluster file')
    args = ap.parse_args(argv)
    if not args.fits and not args.fitsdir:
        ap.error('need -f image.fits or -d fits_directory')

    if args.fitsdir:
        import glob as _glob
        files = sorted(_glob.glob(os.path.join(args.fitsdir, '*.fits')))
        if len(files) < 1:
            print('buildsky: no FITS in directory', file=sys.stderr)
            return 1
        planes, freqs = [], []
        hdr = None
        for fn in files:
            im, h = fitsio.read_fits_image(fn)
            planes.append(-im if args.negative else im)
            freqs.append(float(h.get('RESTFRQ', 150e6)))
            hdr = hdr or h
        img = np.mean(planes, axis=0)
    else:
        img, hdr = fitsio.read_fits_image(args.fits)
        if args.negative:
            img = -img
        planes, freqs = None, None
    mask = None
    if args.mask:
        mask, _ = fitsio.read_fits_image(args.mask)
    islands = find_islands(img, args.threshold, mask)
    d2r = np.pi / 180.0
    psf_pix = 0.0
    beam_rad = 0.0
    if args.bmaj > 0:
        as2rad = d2r / 3600.0
        pscale = abs(hdr['CDELT1']) * d2r
        psf_pix = args.bmaj * as2rad / pscale
        beam_rad = 0.5 * (args.bmaj + args.bmin) * as2rad
    if args.sidelobe_cut > 0:
        islands = [(ys, xs) for ys, xs in islands
                   if img[ys, xs].max() >= args.sidelobe_cut]
    if planes is not None:
        srcs, f0 = fit_spectra(planes, freqs, hdr, islands, args.maxfits,
                               args.criterion, psf_pix)
        if args.negative:
            for s_ in srcs:
                s_['flux'] = -s_['flux']
        freq = f0
        _finish(args, srcs, freq)
        return 0
    srcs = []
    for ys, xs in islands:
        comps = fit_island(img, hdr, ys, xs, args.maxfits, args.criterion,
                           psf_pix)
        if args.rescale:
            ftot = float(img[ys, xs].sum())
            fsum = sum(c['flux'] for c in comps)
            if fsum > 0:
                for c in comps:
                    c['flux'] *= ftot / fsum
        srcs.extend(comps)
    srcs = [s for s in srcs if s['flux'] > 0]
    if args.merge_rd > 0:
        srcs = merge_close(srcs, args.merge_rd, beam_rad)
    if not srcs:
        print("buildsky: no sources found", file=sys.stderr)
        return 1
    if args.negative:
        for s in srcs:
            s['flux'] = -s['flux']
    freq = hdr.get('RESTFRQ', 150e6)
    _finish(args, srcs, freq)
    return 0


def _finish(args, srcs, freq):
    outsky = args.outsky or (args.fits or args.fitsdir) + '.sky.txt'
    outcl = args.outcluster or outsky + '.cluster'
    names = []
    with open(outsky, 'w') as f:
        f.write("# name h m s d m s I Q U V si RM eX eY eP f0\n")
        for i, s in enumerate(srcs):
            name = f"{s['stype']}{i}C{i}"
            names.append(name)
            si = s.get('si', 0.0)
            f.write(f"{name} {_fmt_radec(s['ra'], s['dec'])} "
                    f"{s['flux']:.6f} 0 0 0 {si:.6f} 0 "
                    f"{s['eX']:.8e} {s['eY']:.8e} {s['eP']:.8e} {freq}\n")
    ras = np.array([s['ra'] for s in srcs])
    decs = np.array([s['dec'] for s in srcs])
    w = np.array([abs(s['flux']) for s in srcs])
    assign, cent = weighted_kmeans(ras, decs, w, args.nclusters)
    with open(outcl, 'w') as f:
        f.write("# cluster_id chunk_size source...\n")
        for q in sorted(set(assign)):
            members = [names[i] for i in range(len(srcs)) if assign[i] == q]
            f.write(f"{q + 1} 1 " + ' '.join(members) + "\n")
    print(f"buildsky: {len(srcs)} sources -> {outsky}, "
          f"{len(set(assign))} clusters -> {outcl}")


if __name__ == '__main__':
    sys.exit(main())
