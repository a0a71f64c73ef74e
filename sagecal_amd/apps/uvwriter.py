"""uvwriter — recompute and rewrite the UVW column of an MS.

The reference tool (/root/reference/src/uvwriter/uvwriter.cpp:46-55)
rewrites MS UVW for lunar reference frames via CSPICE. This analog
implements both frames natively: the standard earth path from array
geometry + phase centre, and `--frame lunar` via the WGCCRE analytic
MOON_ME orientation in coords.py (no SPICE kernels needed; ~150 m-class
frame accuracy, ample for UVW)."""
import argparse
import sys

import numpy as np

from .. import msdata


def main(argv=None):
    ap = argparse.ArgumentParser(prog='uvwriter')
    ap.add_argument('-d', dest='ms', required=True, help='NpzMS path')
    ap.add_argument('--lat', type=float, default=0.92)
    ap.add_argument('--tdelta', type=float, default=None)
    ap.add_argument('--frame', choices=['earth', 'lunar'], default='earth',
                    help='earth: ENU + hour angle; lunar: WGCCRE MOON_ME '
                         'body-fixed stations -> J2000 projection')
    ap.add_argument('--jd0', type=float, default=2460000.5,
                    help='lunar frame: JD (TDB) of the first time slot')
    args = ap.parse_args(argv)
    ms = msdata.NpzMS(args.ms)
    td = args.tdelta or ms.tdelta
    pos = msdata.lofar_like_array(ms.N)   # placeholder geometry if none
    if 'pos_enu' in ms._z:
        pos = ms._z['pos_enu']
    p, q = ms.pairs[:, 0], ms.pairs[:, 1]
    us, vs, ws = [], [], []
    for t in range(ms.Ntime):
        if args.frame == 'lunar':
            from .. import coords
            # place ENU metres on the lunar surface at (lon=0.1, lat)
            lon0, lat0 = 0.1, args.lat
            sl, cl = np.sin(lon0), np.cos(lon0)
            sb, cb = np.sin(lat0), np.cos(lat0)
            east = np.array([-sl, cl, 0.0])
            north = np.array([-sb * cl, -sb * sl, cb])
            up = np.array([cb * cl, cb * sl, sb])
            pos_me = (coords.MOON_RADIUS * up
                      + pos[:, 0, None] * east + pos[:, 1, None] * north)
            jd = args.jd0 + (t + 0.5) * td / 86400.0
            uu, vv, ww = coords.lunar_uvw(pos_me, ms.ra0, ms.dec0, jd)
            uu, vv, ww = uu[None], vv[None], ww[None]
        else:
            ha = (t + 0.5) * td * 7.2921150e-5 - 0.2
            uu, vv, ww = msdata.enu_uvw(pos, args.lat, ha, ms.dec0)
        us.append((uu[:, p] - uu[:, q]).reshape(-1))
        vs.append((vv[:, p] - vv[:, q]).reshape(-1))
        ws.append((ww[:, p] - ww[:, q]).reshape(-1))
    ms._z['u'] = np.concatenate(us)
    ms._z['v'] = np.concatenate(vs)
    ms._z['w'] = np.concatenate(ws)
    ms.save()
    print(f"uvwriter: rewrote UVW for {ms.Ntime} slots ({args.frame})")
    return 0


if __name__ == '__main__':
    sys.exit(main())
