"""uvwriter — recompute and rewrite the UVW column of an MS.

The reference tool (/root/reference/src/uvwriter/uvwriter.cpp:46-55)
rewrites MS UVW for lunar reference frames via CSPICE. CSPICE is not
available in this environment; this analog recomputes geocentric UVW from
the array geometry + phase centre (the standard earth-frame path), with a
pluggable frame hook where a lunar ephemeris transform would slot in.
"""
import argparse
import sys

import numpy as np

from .. import msdata
from ..constants import C_LIGHT


def main(argv=None):
    ap = argparse.ArgumentParser(prog='uvwriter')
    ap.add_argument('-d', dest='ms', required=True, help='NpzMS path')
    ap.add_argument('--lat', type=float, default=0.92)
    ap.add_argument('--tdelta', type=float, default=None)
    ap.add_argument('--frame', choices=['earth'], default='earth',
                    help='reference frame (lunar requires an ephemeris '
                         'backend; hook in msdata.enu_uvw)')
    args = ap.parse_args(argv)
    ms = msdata.NpzMS(args.ms)
    td = args.tdelta or ms.tdelta
    pos = msdata.lofar_like_array(ms.N)   # placeholder geometry if none
    if 'pos_enu' in ms._z:
        pos = ms._z['pos_enu']
    p, q = ms.pairs[:, 0], ms.pairs[:, 1]
    us, vs, ws = [], [], []
    for t in range(ms.Ntime):
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
