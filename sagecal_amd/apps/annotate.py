"""annotate — DS9 region file from a sky model + cluster file.

Re-implements src/buildsky/annotate.py (annotate_lsm_sky): one colored
circle (point) / ellipse (gaussian/disc/ring/shapelet) per source,
cluster-colored, with optional labels — without the casacore measures
dependency (no az/el culling; regions are sky-frame fk5).
"""
import argparse
import sys

import numpy as np

from .. import sky as skymod

_COLORS = ['yellow', 'cyan', 'green', 'red', 'magenta', 'blue', 'white',
           'orange']


def write_regions(out, groups, color=None, labels=False):
    """groups: list of (cluster_id, [Source, ...])."""
    out.write("# Region file format: DS9\n")
    out.write("global width=1 font=\"helvetica 10 normal\"\n")
    out.write("fk5\n")
    r2d = 180.0 / np.pi
    for k, (cid, members) in enumerate(groups):
        col = color or _COLORS[k % len(_COLORS)]
        for src in members:
            ra = float(src.ra) * r2d
            dec = float(src.dec) * r2d
            txt = f" # color={col}" + (
                f" text={{{src.name}}}" if labels else "")
            if getattr(src, 'stype', 0) == 0:
                out.write(f"circle({ra:.6f},{dec:.6f},30\")" + txt + "\n")
            else:
                eX = max(float(src.eX) * r2d * 3600, 30.0)
                eY = max(float(src.eY) * r2d * 3600, 30.0)
                pa = float(src.eP) * r2d
                out.write(f"ellipse({ra:.6f},{dec:.6f},{eX:.1f}\","
                          f"{eY:.1f}\",{pa:.1f})" + txt + "\n")


def main(argv=None):
    ap = argparse.ArgumentParser(prog='annotate')
    ap.add_argument('-s', dest='sky', required=True)
    ap.add_argument('-c', dest='cluster', required=True)
    ap.add_argument('-o', dest='outfile', required=True,
                    help='output DS9 region file')
    ap.add_argument('-i', dest='clid', type=int, default=None,
                    help='only this cluster id')
    ap.add_argument('-F', dest='format', type=int, default=0)
    ap.add_argument('-C', dest='color', default=None)
    ap.add_argument('-n', dest='labels', action='store_true',
                    help='label regions with source names')
    args = ap.parse_args(argv)
    sources = skymod.read_sky_model(args.sky, fmt=args.format)
    clist = skymod.read_cluster_file(args.cluster)
    groups = []
    for cid, _, names in clist:
        if args.clid is not None and cid != args.clid:
            continue
        members = [sources[n] for n in names if n in sources]
        for n, src in zip(names, members):
            src.name = n
        groups.append((cid, members))
    with open(args.outfile, 'w') as f:
        write_regions(f, groups, color=args.color, labels=args.labels)
    n = sum(len(m) for _, m in groups)
    print(f"annotate: {n} regions -> {args.outfile}")
    return 0


if __name__ == '__main__':
    sys.exit(main())
