"""create_clusters — build a cluster file from an existing sky model.

Re-implements src/buildsky/create_clusters.py: flux-weighted K-means of
the sky-model sources into Q directions (optionally hierarchical when Q
is negative, matching buildsky -k), writing the `cluster_id chunk_size
sources...` file the calibrator consumes.
"""
import argparse
import sys

import numpy as np

from .. import sky as skymod
from .buildsky import weighted_kmeans


def main(argv=None):
    ap = argparse.ArgumentParser(prog='create_clusters')
    ap.add_argument('-s', dest='sky', required=True, help='LSM sky model')
    ap.add_argument('-c', dest='outcluster', required=True)
    ap.add_argument('-Q', dest='nclusters', type=int, required=True,
                    help='number of clusters (negative: hierarchical '
                         '2-stage split like buildsky -k < 0)')
    ap.add_argument('-t', dest='nchunk', type=int, default=1,
                    help='hybrid chunk size written per cluster')
    ap.add_argument('-F', dest='format', type=int, default=0)
    args = ap.parse_args(argv)
    sources = skymod.read_sky_model(args.sky, fmt=args.format)
    names = list(sources)
    ras = np.array([sources[n].ra for n in names])
    decs = np.array([sources[n].dec for n in names])
    w = np.array([abs(sources[n].sI) for n in names])
    Q = args.nclusters
    if Q < 0:
        # hierarchical: coarse split into sqrt(|Q|) then refine
        Q = abs(Q)
        coarse = max(1, int(np.sqrt(Q)))
        a0, _ = weighted_kmeans(ras, decs, w, coarse)
        assign = np.zeros(len(names), dtype=int)
        nxt = 0
        for g in sorted(set(a0)):
            sel = np.nonzero(a0 == g)[0]
            sub_q = max(1, round(Q * len(sel) / len(names)))
            a1, _ = weighted_kmeans(ras[sel], decs[sel], w[sel], sub_q,
                                    seed=g + 2)
            for u in sorted(set(a1)):
                assign[sel[a1 == u]] = nxt
                nxt += 1
    else:
        assign, _ = weighted_kmeans(ras, decs, w, Q)
    with open(args.outcluster, 'w') as f:
        f.write("# cluster_id chunk_size source...\n")
        for q in sorted(set(assign)):
            members = [names[i] for i in range(len(names))
                       if assign[i] == q]
            f.write(f"{q + 1} {args.nchunk} " + ' '.join(members) + "\n")
    print(f"create_clusters: {len(names)} sources -> "
          f"{len(set(assign))} clusters")
    return 0


if __name__ == '__main__':
    sys.exit(main())
