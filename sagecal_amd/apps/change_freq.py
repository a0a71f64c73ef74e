"""change_freq — rewrite an MS copy's frequency metadata.

Re-implements test/Calibration/Change_freq.py: the documented recipe for
exercising multi-band consensus without a cluster is to duplicate one MS
and shift each copy's frequencies, creating fake sub-bands
(test/Calibration/README.md steps 1-4).
"""
import argparse
import sys

import numpy as np


def main(argv=None):
    ap = argparse.ArgumentParser(prog='change_freq')
    ap.add_argument('-d', dest='ms', required=True, help='NpzMS path')
    ap.add_argument('-f', dest='freq', type=float, required=True,
                    help='new centre frequency (Hz)')
    ap.add_argument('-o', dest='out',
                    help='output path (default: in place)')
    args = ap.parse_args(argv)
    z = dict(np.load(args.ms))
    freqs = np.asarray(z['freqs'], dtype=float)
    shift = args.freq - freqs.mean()
    z['freqs'] = freqs + shift
    np.savez_compressed(args.out or args.ms, **z)
    print(f"change_freq: {freqs.mean():.4g} -> {args.freq:.4g} Hz "
          f"({args.out or args.ms})")
    return 0


if __name__ == '__main__':
    sys.exit(main())
