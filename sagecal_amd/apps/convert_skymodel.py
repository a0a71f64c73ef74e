"""convert_skymodel — LSM <-> BBS sky-model conversion.

Re-implements src/buildsky/convert_skymodel.py (convert_sky_bbs_lsm /
convert_sky_lsm_bbs): BBS lines are
  name, TYPE, ra, dec, I, Q, U, V, ReferenceFrequency, SpectralIndex,
  MajorAxis, MinorAxis, Orientation
with ra `hh:mm:ss.ss` and dec `+dd.mm.ss.ss`; LSM is the reference's
whitespace format (sky.read_sky_model / README.md §2c).
"""
import argparse
import sys

import numpy as np

from .. import sky as skymod

AS2RAD = np.pi / (180.0 * 3600.0)
FWHM = 2.0 * np.sqrt(2.0 * np.log(2.0))


def _ra_str(ra):
    h = (ra % (2 * np.pi)) * 12.0 / np.pi
    hh = int(h)
    mm = int((h - hh) * 60)
    ss = ((h - hh) * 60 - mm) * 60
    return f"{hh:02d}:{mm:02d}:{ss:08.5f}"


def _dec_str(dec):
    sgn = '-' if dec < 0 else '+'
    d = abs(dec) * 180.0 / np.pi
    dd = int(d)
    mm = int((d - dd) * 60)
    ss = ((d - dd) * 60 - mm) * 60
    return f"{sgn}{dd:02d}.{mm:02d}.{ss:08.5f}"


def _parse_ra(s):
    p = s.replace('h', ':').replace('m', ':').split(':')
    return (float(p[0]) + float(p[1]) / 60 + float(p[2]) / 3600) \
        * np.pi / 12.0


def _parse_dec(s):
    p = s.replace('d', '.').split('.')
    sgn = -1.0 if s.strip().startswith('-') else 1.0
    dd = abs(float(p[0]))
    mm = float(p[1]) if len(p) > 1 else 0.0
    frac = '.'.join(p[2:]) if len(p) > 2 else '0'
    ss = float(frac) if frac else 0.0
    return sgn * (dd + mm / 60 + ss / 3600) * np.pi / 180.0


def lsm_to_bbs(infile, outfile, fmt=0):
    sources = skymod.read_sky_model(infile, fmt=fmt)
    with open(outfile, 'w') as f:
        f.write("# (Name, Type, Ra, Dec, I, Q, U, V, "
                "ReferenceFrequency='150e6', SpectralIndex='[0.0]', "
                "MajorAxis, MinorAxis, Orientation) = format\n")
        for name, s in sources.items():
            typ = 'GAUSSIAN' if s.stype != 0 else 'POINT'
            # LSM stores FWHM/2 in rad (readsky.c scaling); BBS wants
            # FWHM arcsec
            maj = s.eX * 2.0 / AS2RAD
            mnr = s.eY * 2.0 / AS2RAD
            pa = np.degrees(s.eP)
            f.write(f"{name}, {typ}, {_ra_str(s.ra)}, {_dec_str(s.dec)}, "
                    f"{s.sI:.6f}, {s.sQ:.6f}, {s.sU:.6f}, {s.sV:.6f}, "
                    f"{s.f0:.1f}, [{s.spec_idx:.4f}], "
                    f"{maj:.3f}, {mnr:.3f}, {pa:.2f}\n")
    return len(sources)


def bbs_to_lsm(infile, outfile):
    n = 0
    with open(infile) as fin, open(outfile, 'w') as f:
        f.write("## LSM file converted from BBS\n")
        f.write("# name h m s d m s I Q U V si RM eX eY eP f0\n")
        for line in fin:
            line = line.strip()
            if not line or line.startswith('#') or line.startswith('('):
                continue
            toks = [t.strip() for t in line.split(',')]
            if len(toks) < 8:
                continue
            name, typ = toks[0], toks[1].upper()
            ra = _parse_ra(toks[2])
            dec = _parse_dec(toks[3])
            sI, sQ, sU, sV = (float(toks[i]) for i in range(4, 8))
            f0 = float(toks[8]) if len(toks) > 8 and toks[8] else 150e6
            si = 0.0
            if len(toks) > 9 and toks[9]:
                si = float(toks[9].strip('[]') or 0.0)
            maj = mnr = pa = 0.0
            if typ.startswith('GAUS') and len(toks) >= 13:
                maj = float(toks[10]) * AS2RAD / 2.0
                mnr = float(toks[11]) * AS2RAD / 2.0
                pa = np.radians(float(toks[12]))
            h = (ra % (2 * np.pi)) * 12.0 / np.pi
            hh, hm = int(h), int((h - int(h)) * 60)
            hs = ((h - hh) * 60 - hm) * 60
            dsgn = -1 if dec < 0 else 1
            dabs = abs(dec) * 180.0 / np.pi
            dd, dm = int(dabs), int((dabs - int(dabs)) * 60)
            ds = ((dabs - dd) * 60 - dm) * 60
            pref = 'G' if typ.startswith('GAUS') else 'P'
            f.write(f"{pref}{name} {hh} {hm} {hs:.5f} {dsgn * dd} {dm} "
                    f"{ds:.5f} {sI} {sQ} {sU} {sV} {si} 0 "
                    f"{maj:.8e} {mnr:.8e} {pa:.6f} {f0}\n")
            n += 1
    return n


def main(argv=None):
    ap = argparse.ArgumentParser(prog='convert_skymodel')
    ap.add_argument('-i', dest='infile', required=True)
    ap.add_argument('-o', dest='outfile', required=True)
    g = ap.add_mutually_exclusive_group(required=True)
    g.add_argument('-b', dest='bbstolsm', action='store_true',
                   help='BBS -> LSM')
    g.add_argument('-l', dest='lsmtobbs', action='store_true',
                   help='LSM -> BBS')
    ap.add_argument('-F', dest='format', type=int, default=0)
    args = ap.parse_args(argv)
    if args.bbstolsm:
        n = bbs_to_lsm(args.infile, args.outfile)
    else:
        n = lsm_to_bbs(args.infile, args.outfile, fmt=args.format)
    print(f"convert_skymodel: {n} sources -> {args.outfile}")
    return 0


if __name__ == '__main__':
    sys.exit(main())
