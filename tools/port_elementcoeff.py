#!/usr/bin/env python3
"""Port the LOFAR/ALO dipole element-beam coefficient tables from the
reference's generated data headers into the framework's .npz format.

The tables are DATA the reference ships as C headers
(/root/reference/src/lib/Radio/elementcoeff.h: LBA 10 freqs + HBA 15
freqs, modes=7 beta=0.5; elementcoeff_ALO.h: ALO 70 freqs), loaded by
set_elementcoeffs (elementbeam.c:40-186). This script parses those
headers and writes sagecal_amd/data/lofar_element_{lba,hba,alo}.npz
with keys: M (mode order), beta, freqs_ghz [Nf], theta [Nf, Nmodes]
complex128, phi [Nf, Nmodes] complex128 (Nmodes = M(M+1)/2 = 28).

Usage: python tools/port_elementcoeff.py [reference_dir] [out_dir]
"""
import os
import re
import sys

import numpy as np

CPLX = re.compile(
    r'([+-]?[\d.]+e?[+-]?\d*)\s*\+\s*_Complex_I\s*\*\s*\(\s*'
    r'([+-]?[\d.]+e?[+-]?\d*)\s*\)')


def _parse_double_array(text, name):
    m = re.search(re.escape(name) + r'\[\d+\]\s*=\s*\{(.*?)\};', text,
                  re.S)
    if m is None:
        raise ValueError(f"array {name} not found")
    return np.array([float(t) for t in
                     re.findall(r'[+-]?[\d.]+e?[+-]?\d*', m.group(1))])


def _parse_complex_table(text, name, nf, nmodes):
    m = re.search(re.escape(name) + r'\[\d+\]\[\d+\]\s*=\s*\{(.*?)\n\};',
                  text, re.S)
    if m is None:
        raise ValueError(f"table {name} not found")
    vals = [complex(float(re_), float(im_))
            for re_, im_ in CPLX.findall(m.group(1))]
    arr = np.array(vals, dtype=np.complex128)
    if arr.size != nf * nmodes:
        raise ValueError(f"{name}: got {arr.size}, want {nf * nmodes}")
    return arr.reshape(nf, nmodes)


def port(ref_dir, out_dir):
    os.makedirs(out_dir, exist_ok=True)
    sets = [
        ('elementcoeff.h', 'lba', 'BEAM_ELEM_MODES', 'BEAM_ELEM_BETA',
         'lba_beam_elem'),
        ('elementcoeff.h', 'hba', 'BEAM_ELEM_MODES', 'BEAM_ELEM_BETA',
         'hba_beam_elem'),
        ('elementcoeff_ALO.h', 'alo', 'ALO_BEAM_ELEM_MODES',
         'ALO_BEAM_ELEM_BETA', 'alo_beam_elem'),
    ]
    for fname, tag, mdef, bdef, prefix in sets:
        text = open(os.path.join(ref_dir, fname)).read()
        M = int(re.search(r'#define\s+' + mdef + r'\s+(\d+)',
                          text).group(1))
        beta = float(re.search(r'#define\s+' + bdef + r'\s+([\d.]+)',
                               text).group(1))
        nmodes = M * (M + 1) // 2
        freqs = _parse_double_array(text, f'{prefix}_freqs')
        theta = _parse_complex_table(text, f'{prefix}_theta', len(freqs),
                                     nmodes)
        phi = _parse_complex_table(text, f'{prefix}_phi', len(freqs),
                                   nmodes)
        out = os.path.join(out_dir, f'lofar_element_{tag}.npz')
        np.savez_compressed(out, M=M, beta=beta, freqs_ghz=freqs,
                            theta=theta, phi=phi)
        print(f"{tag}: M={M} beta={beta} Nf={len(freqs)} "
              f"Nmodes={nmodes} -> {out}")


if __name__ == '__main__':
    ref = sys.argv[1] if len(sys.argv) > 1 else \
        '/root/reference/src/lib/Radio'
    out = sys.argv[2] if len(sys.argv) > 2 else \
        os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), 'sagecal_amd', 'data')
    port(ref, out)
