"""Minimal FITS image I/O (2-D, single HDU) — no astropy in this
environment; buildsky/restore need only plain images with a linear sky
WCS (the reference links cfitsio+wcslib for the same purpose).

Supports BITPIX -32/-64 images with CRPIX/CRVAL/CDELT (deg) in a SIN-like
small-field approximation: pixel -> (l, m) direction cosines via
  l = (x - crpix1) * cdelt1 * pi/180,  m = (y - crpix2) * cdelt2 * pi/180
and (ra, dec) from the phase centre (crval1, crval2 deg).
"""
import numpy as np

BLOCK = 2880


def _card(key, value, comment=''):
    if isinstance(value, bool):
        v = 'T' if value else 'F'
        s = f"{key:8s}= {v:>20s}"
    elif isinstance(value, (int, np.integer)):
        s = f"{key:8s}= {value:20d}"
    elif isinstance(value, float):
        s = f"{key:8s}= {value:20.12E}"
    else:
        s = f"{key:8s}= '{value}'"
    if comment:
        s += f" / {comment}"
    return s[:80].ljust(80)


def write_fits_image(path, data, crval=(0.0, 45.0), cdelt=(-0.01, 0.01),
                     crpix=None, freq=150e6, bunit='JY/PIXEL'):
    """data: [ny, nx] float; WCS values in degrees."""
    data = np.asarray(data, dtype='>f8')
    ny, nx = data.shape
    if crpix is None:
        crpix = (nx / 2 + 1, ny / 2 + 1)
    cards = [
        _card('SIMPLE', True), _card('BITPIX', -64), _card('NAXIS', 2),
        _card('NAXIS1', nx), _card('NAXIS2', ny),
        _card('CTYPE1', 'RA---SIN'), _card('CRVAL1', float(crval[0])),
        _card('CRPIX1', float(crpix[0])), _card('CDELT1', float(cdelt[0])),
        _card('CTYPE2', 'DEC--SIN'), _card('CRVAL2', float(crval[1])),
        _card('CRPIX2', float(crpix[1])), _card('CDELT2', float(cdelt[1])),
        _card('RESTFRQ', float(freq)), _card('BUNIT', bunit),
        'END'.ljust(80),
    ]
    hdr = ''.join(cards).encode('ascii')
    hdr += b' ' * (-len(hdr) % BLOCK)
    body = data.tobytes()
    body += b'\0' * (-len(body) % BLOCK)
    with open(path, 'wb') as f:
        f.write(hdr)
        f.write(body)


def read_fits_image(path):
    """Returns (data [ny, nx] float64, header dict)."""
    with open(path, 'rb') as f:
        raw = f.read()
    hdr = {}
    pos = 0
    while True:
        card = raw[pos:pos + 80].decode('ascii', 'replace')
        pos += 80
        key = card[:8].strip()
        if key == 'END':
            break
        if '=' in card:
            val = card[10:].split('/')[0].strip()
            if val.startswith("'"):
                hdr[key] = val.strip("'").strip()
            elif val in ('T', 'F'):
                hdr[key] = val == 'T'
            else:
                try:
                    hdr[key] = int(val)
                except ValueError:
                    try:
                        hdr[key] = float(val)
                    except ValueError:
                        hdr[key] = val
    pos = (pos + BLOCK - 1) // BLOCK * BLOCK
    nx, ny = hdr['NAXIS1'], hdr['NAXIS2']
    bitpix = hdr['BITPIX']
    dt = {-64: '>f8', -32: '>f4', 16: '>i2', 32: '>i4'}[bitpix]
    n = nx * ny
    data = np.frombuffer(raw[pos:pos + n * abs(bitpix) // 8],
                         dtype=dt).astype(np.float64).reshape(ny, nx)
    if 'BSCALE' in hdr or 'BZERO' in hdr:
        data = data * hdr.get('BSCALE', 1.0) + hdr.get('BZERO', 0.0)
    return data, hdr


def pix_to_radec(hdr, x, y):
    """Pixel (0-based) -> (ra, dec) radians, small-field SIN approx."""
    d2r = np.pi / 180.0
    l = (np.asarray(x) + 1 - hdr['CRPIX1']) * hdr['CDELT1'] * d2r
    m = (np.asarray(y) + 1 - hdr['CRPIX2']) * hdr['CDELT2'] * d2r
    ra0 = hdr['CRVAL1'] * d2r
    dec0 = hdr['CRVAL2'] * d2r
    dec = np.arcsin(np.clip(m * np.cos(dec0)
                            + np.sqrt(np.maximum(1 - l * l - m * m, 0))
                            * np.sin(dec0), -1, 1))
    ra = ra0 + np.arctan2(l, np.cos(dec0)
                          * np.sqrt(np.maximum(1 - l * l - m * m, 0))
                          - m * np.sin(dec0))
    return ra, dec


def radec_to_pix(hdr, ra, dec):
    """(ra, dec) radians -> 0-based pixel coordinates."""
    d2r = np.pi / 180.0
    ra0 = hdr['CRVAL1'] * d2r
    dec0 = hdr['CRVAL2'] * d2r
    dra = np.asarray(ra) - ra0
    l = np.cos(dec) * np.sin(dra)
    m = (np.sin(dec) * np.cos(dec0) - np.cos(dec) * np.sin(dec0)
         * np.cos(dra))
    x = l / (hdr['CDELT1'] * d2r) + hdr['CRPIX1'] - 1
    y = m / (hdr['CDELT2'] * d2r) + hdr['CRPIX2'] - 1
    return x, y
