"""Coordinate transforms: RA/Dec -> direction cosines, GMST, ITRF <-> geodetic.

Re-implements the functionality of /root/reference/src/lib/Radio/transforms.c
(xyz2llh, radec2azel_gmst) with numpy, plus the lmn computation used by
readsky.c (the phase-centre projection; nn is stored as n-1 there,
readsky.c:628).
"""
import numpy as np


def radec_to_lmn(ra, dec, ra0, dec0):
    """Direction cosines (l, m, n) of sources at (ra, dec) relative to phase
    centre (ra0, dec0). Returns n itself (callers subtract 1 for the w term,
    matching reference readsky.c:628 which stores nn-1)."""
    ra = np.asarray(ra, dtype=np.float64)
    dec = np.asarray(dec, dtype=np.float64)
    dra = ra - ra0
    l = np.cos(dec) * np.sin(dra)
    m = np.sin(dec) * np.cos(dec0) - np.cos(dec) * np.sin(dec0) * np.cos(dra)
    n = np.sin(dec) * np.sin(dec0) + np.cos(dec) * np.cos(dec0) * np.cos(dra)
    return l, m, n


def jd_to_gmst(jd):
    """Greenwich mean sidereal time [rad] from Julian date (UT1).

    Same truncated IAU expression class as the reference's GPU
    kernel_convert_time (predict_model.cu:1886)."""
    t = (jd - 2451545.0) / 36525.0
    gmst_sec = (67310.54841
                + (876600.0 * 3600.0 + 8640184.812866) * t
                + 0.093104 * t * t
                - 6.2e-6 * t * t * t)
    gmst = np.remainder(np.remainder(gmst_sec, 86400.0) * (2.0 * np.pi / 86400.0),
                        2.0 * np.pi)
    return gmst


def radec_to_azel_gmst(ra, dec, lon, lat, gmst):
    """RA/Dec -> azimuth/elevation given observer lon/lat and GMST [rad].
    Mirrors transforms.c radec2azel_gmst."""
    ha = gmst + lon - ra  # local hour angle
    sin_el = np.sin(lat) * np.sin(dec) + np.cos(lat) * np.cos(dec) * np.cos(ha)
    sin_el = np.clip(sin_el, -1.0, 1.0)
    el = np.arcsin(sin_el)
    az = np.arctan2(-np.cos(dec) * np.sin(ha),
                    np.sin(dec) * np.cos(lat) - np.cos(dec) * np.sin(lat) * np.cos(ha))
    az = np.remainder(az, 2.0 * np.pi)
    return az, el


def xyz_to_llh(x, y, z):
    """ITRF cartesian -> geodetic lon/lat/height (WGS84), iterative method as
    transforms.c xyz2llh."""
    a = 6378137.0
    f = 1.0 / 298.257223563
    e2 = 2 * f - f * f
    lon = np.arctan2(y, x)
    p = np.sqrt(np.asarray(x) ** 2 + np.asarray(y) ** 2)
    lat = np.arctan2(z, p * (1 - e2))
    for _ in range(10):
        N = a / np.sqrt(1 - e2 * np.sin(lat) ** 2)
        h = p / np.cos(lat) - N
        lat = np.arctan2(z, p * (1 - e2 * N / (N + h)))
    N = a / np.sqrt(1 - e2 * np.sin(lat) ** 2)
    h = p / np.cos(lat) - N
    return lon, lat, h


def hms_to_rad(h, m, s):
    """Hours/min/sec of RA -> radians (readsky.c sign conventions)."""
    sign = -1.0 if (h < 0 or (h == 0 and (m < 0 or s < 0))) else 1.0
    return sign * (abs(h) + abs(m) / 60.0 + abs(s) / 3600.0) * np.pi / 12.0


def dms_to_rad(d, m, s):
    """Deg/min/sec of Dec -> radians."""
    sign = -1.0 if (d < 0 or (d == 0 and (m < 0 or s < 0))) else 1.0
    return sign * (abs(d) + abs(m) / 60.0 + abs(s) / 3600.0) * np.pi / 180.0
