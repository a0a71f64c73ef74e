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


# ---------------------------------------------------------------------------
# Lunar frames (native replacement for the reference's CSPICE path,
# /root/reference/src/lib/Radio/cspice_utils.c). The reference shells out to
# SPICE (pxform "J2000"->"MOON_ME" with DE440 kernels); here the Moon's
# orientation comes from the IAU/WGCCRE 2009 analytic series, which defines
# the lunar mean-Earth/polar-axis (ME) system to ~150 m on the surface
# (the documented IAU_MOON-vs-MOON_ME agreement) — ephemeris-file-free and
# ample for beam pointing / UVW work.
# ---------------------------------------------------------------------------

_J2000_JD = 2451545.0


def moon_orientation(jd):
    """WGCCRE 2009 lunar orientation: (alpha0, delta0, W) in radians at
    julian date `jd` (TDB). alpha0/delta0 = ICRF pole of the Moon, W =
    prime-meridian angle of the mean-Earth system."""
    d = np.asarray(jd, dtype=float) - _J2000_JD
    T = d / 36525.0
    E = np.deg2rad(np.array([
        [125.045, -0.0529921], [250.089, -0.1059842], [260.008, 13.0120009],
        [176.625, 13.3407154], [357.529, 0.9856003], [311.589, 26.4057084],
        [134.963, 13.0649930], [276.617, 0.3287146], [34.226, 1.7484877],
        [15.134, -0.1589763], [119.743, 0.0036096], [239.961, 0.1643573],
        [25.053, 12.9590088]]))
    ang = E[:, 0] + E[:, 1] * np.expand_dims(d, -1)
    s, c = np.sin(ang), np.cos(ang)
    a0 = (269.9949 + 0.0031 * T - 3.8787 * s[..., 0] - 0.1204 * s[..., 1]
          + 0.0700 * s[..., 2] - 0.0172 * s[..., 3] + 0.0072 * s[..., 5]
          - 0.0052 * s[..., 9] + 0.0043 * s[..., 12])
    d0 = (66.5392 + 0.0130 * T + 1.5419 * c[..., 0] + 0.0239 * c[..., 1]
          - 0.0278 * c[..., 2] + 0.0068 * c[..., 3] - 0.0029 * c[..., 5]
          + 0.0009 * c[..., 6] + 0.0008 * c[..., 9] - 0.0009 * c[..., 12])
    W = (38.3213 + 13.17635815 * d - 1.4e-12 * d * d + 3.5610 * s[..., 0]
         + 0.1208 * s[..., 1] - 0.0642 * s[..., 2] + 0.0158 * s[..., 3]
         + 0.0252 * s[..., 4] - 0.0066 * s[..., 5] - 0.0047 * s[..., 6]
         - 0.0046 * s[..., 7] + 0.0028 * s[..., 8] + 0.0052 * s[..., 9]
         + 0.0040 * s[..., 10] + 0.0019 * s[..., 11] - 0.0044 * s[..., 12])
    return np.deg2rad(a0), np.deg2rad(d0), np.deg2rad(W % 360.0)


def _rot_z(a):
    ca, sa = np.cos(a), np.sin(a)
    return np.array([[ca, sa, 0.0], [-sa, ca, 0.0], [0.0, 0.0, 1.0]])


def _rot_x(a):
    ca, sa = np.cos(a), np.sin(a)
    return np.array([[1.0, 0.0, 0.0], [0.0, ca, sa], [0.0, -sa, ca]])


def j2000_to_moon_me(jd):
    """3x3 rotation taking a J2000/ICRF vector into the lunar mean-Earth
    body-fixed frame (the reference's pxform_c("J2000","MOON_ME",...),
    cspice_utils.c:150): v_me = R @ v_j2000, with the standard IAU
    construction R = Rz(W) Rx(pi/2 - delta0) Rz(pi/2 + alpha0)."""
    a0, d0, W = moon_orientation(jd)
    return _rot_z(W) @ _rot_x(np.pi / 2 - d0) @ _rot_z(np.pi / 2 + a0)


MOON_RADIUS = 1737.4e3   # m (pck00011 triaxial radii are all 1737.4 km)


def xyz_to_lunar_latlon(x, y, z):
    """Selenocentric rectangular (m) -> (lon, lat, alt); the Moon's
    reference ellipsoid is a sphere, so planetographic == planetocentric
    (cspice_xyz_to_latlon, cspice_utils.c:115)."""
    r = np.sqrt(x * x + y * y + z * z)
    return np.arctan2(y, x), np.arcsin(np.clip(z / r, -1, 1)), r - MOON_RADIUS


def lunar_radec_to_latlon(ra, dec, jd):
    """J2000 (ra, dec) -> sub-source lunar (lon, lat) at epoch jd
    (cspice_longitude_latitude, cspice_utils.c:249)."""
    v = np.array([np.cos(dec) * np.cos(ra), np.cos(dec) * np.sin(ra),
                  np.sin(dec)])
    s = j2000_to_moon_me(jd) @ v
    return np.arctan2(s[1], s[0]), np.arcsin(np.clip(s[2], -1, 1))


def lunar_azel(ra, dec, lon, lat, jd):
    """Topocentric (az, el) of a J2000 source seen from lunar (lon, lat)
    at epoch jd: source direction in the ME frame -> local ENU. The
    reference computes only the zenith distance via a haversine between
    sub-source and station lon/lat (cspice_element_beam_lunar,
    cspice_utils.c:166-180); the ENU construction is equivalent for el
    and adds az."""
    v = np.array([np.cos(dec) * np.cos(ra), np.cos(dec) * np.sin(ra),
                  np.sin(dec)])
    s = j2000_to_moon_me(jd) @ v
    sl, cl = np.sin(lon), np.cos(lon)
    sb, cb = np.sin(lat), np.cos(lat)
    e = -sl * s[0] + cl * s[1]
    n = -sb * cl * s[0] - sb * sl * s[1] + cb * s[2]
    u = cb * cl * s[0] + cb * sl * s[1] + sb * s[2]
    return np.arctan2(e, n), np.arcsin(np.clip(u, -1, 1))


def lunar_uvw(pos_me, ra0, dec0, jd):
    """Station UVW (m) toward J2000 phase centre (ra0, dec0) for lunar
    body-fixed station positions pos_me [N,3] (m) at epoch jd — the
    uvwriter lunar mode (uvwriter.cpp:46): rotate stations to J2000 with
    the transpose frame matrix, then project on the standard (u,v,w)
    basis."""
    M = j2000_to_moon_me(jd)
    p = np.asarray(pos_me) @ M          # == (M.T @ pos.T).T
    sr, cr = np.sin(ra0), np.cos(ra0)
    sd, cd = np.sin(dec0), np.cos(dec0)
    uhat = np.array([-sr, cr, 0.0])
    vhat = np.array([-sd * cr, -sd * sr, cd])
    what = np.array([cd * cr, cd * sr, sd])
    return p @ uhat, p @ vhat, p @ what


def moon_position_j2000(jd):
    """Geocentric Moon direction (unit vector, J2000) from a truncated
    ELP/Meeus series (~0.3 deg worst case) — used to validate the ME frame
    orientation against the optical-libration bound, and available for
    earth-moon geometry checks."""
    T = (np.asarray(jd, dtype=float) - _J2000_JD) / 36525.0
    Lp = np.deg2rad(218.3164477 + 481267.88123421 * T)
    D = np.deg2rad(297.8501921 + 445267.1114034 * T)
    Ms = np.deg2rad(357.5291092 + 35999.0502909 * T)
    Mp = np.deg2rad(134.9633964 + 477198.8675055 * T)
    F = np.deg2rad(93.2720950 + 483202.0175233 * T)
    lam = Lp + np.deg2rad(
        6.288774 * np.sin(Mp) + 1.274027 * np.sin(2 * D - Mp)
        + 0.658314 * np.sin(2 * D) + 0.213618 * np.sin(2 * Mp)
        - 0.185116 * np.sin(Ms) - 0.114332 * np.sin(2 * F)
        + 0.058793 * np.sin(2 * D - 2 * Mp)
        + 0.057066 * np.sin(2 * D - Ms - Mp)
        + 0.053322 * np.sin(2 * D + Mp) + 0.045758 * np.sin(2 * D - Ms))
    bet = np.deg2rad(
        5.128122 * np.sin(F) + 0.280602 * np.sin(Mp + F)
        + 0.277693 * np.sin(Mp - F) + 0.173237 * np.sin(2 * D - F)
        + 0.055413 * np.sin(2 * D - Mp + F)
        + 0.046271 * np.sin(2 * D - Mp - F)
        + 0.032573 * np.sin(2 * D + F) + 0.017198 * np.sin(2 * Mp + F))
    eps = np.deg2rad(23.4392911 - 0.0130042 * T)
    x = np.cos(bet) * np.cos(lam)
    y = (np.cos(bet) * np.sin(lam) * np.cos(eps)
         - np.sin(bet) * np.sin(eps))
    z = (np.cos(bet) * np.sin(lam) * np.sin(eps)
         + np.sin(bet) * np.cos(eps))
    return np.array([x, y, z])


def precession_matrix(jd):
    """IAU-1976 precession rotation from J2000 mean equinox to the mean
    equinox of date `jd` (the role of the reference's NOVAS-derived
    get_precession_params/precession, transforms.c; used to precess
    catalogue positions to apparent, data.cpp:1616):
    R = Rz(-z) Ry(theta) Rz(-zeta), v_date = R @ v_J2000."""
    T = (np.asarray(jd, dtype=float) - _J2000_JD) / 36525.0
    arc = np.deg2rad(1.0 / 3600.0)
    zeta = (2306.2181 * T + 0.30188 * T ** 2 + 0.017998 * T ** 3) * arc
    z = (2306.2181 * T + 1.09468 * T ** 2 + 0.018203 * T ** 3) * arc
    th = (2004.3109 * T - 0.42665 * T ** 2 - 0.041833 * T ** 3) * arc

    def Rz(a):
        ca, sa = np.cos(a), np.sin(a)
        return np.array([[ca, sa, 0.0], [-sa, ca, 0.0], [0.0, 0.0, 1.0]])

    def Ry(a):
        ca, sa = np.cos(a), np.sin(a)
        return np.array([[ca, 0.0, -sa], [0.0, 1.0, 0.0], [sa, 0.0, ca]])
    return Rz(-z) @ Ry(th) @ Rz(-zeta)


def precess_radec(ra, dec, jd):
    """(ra, dec) at J2000 -> mean of date jd."""
    v = np.array([np.cos(dec) * np.cos(ra), np.cos(dec) * np.sin(ra),
                  np.sin(dec)])
    p = precession_matrix(jd) @ v
    return float(np.arctan2(p[1], p[0]) % (2 * np.pi)), \
        float(np.arcsin(np.clip(p[2], -1, 1)))
