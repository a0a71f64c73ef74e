"""Station beam models: array factor, HBA tile beamformer, dipole element
patterns.

Re-implements the physics of /root/reference/src/lib/Radio/stationbeam.c
(arraybeam :49 — delay-steered element sum), the two-stage tile beamformer
(predict_model.cu kernel_tile_array_beam:236), and elementbeam.c
(spherical-harmonic dipole patterns, set_elementcoeffs/eval_elementcoeffs).

Differences from the reference, by design:
  - geometry is ENU/az-el based (lon/lat + GMST) rather than casacore
    ITRF tables; the LOFAR_ANTENNA_FIELD metadata (element/tile offsets,
    pointing) maps onto ArrayConfig directly;
  - the LOFAR LBA/HBA/ALO element coefficient tables are DATA the
    reference ships as headers (elementcoeff.h); they are ported into
    sagecal_amd/data/lofar_element_*.npz (tools/port_elementcoeff.py)
    and evaluated by LofarElementCoeffs in the reference's own
    Laguerre-Gaussian basis; a synthetic generator + documented .npz
    format (ElementCoeffs) remain for custom dipole models.

All evaluation is vectorized torch over (source, time, station).
"""
import math

import numpy as np
import torch

from . import coords
from .constants import C_LIGHT


class ArrayConfig:
    """Per-station element layout + beam pointing (LOFAR_ANTENNA_FIELD
    analog, data.cpp:268-288)."""

    def __init__(self, element_enu, lon, lat, b_ra0, b_dec0,
                 tile_enu=None, freq0_tile=None):
        # element_enu: list of [K_s, 3] arrays (per station) or one shared
        self.element_enu = element_enu
        self.lon, self.lat = lon, lat
        self.b_ra0, self.b_dec0 = b_ra0, b_dec0
        self.tile_enu = tile_enu        # dipole offsets within a tile (HBA)
        self.freq0_tile = freq0_tile

    def elements(self, s):
        e = self.element_enu
        return e[s] if isinstance(e, (list, tuple)) else e


def _direction_enu(ra, dec, lon, lat, gmst):
    """Unit vector(s) toward (ra, dec) in ENU at the station."""
    az, el = coords.radec_to_azel_gmst(np.asarray(ra), np.asarray(dec),
                                       lon, lat, gmst)
    ce = np.cos(el)
    return np.stack([ce * np.sin(az), ce * np.cos(az), np.sin(el)],
                    axis=-1), az, el


def array_beam(cfg, ra, dec, freqs, tmjd, station_ids=None, device='cpu'):
    """Scalar array factor a[s, k, t, f] for stations s, sources k, times
    tmjd (days), freqs.

    arraybeam (stationbeam.c:49): af = (1/K) sum_elem exp(j 2 pi f/c
    r_e . (u_src - u_point)) — delay steering toward the beam pointing.
    """
    ra = np.atleast_1d(ra)
    tmjd = np.atleast_1d(tmjd)
    freqs = np.atleast_1d(freqs)
    gmst = coords.jd_to_gmst(np.asarray(tmjd) + 2400000.5)
    nsta = len(cfg.element_enu) if isinstance(cfg.element_enu,
                                              (list, tuple)) else 1
    if station_ids is None:
        station_ids = list(range(nsta))
    station_ids = list(station_ids)
    elem_list = [cfg.elements(s) for s in station_ids]
    counts = {np.asarray(e).shape[0] for e in elem_list}
    fq = torch.as_tensor(freqs, dtype=torch.float64, device=device)
    # per-time geometry (shared across stations: one array origin)
    dus, belows = [], []
    for g in np.atleast_1d(gmst):
        usrc, az, el = _direction_enu(ra, dec, cfg.lon, cfg.lat, g)
        upnt, _, _ = _direction_enu(cfg.b_ra0, cfg.b_dec0, cfg.lon,
                                    cfg.lat, g)
        dus.append(torch.as_tensor(usrc - upnt, dtype=torch.float64,
                                   device=device))     # [K, 3]
        belows.append(torch.as_tensor(el < 0, device=device))
    du = torch.stack(dus)                              # [T, K, 3]
    below = torch.stack(belows)                        # [T, K]
    if len(counts) == 1:
        # equal element counts: ONE batched pass — phases
        # [S,T,E,K] = elems[S,E,3] . du[T,K,3], then exp+mean over E and
        # the frequency axis broadcast; fused device-wide tensor ops
        # instead of per-(station,time,freq) python loops (the role of
        # kernel_array_beam's 2-D grid, predict_model.cu:129).
        elems = torch.as_tensor(np.stack(elem_list), dtype=torch.float64,
                                device=device)         # [S, E, 3]
        proj = torch.einsum('sex,tkx->stek', elems, du)
        ph = proj.unsqueeze(-1) * (2.0 * math.pi / C_LIGHT) * fq
        af = torch.complex(torch.cos(ph), torch.sin(ph)).mean(dim=2)
        af = torch.where(below[None, :, :, None],
                         torch.zeros_like(af), af)     # [S, T, K, F]
        return af.permute(0, 2, 1, 3).contiguous()     # [S, K, T, F]
    # jagged element counts: per-station batched over (T, K, F)
    out = []
    for elems_np in elem_list:
        elems = torch.as_tensor(elems_np, dtype=torch.float64,
                                device=device)
        proj = torch.einsum('ex,tkx->tek', elems, du)
        ph = proj.unsqueeze(-1) * (2.0 * math.pi / C_LIGHT) * fq
        af = torch.complex(torch.cos(ph), torch.sin(ph)).mean(dim=1)
        af = torch.where(below[:, :, None], torch.zeros_like(af), af)
        out.append(af.permute(1, 0, 2))                # [K, T, F]
    return torch.stack(out)


def tile_beam(cfg, ra, dec, freqs, tmjd, station_ids=None, device='cpu'):
    """Two-stage HBA beamformer (kernel_tile_array_beam,
    predict_model.cu:236): dipole-in-tile factor (steered at the tile
    reference frequency) x tile-centroid array factor."""
    a_tiles = array_beam(cfg, ra, dec, freqs, tmjd, station_ids, device)
    dip_cfg = ArrayConfig(cfg.tile_enu, cfg.lon, cfg.lat, cfg.b_ra0,
                          cfg.b_dec0)
    a_dip = array_beam(dip_cfg, ra, dec, freqs, tmjd, [0], device)
    return a_tiles * a_dip  # broadcast single dipole layout over stations


# ---------------------------------------------------------------------------
# Element (dipole) beams: spherical-harmonic patterns
# ---------------------------------------------------------------------------

class ElementCoeffs:
    """Dipole pattern coefficient set (elementcoeff.h analog).

    Stored as .npz with: 'n0' (modes per dim), 'freqs' [Nf] (wideband
    sets) and complex coefficient arrays 'ctheta_x','cphi_x','ctheta_y',
    'cphi_y' of shape [Nf, n0*n0]: pattern
    E_pol(theta, phi) = sum_{n,m} c_{nm} f_nm(theta, phi) with the same
    spherical-harmonic mode stack as sharmonic_modes (Dirac_radio.h:366).
    """

    def __init__(self, n0, freqs, ctheta_x, cphi_x, ctheta_y, cphi_y):
        self.n0 = n0
        self.freqs = np.atleast_1d(freqs)
        self.ctheta_x = ctheta_x
        self.cphi_x = cphi_x
        self.ctheta_y = ctheta_y
        self.cphi_y = cphi_y

    @staticmethod
    def load(path):
        z = np.load(path)
        return ElementCoeffs(int(z['n0']), z['freqs'], z['ctheta_x'],
                             z['cphi_x'], z['ctheta_y'], z['cphi_y'])

    def save(self, path):
        np.savez(path, n0=self.n0, freqs=self.freqs,
                 ctheta_x=self.ctheta_x, cphi_x=self.cphi_x,
                 ctheta_y=self.ctheta_y, cphi_y=self.cphi_y)

    def at_freq(self, f):
        """Nearest-frequency coefficient slice (wideband sets,
        set_elementcoeffs_wb elementbeam.c:186)."""
        i = int(np.argmin(np.abs(self.freqs - f)))
        return (self.ctheta_x[i], self.cphi_x[i], self.ctheta_y[i],
                self.cphi_y[i])


def sharmonic_basis(theta, phi, n0):
    """Real-form spherical-harmonic mode stack f_nm(theta, phi), modes
    (n, m) with n < n0, |m| <= n, flattened; theta: zenith angle."""
    try:
        from scipy.special import sph_harm_y
        def _sh(m, n, ph, th):
            return sph_harm_y(n, m, th, ph)
    except ImportError:                  # older scipy
        from scipy.special import sph_harm
        def _sh(m, n, ph, th):
            return sph_harm(m, n, ph, th)
    th = np.asarray(theta)
    ph = np.asarray(phi)
    cols = []
    for n in range(n0):
        for m in range(-n, n + 1):
            cols.append(_sh(m, n, ph, th))
    return np.stack(cols, axis=-1)      # [..., nmodes]


def n_modes(n0):
    return n0 * n0


def element_beam(coeffs, az, el, freq):
    """E-Jones per direction from the coefficient tables
    (eval_elementcoeffs, elementbeam.c): rows (theta, phi) pattern of the
    X and Y dipoles -> 2x2 complex [K, 2, 2]. Accepts either the
    synthetic spherical-harmonic ElementCoeffs or the real LOFAR tables
    (LofarElementCoeffs)."""
    if isinstance(coeffs, LofarElementCoeffs):
        zen = np.pi / 2 - np.asarray(el)
        return coeffs.eval(zen, np.asarray(az), freq)
    theta = np.pi / 2 - np.asarray(el)      # zenith angle
    phi = np.asarray(az)
    ct_x, cp_x, ct_y, cp_y = coeffs.at_freq(freq)
    Bs = sharmonic_basis(theta, phi, coeffs.n0)   # [K, modes]
    Ext = Bs @ ct_x
    Exp = Bs @ cp_x
    Eyt = Bs @ ct_y
    Eyp = Bs @ cp_y
    E = np.stack([np.stack([Ext, Exp], -1),
                  np.stack([Eyt, Eyp], -1)], -2)  # [K, 2, 2]
    return torch.from_numpy(E)


class LofarElementCoeffs:
    """Real LOFAR/ALO dipole element patterns in the reference's own
    Laguerre-Gaussian polar basis (elementbeam.c set_elementcoeffs:40 +
    eval_elementcoeffs:384; tables ported from elementcoeff.h /
    elementcoeff_ALO.h by tools/port_elementcoeff.py).

    Basis modes (n, m), n < M, m = -n..n step 2 (Nmodes = M(M+1)/2):
      f_nm(r, t) = P_nm (pi/4 + r)^|m| L_{(n-|m|)/2}^{|m|}(r^2/beta^2)
                   exp(-r^2/(2 beta^2)) exp(-j m t)
      P_nm = sqrt(((n-|m|)/2)! / (pi ((n+|m|)/2)!)) (-1)^{(n-|m|)/2}
             beta^{-1-|m|}
    with r the zenith angle and t the dipole-frame azimuth; the X dipole
    is evaluated at t = az - pi/4, the Y dipole at t = az + pi/4
    (stationbeam.c:331-345)."""

    def __init__(self, M, beta, freqs_ghz, theta, phi):
        self.M = int(M)
        self.beta = float(beta)
        self.freqs_ghz = np.atleast_1d(freqs_ghz).astype(float)
        self.theta = np.atleast_2d(theta)     # [Nf, Nmodes] complex
        self.phi = np.atleast_2d(phi)
        self.Nmodes = self.M * (self.M + 1) // 2
        assert self.theta.shape[1] == self.Nmodes
        # preamble per mode (elementbeam.c:160-175)
        import math
        pre = np.empty(self.Nmodes)
        idx = 0
        for n in range(self.M):
            for m in range(-n, n + 1, 2):
                am = abs(m)
                v = math.sqrt(math.factorial((n - am) // 2)
                              / (math.pi * math.factorial((n + am) // 2)))
                if ((n - am) // 2) % 2:
                    v = -v
                pre[idx] = v * self.beta ** (-1.0 - am)
                idx += 1
        self.preamble = pre
        self._nm = [(n, m) for n in range(self.M)
                    for m in range(-n, n + 1, 2)]

    @staticmethod
    def load(kind_or_path):
        """Load 'lba' / 'hba' / 'alo' (packaged tables) or an .npz path."""
        import os
        p = str(kind_or_path)
        if p.lower() in ('lba', 'hba', 'alo'):
            p = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                             'data', f'lofar_element_{p.lower()}.npz')
        z = np.load(p)
        return LofarElementCoeffs(z['M'], z['beta'], z['freqs_ghz'],
                                  z['theta'], z['phi'])

    def at_freq(self, freq_hz):
        """(theta, phi) coefficient vectors at a frequency: clamp at the
        table edges, linear interpolation inside (elementbeam.c:105-142)."""
        f = freq_hz / 1e9
        fr = self.freqs_ghz
        idh = int(np.searchsorted(fr, f, side='left'))
        if idh >= len(fr):
            idl = idh = len(fr) - 1
        elif idh == 0:
            idl = 0
        else:
            idl = idh - 1
        if idl == idh:
            return self.theta[idl], self.phi[idl]
        w1 = (f - fr[idl]) / (fr[idh] - fr[idl])
        return ((1 - w1) * self.theta[idl] + w1 * self.theta[idh],
                (1 - w1) * self.phi[idl] + w1 * self.phi[idh])

    def basis(self, r, t):
        """Mode stack f_nm at zenith angle r, basis azimuth t: [K, Nmodes]
        complex (vectorized eval_elementcoeffs inner loop)."""
        r = np.atleast_1d(np.asarray(r, dtype=float))
        t = np.atleast_1d(np.asarray(t, dtype=float))
        rb = (r / self.beta) ** 2
        ex = np.exp(-0.5 * rb)
        cols = np.empty((r.shape[0], self.Nmodes), dtype=np.complex128)
        # Laguerre L_p^q(rb) by the reference's recurrence (L_g1)
        lag = {}
        maxp = (self.M - 1) // 2 + 1
        for q in range(self.M):
            Lp2 = np.ones_like(rb)
            Lp1 = 1.0 - rb + q
            lag[(0, q)] = Lp2
            lag[(1, q)] = Lp1
            for i in range(2, maxp + 1):
                p1 = 1.0 / i
                Lp = (2.0 + p1 * (q - 1.0 - rb)) * Lp1 \
                    - (1.0 + p1 * (q - 1)) * Lp2
                lag[(i, q)] = Lp
                Lp2, Lp1 = Lp1, Lp
        for idx, (n, m) in enumerate(self._nm):
            am = abs(m)
            Lg = lag[((n - am) // 2, am)]
            rm = (np.pi / 4 + r) ** am
            pr = rm * Lg * ex * self.preamble[idx]
            cols[:, idx] = pr * np.exp(-1j * m * t)
        return cols

    def eval(self, zen, az, freq_hz):
        """E-Jones rows per direction: [K, 2, 2] complex with
        row 0 = X dipole (Etheta, Ephi) at t = az - pi/4,
        row 1 = Y dipole at t = az + pi/4 (stationbeam.c:331-345)."""
        ct, cp = self.at_freq(freq_hz)
        bx = self.basis(zen, np.asarray(az) - np.pi / 4)
        by = self.basis(zen, np.asarray(az) - np.pi / 4 + np.pi / 2)
        E = np.stack([np.stack([bx @ ct, bx @ cp], -1),
                      np.stack([by @ ct, by @ cp], -1)], -2)
        return torch.from_numpy(E)


def make_synthetic_element_coeffs(n0=4, freqs=(150e6,), seed=0,
                                  dipole_gain=1.0):
    """Physically-shaped synthetic dipole pattern: dominant n=0/1 terms
    (cos(theta)-like rolloff, orthogonal X/Y dipoles) + small higher-order
    structure. Real LBA/HBA tables in the same format drop in."""
    rng = np.random.default_rng(seed)
    Nf = len(freqs)
    M = n_modes(n0)
    def mk(main_m):
        c = np.zeros((Nf, M), dtype=np.complex128)
        idx = 0
        for n in range(n0):
            for m in range(-n, n + 1):
                if n == 0:
                    c[:, idx] = dipole_gain
                elif n == 1 and m == main_m:
                    c[:, idx] = 0.4 * dipole_gain
                else:
                    c[:, idx] = 0.02 * (rng.standard_normal(Nf)
                                        + 1j * rng.standard_normal(Nf))
                idx += 1
        return c
    return ElementCoeffs(n0, freqs, mk(1), mk(-1) * 0.1, mk(-1) * 0.1,
                         mk(1))


# ---------------------------------------------------------------------------
# Beam-aware predict (CPU/torch; predict_withbeam.c analog)
# ---------------------------------------------------------------------------

def predict_coh_withbeam(pack, u, v, w, freq, freq0, fdelta, tdelta, dec0,
                         cfg, tmjd, bb, Nbase, T, mode=1, coeffs=None):
    """Coherencies with the station beam applied per source/station
    (precalculate_coherencies_withbeam semantics, Dirac_radio.h:471):
    C'_pq(src) = g_p(src) C g_q(src)^* for the scalar array factor
    (mode 1), or E_p C E_q^H with element beams (mode 2).

    Returns [M, B, 2, 2]. Times tmjd: [T] MJD days (one per timeslot).

    The array factor is applied as its REAL magnitude |af| like the
    reference (beamgain = |sum phasors|/K, stationbeam.c:318 and the GPU
    beam buffer predict_model.cu:843-852), not as a complex gain.

    On GPU with mode 1 the whole predict runs as ONE fused kernel call:
    the torch-computed gains feed k_predict_coh's beam argument
    (kernel_array_beam -> kernel_coherencies structure).
    """
    from .ops import reference as R
    B = u.shape[0]
    M = pack.M
    # per-source single predicts (phase/smear only), then scale by beam
    cdtype = torch.complex128 if u.dtype == torch.float64 \
        else torch.complex64
    ra, dec = _pack_radec(pack, dec0)
    t_idx = torch.arange(B) // Nbase
    p = bb[:, 0]
    q = bb[:, 1]
    use_array = mode in (1, 2)
    use_elem = mode in (2, 3)
    if use_array:
        af = array_beam(cfg, ra, dec, [freq], tmjd)  # [S, K, T, 1]
        af = af[..., 0].abs()                        # [S, K, T] real
    if mode == 1 and u.is_cuda:
        # fused kernel path (-B 1): beam [T, K, N] + pairs into
        # k_predict_coh; falls back to the torch loop if the extension
        # is unavailable (dispatch raises on GPU by policy)
        from .ops import hip_host
        beam_t = af.permute(2, 1, 0).to(device=u.device,
                                        dtype=torch.float32).contiguous()
        return hip_host.predict_coh(
            pack, u, v, w, freq, freq0, fdelta, tdelta, dec0,
            beam=beam_t, pairs=bb[:Nbase].to(torch.int32),
            Nbase=Nbase).to(cdtype)
    out = torch.zeros(M, B, 2, 2, dtype=cdtype, device=u.device)
    if use_elem:
        # element E-Jones per (source, time): one dipole pattern per
        # array origin (kernel_element_beam; DOBEAM_ELEMENT/FULL modes)
        if coeffs is None:
            coeffs = make_synthetic_element_coeffs(freqs=(freq,))
        elif isinstance(coeffs, str):
            coeffs = LofarElementCoeffs.load(coeffs)
        gmst = coords.jd_to_gmst(np.asarray(np.atleast_1d(tmjd))
                                 + 2400000.5)
        Es = []
        for g in np.atleast_1d(gmst):
            _, az, el = _direction_enu(ra, dec, cfg.lon, cfg.lat, g)
            E = element_beam(coeffs, az, np.maximum(el, 0.0), freq)
            E[torch.from_numpy(el < 0)] = 0
            Es.append(E)
        Ej = torch.stack(Es, dim=1).to(cdtype)       # [K, T, 2, 2]
    for ci in range(M):
        s0, s1 = int(pack.cluster_off[ci]), int(pack.cluster_off[ci + 1])
        for gi in range(s0, s1):
            sub = _single_source_coh(pack, gi, u, v, w, freq, freq0,
                                     fdelta, tdelta, dec0)   # [B,2,2]
            if use_elem:
                Eb = Ej[gi, t_idx]                   # [B, 2, 2]
                sub = Eb @ sub @ Eb.conj().transpose(-1, -2)
            if use_array:
                gp = af[p, gi, t_idx]                # [B] real |af|
                gq = af[q, gi, t_idx]
                sub = (gp * gq).to(sub.dtype)[:, None, None] * sub
            out[ci] += sub
    return out


def _pack_radec(pack, dec0):
    """Recover per-source (ra, dec) from direction cosines (inverse of
    radec_to_lmn at ra0=0)."""
    ll = pack.ll.cpu().numpy()
    mm = pack.mm.cpu().numpy()
    nn = pack.nn1.cpu().numpy() + 1.0
    dec = np.arcsin(np.clip(mm * np.cos(dec0) + nn * np.sin(dec0), -1, 1))
    ra = np.arctan2(ll, nn * np.cos(dec0) - mm * np.sin(dec0))
    return ra, dec


def _single_source_coh(pack, gi, u, v, w, freq, freq0, fdelta, tdelta,
                       dec0):
    """Coherency of a single source (phase+smear+envelope+Stokes)."""
    from .ops import reference as R

    class _P:
        pass
    one = _P()
    for f_ in R.SourcePack.FIELDS:
        setattr(one, f_, getattr(pack, f_)[gi:gi + 1])
    one.stype = pack.stype[gi:gi + 1]
    one.use_proj = pack.use_proj[gi:gi + 1]
    one.cluster_off = torch.tensor([0, 1])
    one.M = 1
    one.shapelets = ({0: pack.shapelets[gi]}
                     if getattr(pack, 'shapelets', None)
                     and gi in pack.shapelets else {})
    return R.predict_coh(one, u, v, w, freq, freq0, fdelta, tdelta,
                         dec0)[0]


def lunar_station_azel(ra, dec, lon, lat, jd):
    """Vectorized topocentric (az, el) of J2000 sources over lunar
    stations: sources [S] x stations [N] -> ([N,S], [N,S]).
    Native replacement for the per-station loop in
    cspice_element_beam_lunar (cspice_utils.c:160-195); the reference
    approximates el = pi/2 - |dlat| from a haversine between sub-source
    and station lon/lat — here the exact local-ENU elevation is used."""
    from . import coords
    ra = np.atleast_1d(np.asarray(ra, dtype=float))
    dec = np.atleast_1d(np.asarray(dec, dtype=float))
    lon = np.atleast_1d(np.asarray(lon, dtype=float))
    lat = np.atleast_1d(np.asarray(lat, dtype=float))
    v = np.stack([np.cos(dec) * np.cos(ra), np.cos(dec) * np.sin(ra),
                  np.sin(dec)])                       # [3, S]
    s = coords.j2000_to_moon_me(jd) @ v               # [3, S]
    sl, cl = np.sin(lon)[:, None], np.cos(lon)[:, None]
    sb, cb = np.sin(lat)[:, None], np.cos(lat)[:, None]
    e = -sl * s[0] + cl * s[1]
    n = -sb * cl * s[0] - sb * sl * s[1] + cb * s[2]
    u = cb * cl * s[0] + cb * sl * s[1] + sb * s[2]
    return np.arctan2(e, n), np.arcsin(np.clip(u, -1.0, 1.0))


def element_beam_lunar(coeffs, ra, dec, lon, lat, jd, freq):
    """Per-station lunar element E-Jones [N, S, 2, 2]
    (cspice_element_beam_lunar): sources below the lunar horizon get a
    zero response (the reference's el<0 'invalid' branch)."""
    az, el = lunar_station_azel(ra, dec, lon, lat, jd)
    N, S = az.shape
    E = element_beam(coeffs, az.reshape(-1), el.reshape(-1), freq)
    E = E.reshape(N, S, 2, 2)
    E[torch.from_numpy(el < 0)] = 0
    return E
