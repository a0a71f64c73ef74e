"""Sky model and cluster-file handling.

Re-implements the semantics of /root/reference/src/lib/Radio/readsky.c
(read_sky_cluster, readsky.c:195) with numpy arrays instead of glib hash
tables: LSM sky-model parsing (both the 1-term and 3-term log-spectral-index
formats, README.md "Sky model format"), cluster-file parsing with hybrid
chunk counts, per-cluster regularization files (read_arho_fromfile,
Dirac_radio.h:143), and ignore lists.

Cluster data is stored struct-of-arrays, GPU-ready: each cluster carries
contiguous float64 arrays (ll, mm, nn-1, fluxes, spectral indices, shape
params) so the predict kernels consume them without per-source pointers.
"""
import numpy as np
from dataclasses import dataclass, field
from . import coords
from .constants import (STYPE_POINT, STYPE_GAUSSIAN, STYPE_DISK, STYPE_RING,
                        STYPE_SHAPELET)

PROJ_CUT = 0.998  # reference: Dirac_common.h:90
FWHM_TO_SIGMA = 1.0 / (2.0 * np.sqrt(2.0 * np.log(2.0)))  # readsky.c:415


@dataclass
class Source:
    name: str
    ra: float
    dec: float
    sI: float
    sQ: float
    sU: float
    sV: float
    spec_idx: float = 0.0
    spec_idx1: float = 0.0
    spec_idx2: float = 0.0
    RM: float = 0.0
    eX: float = 0.0
    eY: float = 0.0
    eP: float = 0.0
    f0: float = 0.0
    stype: int = STYPE_POINT
    # shapelet data (modes file), set for STYPE_SHAPELET
    sh_n0: int = 0
    sh_beta: float = 0.0
    sh_coeff: np.ndarray = None


@dataclass
class Cluster:
    """One direction cluster: struct-of-arrays over its sources.

    Mirrors clus_source_t (Dirac_common.h:173-197): ll/mm/nn (nn stores n-1,
    readsky.c:628), per-source fluxes already scaled to the data reference
    frequency, original fluxes + spectral indices for multi-frequency
    evaluation, source types and shape parameters.
    """
    cluster_id: int
    nchunk: int                     # hybrid chunk count ("chunk_size" col)
    names: list
    ll: np.ndarray                  # direction cosines
    mm: np.ndarray
    nn1: np.ndarray                 # n - 1
    sI: np.ndarray                  # I at data freq0 (scaled by spec idx)
    sQ: np.ndarray
    sU: np.ndarray
    sV: np.ndarray
    sI0: np.ndarray                 # original catalogue fluxes
    sQ0: np.ndarray
    sU0: np.ndarray
    sV0: np.ndarray
    spec_idx: np.ndarray
    spec_idx1: np.ndarray
    spec_idx2: np.ndarray
    f0: np.ndarray                  # per-source reference frequency
    stype: np.ndarray               # int8 source types
    # extended-source params (valid where stype != POINT):
    eX: np.ndarray
    eY: np.ndarray
    eP: np.ndarray
    # projection terms (readsky.c:405-422): cos(xi), sin(-xi), cos(phi), sin(-phi)
    cxi: np.ndarray
    sxi: np.ndarray
    cphi: np.ndarray
    sphi: np.ndarray
    use_proj: np.ndarray            # bool
    shapelets: list = field(default_factory=list)  # (index, n0, beta, coeff[n0*n0])

    @property
    def nsrc(self):
        return len(self.ll)


def _parse_sky_lines(path):
    """Yield token lists of non-comment, non-empty lines."""
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line or line.startswith('#'):
                continue
            yield line.split()


def read_sky_model(path, fmt=0, modes_dir=None):
    """Parse an LSM sky model file into a dict name -> Source.

    fmt=0: 1-term spectral index (16 numeric cols after name)
    fmt=1: 3-term spectral index (18 numeric cols after name), reference -F 1.
    Column order per README.md §2c / readsky.c:305-308.
    """
    sources = {}
    for toks in _parse_sky_lines(path):
        name = toks[0]
        need = 18 if fmt == 1 else 16
        try:
            vals = [float(t) for t in toks[1:]]
            if len(vals) < need:
                raise ValueError(
                    f"{len(vals)} numeric columns, need {need}")
        except ValueError as e:
            raise ValueError(
                f"sky model {path}: bad line for source '{name}': {e} "
                f"(format {fmt}: name + {need} columns per README §2c)"
            ) from e
        if fmt == 1:
            (rahr, ramin, rasec, decd, decmin, decsec, sI, sQ, sU, sV,
             si0, si1, si2, RM, eX, eY, eP, f0) = vals[:18]
        else:
            (rahr, ramin, rasec, decd, decmin, decsec, sI, sQ, sU, sV,
             si0, RM, eX, eY, eP, f0) = vals[:16]
            si1 = si2 = 0.0
        ra = coords.hms_to_rad(rahr, ramin, rasec)
        dec = coords.dms_to_rad(decd, decmin, decsec)
        stype = STYPE_POINT
        c0 = name[0].upper()
        # source type from name prefix (readsky.c: G*/D*/R*/S* naming rule)
        if eX != 0.0 or eY != 0.0 or c0 in 'GDRS':
            if c0 == 'G':
                stype = STYPE_GAUSSIAN
            elif c0 == 'D':
                stype = STYPE_DISK
            elif c0 == 'R':
                stype = STYPE_RING
            elif c0 == 'S':
                stype = STYPE_SHAPELET
        src = Source(name, ra, dec, sI, sQ, sU, sV,
                     si0, si1, si2, RM, eX, eY, eP, f0, stype)
        if stype == STYPE_SHAPELET:
            # reference looks for <name>.fits.modes in the working dir
            # (readsky.c:149); we search next to the sky file as well
            import os
            from . import shapelet as shmod
            for d in (modes_dir, os.path.dirname(os.path.abspath(path)),
                      '.'):
                if d is None:
                    continue
                mf = os.path.join(d, name + '.fits.modes')
                if os.path.exists(mf):
                    src.sh_n0, src.sh_beta, src.sh_coeff = \
                        shmod.read_modes_file(mf)
                    break
            if src.eX == 0:
                src.eX = 1.0
            if src.eY == 0:
                src.eY = 1.0
        sources[name] = src
    return sources


def read_cluster_file(path):
    """Parse a cluster file: rows of `cluster_id chunk_size src1 src2 ...`
    (README.md §2b). Returns list of (cluster_id, nchunk, [names])."""
    out = []
    for toks in _parse_sky_lines(path):
        cid = int(toks[0])
        nchunk = int(toks[1])
        out.append((cid, nchunk, toks[2:]))
    return out


def _scaled_flux(s, f_ratio_log, f):
    """Flux at frequency f given catalogue flux s and log(f/f0) terms,
    preserving sign as readsky.c:353-376 does."""
    if s == 0.0:
        return 0.0
    mag = np.exp(np.log(abs(s)) + f_ratio_log)
    return mag if s > 0 else -mag


def build_clusters(sources, cluster_list, ra0, dec0, freq0,
                   ignore_ids=()):
    """Assemble Cluster SoA structures for the given phase centre and data
    reference frequency (the channel-averaged solve frequency).

    Mirrors the tail of read_sky_cluster (readsky.c:580-650): computes
    direction cosines w.r.t. (ra0, dec0), scales fluxes to freq0 using the
    log-polynomial spectral index, prepares extended-source projection terms.
    """
    clusters = []
    ignore = set(ignore_ids)
    for cid, nchunk, names in cluster_list:
        if cid in ignore:
            continue
        srcs = [sources[n] for n in names if n in sources]
        if not srcs:
            continue
        n = len(srcs)
        ll = np.empty(n); mm = np.empty(n); nn = np.empty(n)
        arr = {k: np.zeros(n) for k in
               ('sI', 'sQ', 'sU', 'sV', 'sI0', 'sQ0', 'sU0', 'sV0',
                'spec_idx', 'spec_idx1', 'spec_idx2', 'f0',
                'eX', 'eY', 'eP', 'cxi', 'sxi', 'cphi', 'sphi')}
        stype = np.zeros(n, dtype=np.int8)
        use_proj = np.zeros(n, dtype=bool)
        shapelets = []
        for i, s in enumerate(srcs):
            l, m, nd = coords.radec_to_lmn(s.ra, s.dec, ra0, dec0)
            ll[i], mm[i], nn[i] = l, m, nd
            f0 = s.f0 if s.f0 > 0 else freq0
            lf = np.log(freq0 / f0)
            flog = s.spec_idx * lf + s.spec_idx1 * lf ** 2 + s.spec_idx2 * lf ** 3
            arr['sI'][i] = _scaled_flux(s.sI, flog, freq0)
            arr['sQ'][i] = _scaled_flux(s.sQ, flog, freq0)
            arr['sU'][i] = _scaled_flux(s.sU, flog, freq0)
            arr['sV'][i] = _scaled_flux(s.sV, flog, freq0)
            arr['sI0'][i], arr['sQ0'][i] = s.sI, s.sQ
            arr['sU0'][i], arr['sV0'][i] = s.sU, s.sV
            arr['spec_idx'][i] = s.spec_idx
            arr['spec_idx1'][i] = s.spec_idx1
            arr['spec_idx2'][i] = s.spec_idx2
            arr['f0'][i] = f0
            stype[i] = s.stype
            if s.stype != STYPE_POINT:
                # projection terms, readsky.c:399-422
                phi = np.arccos(np.clip(nd, -1.0, 1.0))
                xi = np.arctan2(-l, m)
                arr['cxi'][i] = np.cos(xi)
                arr['sxi'][i] = np.sin(-xi)
                arr['cphi'][i] = np.cos(phi)
                arr['sphi'][i] = np.sin(-phi)
                use_proj[i] = nd < PROJ_CUT
                if s.stype == STYPE_GAUSSIAN:
                    arr['eX'][i] = s.eX * FWHM_TO_SIGMA
                    arr['eY'][i] = s.eY * FWHM_TO_SIGMA
                else:
                    arr['eX'][i] = s.eX
                    arr['eY'][i] = s.eY
                arr['eP'][i] = s.eP
                if s.stype == STYPE_SHAPELET and s.sh_coeff is not None:
                    shapelets.append((i, s.sh_n0, s.sh_beta,
                                      np.asarray(s.sh_coeff, dtype=np.float64)))
        clusters.append(Cluster(
            cluster_id=cid, nchunk=nchunk, names=[s.name for s in srcs],
            ll=ll, mm=mm, nn1=nn - 1.0,
            sI=arr['sI'], sQ=arr['sQ'], sU=arr['sU'], sV=arr['sV'],
            sI0=arr['sI0'], sQ0=arr['sQ0'], sU0=arr['sU0'], sV0=arr['sV0'],
            spec_idx=arr['spec_idx'], spec_idx1=arr['spec_idx1'],
            spec_idx2=arr['spec_idx2'], f0=arr['f0'], stype=stype,
            eX=arr['eX'], eY=arr['eY'], eP=arr['eP'],
            cxi=arr['cxi'], sxi=arr['sxi'], cphi=arr['cphi'], sphi=arr['sphi'],
            use_proj=use_proj, shapelets=shapelets))
    return clusters


def read_sky_cluster(sky_path, cluster_path, ra0, dec0, freq0, fmt=0,
                     ignore_ids=(), jd=None):
    """One-call equivalent of reference read_sky_cluster (readsky.c:195).

    jd: if given, precess the J2000 catalogue positions AND the phase
    centre to the mean equinox of that epoch before computing direction
    cosines (the reference does this from the MS time, data.cpp:1616)."""
    sources = read_sky_model(sky_path, fmt=fmt)
    if jd is not None:
        from . import coords
        for s in sources.values():
            s.ra, s.dec = coords.precess_radec(s.ra, s.dec, jd)
        ra0, dec0 = coords.precess_radec(ra0, dec0, jd)
    clist = read_cluster_file(cluster_path)
    return build_clusters(sources, clist, ra0, dec0, freq0, ignore_ids)


def read_arho_file(path, clusters):
    """Per-cluster regularization file (-G option): rows of
    `cluster_id hybrid_factor spectral_reg [spatial_reg]`
    (README.md §4; reference read_arho_fromfile). Returns (arho, arho_spatial)
    arrays aligned with `clusters` order."""
    table = {}
    for toks in _parse_sky_lines(path):
        cid = int(toks[0])
        spec = float(toks[2])
        spat = float(toks[3]) if len(toks) > 3 else 0.0
        table[cid] = (spec, spat)
    arho = np.zeros(len(clusters))
    arho_s = np.zeros(len(clusters))
    for i, c in enumerate(clusters):
        if c.cluster_id in table:
            arho[i], arho_s[i] = table[c.cluster_id]
    return arho, arho_s


def read_ignore_file(path):
    """Ignore list: one cluster id per line."""
    ids = []
    for toks in _parse_sky_lines(path):
        ids.append(int(toks[0]))
    return ids


def make_synthetic_sky(M=10, nsrc_per_cluster=5, seed=0, fov_rad=0.05,
                       freq0=150e6, flux_lo=0.5, flux_hi=10.0):
    """Random synthetic sky for benchmarks/tests: M clusters of point sources
    scattered in a field of view. Returns (sources dict, cluster list)."""
    rng = np.random.default_rng(seed)
    sources = {}
    clist = []
    ra0, dec0 = 0.0, np.pi / 4
    for c in range(M):
        names = []
        # cluster centres evenly spread on a ring (real calibration
        # directions are well separated in the field)
        ang = 2 * np.pi * c / max(M, 1) + 0.3
        rad = fov_rad * (0.4 + 0.6 * ((c * 7919) % M + 1) / max(M, 1)) if M > 1 else 0.0
        cra = ra0 + rad * np.cos(ang)
        cdec = dec0 + rad * np.sin(ang)
        for s in range(nsrc_per_cluster):
            name = f"PC{c}S{s}"
            ra = cra + rng.uniform(-0.1, 0.1) * fov_rad
            dec = cdec + rng.uniform(-0.1, 0.1) * fov_rad
            flux = float(rng.uniform(flux_lo, flux_hi))
            sources[name] = Source(name, ra, dec, flux, 0.0, 0.0, 0.0,
                                   spec_idx=float(rng.uniform(-0.9, 0.1)),
                                   f0=freq0)
            names.append(name)
        clist.append((c, 1, names))
    return sources, clist
