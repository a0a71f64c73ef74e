"""Host-side adapters between the dispatch API and the gfx950 kernels.

Handles layout preparation: per-frequency flux scaling (done with torch on
device, keeping the predict kernel free of spectral-index math), baking the
projection identity into unused rotation terms, chunk tables, station-pair
index tables, and complex64 views.
"""
import math
import os

import torch

from . import dispatch

_pidx_cache = {}


def _ext():
    e = dispatch._load_ext()
    if e is None:
        raise RuntimeError(
            f"dirac_hip extension unavailable: {dispatch._ext_err}")
    return e


class GPUPack:
    """Device-resident SourcePack in kernel layout (float32 props, float64
    direction cosines, projection identity baked for use_proj=False)."""

    def __init__(self, pack, device):
        f32 = lambda t: t.to(device=device, dtype=torch.float32).contiguous()
        f64 = lambda t: t.to(device=device, dtype=torch.float64).contiguous()
        self.M = pack.M
        self.ll, self.mm, self.nn1 = f64(pack.ll), f64(pack.mm), f64(pack.nn1)
        for k in ('sI', 'sQ', 'sU', 'sV', 'sI0', 'sQ0', 'sU0', 'sV0',
                  'spec_idx', 'spec_idx1', 'spec_idx2', 'eX', 'eY', 'eP'):
            setattr(self, k, f32(getattr(pack, k)))
        self.f0 = pack.f0.to(device=device, dtype=torch.float64)
        # gaussian honors use_projection; disk/ring ALWAYS project
        # (disk_contrib/ring_contrib predict.c:60-90, oracle-verified)
        up = pack.use_proj.to(device=device).bool() | \
            (pack.stype.to(device) >= 2)
        one = torch.ones_like(f32(pack.cxi))
        zero = torch.zeros_like(one)
        self.cxi = torch.where(up, f32(pack.cxi), one).contiguous()
        self.sxi = torch.where(up, f32(pack.sxi), zero).contiguous()
        self.cphi = torch.where(up, f32(pack.cphi), one).contiguous()
        self.sphi = torch.where(up, f32(pack.sphi), zero).contiguous()
        self.stype = pack.stype.to(device=device,
                                   dtype=torch.int32).contiguous()
        # shapelet sources are handled by the torch path (shapelet.py):
        # zero their fluxes for the kernel so they contribute nothing there
        self.shapelets = dict(getattr(pack, 'shapelets', {}))
        if self.shapelets:
            # keep the true fluxes for the host shapelet pass, zero them
            # in the kernel arrays
            sh = self.stype == 4
            self._shfull = {}
            for k in ('sI', 'sQ', 'sU', 'sV', 'sI0', 'sQ0', 'sU0', 'sV0'):
                t = getattr(self, k)
                self._shfull[k] = t.clone()
                t = t.clone()
                t[sh] = 0.0
                setattr(self, k, t.contiguous())
        self.pack_ref = pack
        # zero shape angles for point sources keep the envelope branch out
        self.cluster_off = pack.cluster_off.to(device=device,
                                               dtype=torch.int32).contiguous()
        self.freq0 = None
        self._flux_cache = {}

    def fluxes_at(self, freq, freq0):
        """Per-source fluxes at channel `freq`; pack fluxes are given at
        freq0 (sI..) and catalogue f0 (sI0..)."""
        key = round(float(freq), 3)
        if key in self._flux_cache:
            return self._flux_cache[key]
        if abs(freq - freq0) < 1.0:
            out = (self.sI, self.sQ, self.sU, self.sV)
        else:
            lf = torch.log(torch.tensor(float(freq), dtype=torch.float64,
                                        device=self.ll.device) / self.f0)
            lf = lf.to(torch.float32)
            flog = (self.spec_idx * lf + self.spec_idx1 * lf ** 2
                    + self.spec_idx2 * lf ** 3)

            def scale(s0):
                mag = torch.exp(torch.log(s0.abs().clamp_min(1e-30)) + flog)
                return torch.where(s0 == 0, torch.zeros_like(s0),
                                   torch.sign(s0) * mag)
            out = (scale(self.sI0), scale(self.sQ0), scale(self.sU0),
                   scale(self.sV0))
        self._flux_cache[key] = out
        return out

    def r1_for(self, dec0):
        """Time-smear source-distance term sqrt(ll^2+(sin(dec0) mm)^2)
        (predict.c:98-100), per source."""
        ds = math.sin(dec0)
        return torch.sqrt(self.ll ** 2 + (ds * self.mm) ** 2).to(
            torch.float32).contiguous()


_gpu_pack_cache = {}


def gpu_pack(pack, device):
    key = (dispatch.obj_token(pack), str(device))
    if key not in _gpu_pack_cache:
        if len(_gpu_pack_cache) > 16:   # bound long multi-MS runs
            _gpu_pack_cache.pop(next(iter(_gpu_pack_cache)))
        _gpu_pack_cache[key] = GPUPack(pack, device)
    return _gpu_pack_cache[key]


def predict_coh(pack, u, v, w, freq, freq0, fdelta, tdelta, dec0,
                beam=None, pairs=None, Nbase=0, **kw):
    """Kernel-backed coherency predict. Optional fused station beam
    (-B 1, reference DOBEAM_ARRAY): `beam` [T, K, N] float32 real
    array-factor gains on device + `pairs` [Nbase, 2] int32 — each
    source's contribution is scaled by beam[t,k,p]*beam[t,k,q] inside
    k_predict_coh (predict_model.cu:843-852 semantics)."""
    gp = pack if isinstance(pack, GPUPack) else gpu_pack(pack, u.device)
    sI, sQ, sU, sV = gp.fluxes_at(freq, freq0)
    r1 = gp.r1_for(dec0)
    u64 = u.to(torch.float64).contiguous()
    v64 = v.to(torch.float64).contiguous()
    w64 = w.to(torch.float64).contiguous()
    if beam is not None:
        beam = beam.to(device=u.device, dtype=torch.float32).contiguous()
        pairs = pairs.to(device=u.device, dtype=torch.int32).contiguous()
        Nbase = int(Nbase) or int(pairs.shape[0])
    out = _ext().predict_coh(
        u64, v64, w64, gp.ll, gp.mm, gp.nn1, sI, sQ, sU, sV,
        gp.eX, gp.eY, gp.eP, gp.cxi, gp.sxi, gp.cphi, gp.sphi, r1,
        gp.stype, gp.cluster_off, float(freq), float(fdelta) * 0.5,
        float(tdelta), beam=beam, pairs=pairs, Nbase=int(Nbase))
    out = out.view(gp.M, -1, 2, 2)
    if gp.shapelets:
        out = out + _shapelet_coh(gp, u64, v64, w64, freq, freq0, fdelta,
                                  tdelta, dec0).to(out.dtype)
    return out


def _shapelet_coh(gp, u, v, w, freq, freq0, fdelta, tdelta, dec0):
    """Additive coherency of shapelet sources, computed with torch on
    device (mirrors reference.predict_coh for stype==4 sources only)."""
    from .. import shapelet as shmod
    import numpy as np
    M = gp.M
    B = u.shape[0]
    out = torch.zeros(M, B, 2, 2, dtype=torch.complex64, device=u.device)
    co = gp.cluster_off.cpu().numpy()
    pk = gp.pack_ref
    # true (un-zeroed) fluxes at this channel
    full = gp._shfull
    if abs(freq - freq0) < 1.0:
        sIs, sQs, sUs, sVs = (full['sI'], full['sQ'], full['sU'],
                              full['sV'])
    else:
        lf = torch.log(torch.tensor(float(freq), dtype=torch.float64,
                                    device=gp.ll.device) / gp.f0)
        lf = lf.to(torch.float32)
        flog = (gp.spec_idx * lf + gp.spec_idx1 * lf ** 2
                + gp.spec_idx2 * lf ** 3)

        def _scale(s0):
            mag = torch.exp(torch.log(s0.abs().clamp_min(1e-30)) + flog)
            return torch.where(s0 == 0, torch.zeros_like(s0),
                               torch.sign(s0) * mag)
        sIs, sQs, sUs, sVs = (_scale(full['sI0']), _scale(full['sQ0']),
                              _scale(full['sU0']), _scale(full['sV0']))
    for gi, (n0, beta, modes) in gp.shapelets.items():
        ci = int(np.searchsorted(co, gi, side='right') - 1)
        ll = float(gp.ll[gi]); mm = float(gp.mm[gi]); nn1 = float(gp.nn1[gi])
        G = 2.0 * math.pi * (u * ll + v * mm + w * nn1)
        ph = torch.remainder(G * freq, 2.0 * math.pi)
        phc = torch.complex(torch.cos(ph), torch.sin(ph))
        smf = G * (fdelta * 0.5)
        sm = torch.where(smf.abs() > 1e-12,
                         (torch.sin(smf) / smf).abs(),
                         torch.ones_like(smf))
        if tdelta > 0:
            bl = torch.sqrt(u * u + v * v + w * w) * freq
            r1 = math.sqrt(ll * ll + (math.sin(dec0) * mm) ** 2)
            prod = 7.2921150e-5 * tdelta * bl * r1
            sm = sm * torch.where(prod > 1e-12,
                                  1.0645 * torch.erf(0.8326 * prod)
                                  / prod.clamp_min(1e-12),
                                  torch.ones_like(prod))
        envc = shmod.shapelet_contrib(
            u * freq, v * freq, w * freq, float(gp.eX[gi]),
            float(gp.eY[gi]), float(gp.eP[gi]), float(pk.cxi[gi]),
            float(pk.sxi[gi]), float(pk.cphi[gi]), float(pk.sphi[gi]),
            bool(pk.use_proj[gi]), beta, n0, torch.as_tensor(modes))
        term = (phc * sm.to(phc.dtype) * envc).to(torch.complex64)
        # note: use_proj baked as identity when off; contrib uses proj form
        I = float(sIs[gi]); Q = float(sQs[gi])
        U_ = float(sUs[gi]); V_ = float(sVs[gi])
        out[ci, :, 0, 0] += term * (I + Q)
        out[ci, :, 0, 1] += term * complex(U_, V_)
        out[ci, :, 1, 0] += term * complex(U_, -V_)
        out[ci, :, 1, 1] += term * (I - Q)
    return out


class BaselineLayout:
    """Row-structure descriptor for the solver kernels: rows are
    seg*(T*Nbase) + t*Nbase + b with a fixed pair table. Built once per
    tile; pair-index table cached per (N, Nbase)."""

    def __init__(self, bb, Nbase, T, nseg, N, device):
        self.Nbase, self.T, self.nseg, self.N = Nbase, T, nseg, N
        self.pairs = bb[:Nbase].to(device=device,
                                   dtype=torch.int32).contiguous()
        key = (N, Nbase, str(device))
        if key not in _pidx_cache:
            pidx = torch.full((N, N), -1, dtype=torch.int32)
            pcpu = self.pairs.cpu().long()
            ok = (pcpu[:, 0] >= 0) & (pcpu[:, 1] >= 0)
            lo = torch.minimum(pcpu[:, 0], pcpu[:, 1])[ok]
            hi = torch.maximum(pcpu[:, 0], pcpu[:, 1])[ok]
            pidx[lo, hi] = torch.arange(Nbase, dtype=torch.int32)[ok]
            _pidx_cache[key] = pidx.to(device).contiguous()
        self.pidx = _pidx_cache[key]

    def chunk_tab(self, chunk_rows):
        """Per-(seg,t) global chunk id from per-row chunk indices."""
        if chunk_rows is None:
            return torch.zeros(self.nseg * self.T, dtype=torch.int32,
                               device=self.pairs.device)
        return chunk_rows[::self.Nbase].to(torch.int32).contiguous()


def _c64(t):
    return t.to(torch.complex64).reshape(t.shape[0], 4).contiguous()


def jtj_jtr(x, coh, J, bb, N, weights=None, chunk_rows=None, nchunk=1,
            layout=None):
    if layout is None:
        raise RuntimeError("GPU jtj_jtr requires a BaselineLayout "
                           "(structured rows); got none")
    ct = layout.chunk_tab(chunk_rows)
    w32 = weights.to(torch.float32).contiguous() if weights is not None \
        else None
    Jc = J.to(torch.complex64).reshape(-1, 4).contiguous()
    JtJ, Jtr, cost = _ext().jtj_jtr(
        _c64(x), _c64(coh), Jc, layout.pairs, ct, layout.pidx, w32,
        layout.Nbase, layout.T, N, layout.nseg, nchunk)
    return JtJ, Jtr, cost.sum()


def model_cost_per_chunk(x, coh, J, bb, N, weights=None, chunk_rows=None,
                         nchunk=1, layout=None):
    ct = layout.chunk_tab(chunk_rows)
    w32 = weights.to(torch.float32).contiguous() if weights is not None \
        else None
    Jc = J.to(torch.complex64).reshape(-1, 4).contiguous()
    return _ext().model_cost(_c64(x), _c64(coh), Jc, layout.pairs, ct, w32,
                             layout.Nbase, layout.T, N, layout.nseg, nchunk)


def apply_jones(coh, J, bb, chunk_rows=None, layout=None):
    if layout is None:
        # fall back to torch complex math on GPU (cold path)
        from . import reference as R
        return R.apply_jones(coh, J, bb, chunk_rows)
    ct = layout.chunk_tab(chunk_rows)
    Jc = J.to(torch.complex64).reshape(-1, 4).contiguous()
    cohs = _c64(coh).unsqueeze(0)
    out = _ext().apply_jones(None, cohs.reshape(-1, 4), Jc, layout.pairs,
                             ct, layout.Nbase, layout.T, J.shape[1],
                             layout.nseg, 1, 0)
    return out.view(-1, 2, 2)



def lbfgs_cost_grad(x, cohs, J_packed, chunk_off, nchunks, bb, T, Nbase,
                    robust_nu=None, weights=None):
    """Full-parameter cost/gradient on GPU via the grad-only accumulation
    kernel: r = x - sum_ci V_ci (fused residual kernel), then per cluster
    Jtr with x := r + V_ci (so the kernel's internal residual IS the total
    residual); robust scale folded as per-baseline weights. Falls back to
    the torch path when the row structure is absent."""
    from . import reference as R
    B = x.shape[0]
    M = cohs.shape[0]
    N = J_packed.shape[1]
    Mt = J_packed.shape[0]
    dev = x.device
    if B % Nbase != 0 or (B // Nbase) != T:
        return R.lbfgs_cost_grad(x, cohs, J_packed, chunk_off, nchunks,
                                 bb, T, Nbase, robust_nu, weights)
    lay = BaselineLayout(bb, Nbase, T, 1, N, dev)
    ext = _ext()
    Jc = J_packed.to(torch.complex64).reshape(-1, 4).contiguous()
    # per-cluster chunk tables (global chunk ids)
    cts = []
    Vs = []
    Vtot = torch.zeros(B, 4, dtype=torch.complex64, device=dev)
    cohs4 = cohs.to(torch.complex64).reshape(M, B, 4)
    for ci in range(M):
        rows = R.chunk_rows_for(ci, nchunks, T, Nbase, B, dev)
        if rows is None:
            rows = torch.zeros(B, dtype=torch.long, device=dev)
        rows = rows + chunk_off[ci]
        ct = rows[::Nbase].to(torch.int32).contiguous()
        cts.append(ct)
        V = ext.apply_jones(None, cohs4[ci].contiguous(), Jc, lay.pairs,
                            ct, Nbase, T, N, 1, 1, 0)
        Vs.append(V)
        Vtot += V
    x4 = x.to(torch.complex64).reshape(B, 4)
    rtot = x4 - Vtot
    e2 = (rtot.abs() ** 2).sum(dim=1)
    if robust_nu is not None:
        cost = torch.log1p(e2 / robust_nu).sum()
        scale = (1.0 / (robust_nu + e2)).to(torch.float32).contiguous()
    else:
        cost = e2.sum()
        scale = None
    grad = torch.zeros(Mt * N, 4, dtype=torch.complex64, device=dev)
    for ci in range(M):
        xsub = (rtot + Vs[ci]).contiguous()
        gci, _ = ext.jtr_grad(xsub, cohs4[ci].contiguous(), Jc, lay.pairs,
                              cts[ci], scale, Nbase, T, N, 1, Mt)
        grad += gci
    g = -2.0 * torch.view_as_real(grad).reshape(-1)
    return cost.to(torch.float32), g


_chol_scratch = {}


def chol_solve_damped(JtJ, Jtr, mu):
    """dp = (JtJ + mu I)^-1 Jtr via the fused gfx950 kernel (no A
    materialization, no rocSOLVER). Failed factorizations return NaN rows
    so the LM accept mask rejects them. n is padded to a multiple of 32
    with identity diagonal (kernel requires it; pad solutions are 0)."""
    n = JtJ.shape[1]
    npad = (n + 31) // 32 * 32
    if npad != n:
        batch = JtJ.shape[0]
        J2 = torch.zeros(batch, npad, npad, dtype=JtJ.dtype,
                         device=JtJ.device)
        J2[:, :n, :n] = JtJ
        J2.diagonal(dim1=1, dim2=2)[:, n:].fill_(1.0)
        b2 = torch.zeros(batch, npad, dtype=Jtr.dtype, device=Jtr.device)
        b2[:, :n] = Jtr
        JtJ, Jtr = J2, b2
    key = (tuple(JtJ.shape), str(JtJ.device))
    sc = _chol_scratch.get(key)
    want = (JtJ.shape[0], 2 * JtJ.shape[1] * JtJ.shape[2])
    if sc is None or tuple(sc.shape) != want:
        if len(_chol_scratch) > 8:   # bound long multi-config runs
            _chol_scratch.pop(next(iter(_chol_scratch)))
        sc = torch.empty(want, dtype=JtJ.dtype, device=JtJ.device)
        _chol_scratch[key] = sc
    # n >= 256: multi-workgroup right-looking path — the trailing-update
    # SYRK tiles spread over the whole chip instead of 1 WG/problem
    # (65% of step time at [5,512,512]; see profiles/). Small n stays on
    # the single fused kernel (lower launch count wins).
    mw = os.environ.get('SAGECAL_CHOL_MW')
    use_mw = (JtJ.shape[1] >= 384) if mw is None else mw == '1'
    if use_mw:
        # chunked panels (round 2) support up to n=4096
        assert JtJ.shape[1] <= 4096, "mw Cholesky supports n <= 4096"
        dp, info = _ext().chol_solve_mw(JtJ.contiguous(), Jtr.contiguous(),
                                        mu.to(torch.float32).contiguous(),
                                        sc, 4)
    else:
        dp, info = _ext().chol_solve(JtJ.contiguous(), Jtr.contiguous(),
                                     mu.to(torch.float32).contiguous(),
                                     sc, 3)
    # failed factorizations already return NaN rows (kernel poisons dp)
    return dp[:, :n]
