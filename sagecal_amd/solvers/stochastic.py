"""Stochastic (minibatch) calibration with persistent LBFGS, and the
single-node bandpass consensus mode.

Re-implements:
  - bfgsfit_minibatch_visibilities (robust_batchmode_lbfgs.c, API
    Dirac.h:291-349): time-minibatch LBFGS with curvature memory persisted
    across minibatches (persistent_data_t, Dirac.h:84-110);
  - bfgsfit_minibatch_consensus: + consensus term y^T(x-Bz)+rho/2||x-Bz||^2;
  - run_minibatch_consensus_calibration (minibatch_consensus_mode.cpp:47):
    channels split into `nsolbw` mini-bands, each solved stochastically,
    coupled by the frequency polynomial consensus;
  - the federated-averaging variant (sagecal_stochastic_master.cpp:340):
    per-worker Z averaged on the manifold, local solves regularized by
    alpha (find_prod_inverse_full_fed).
"""
import numpy as np
import torch

from ..ops import dispatch as ops
from ..ops import reference as R
from . import lbfgs as lbfgs_mod
from ..consensus import poly as poly_mod


class BandState:
    """Per mini-band persistent solver state (lbfgs_persist_init analog)."""

    def __init__(self, Mt, N, m=7, device='cpu', dtype=torch.complex128):
        rdt = torch.float32 if dtype == torch.complex64 else torch.float64
        eye = torch.eye(2, dtype=dtype, device=device)
        self.J = eye.expand(Mt, N, 2, 2).clone()
        self.mem = lbfgs_mod.LBFGSMemory(m, Mt * N * 8, dtype=rdt,
                                         device=device)
        self.Y = torch.zeros(Mt * N * 8, dtype=rdt, device=device)
        self.nu = 5.0

    def reset(self):
        self.mem.reset()


def _cost_grad_band(v, x, cohs, chunk_off, nchunks, bb, T, Nbase, Mt, N,
                    cdtype, robust_nu, consensus=None):
    """LBFGS cost/grad over the full parameter vector of one band,
    optionally with the consensus augmentation (Dirac.h:325-340)."""
    J = torch.view_as_complex(v.reshape(Mt, N, 2, 2, 2).contiguous())
    cost, grad = ops.lbfgs_cost_grad(x, cohs, J, chunk_off, nchunks, bb, T,
                                     Nbase, robust_nu=robust_nu)
    if consensus is not None:
        y, bz, rho = consensus       # flattened real vectors + scalar/vec
        d = v - bz
        cost = cost + (y * d).sum() + 0.5 * (rho * d * d).sum()
        grad = grad + y + rho * d
    return cost, grad


def _cost_grad_band_multifreq(v, xs, cohs_f, chunk_off, nchunks, bb, T,
                              Nbase, Mt, N, cdtype, robust_nu,
                              consensus=None):
    """Multi-channel LBFGS cost/grad: per-channel residuals summed over
    the mini-band's channels instead of fitting the channel average —
    the role of the reference's fused multifreq gradient kernel
    (lbfgs_multifreq.cu:29 kernel_deriv_r_robust / cudakernel_lbfgs_
    multifreq_r_robust, Dirac.h:523-525). xs: [F, B, 2, 2];
    cohs_f: list of [M, B, 2, 2] per channel."""
    J = torch.view_as_complex(v.reshape(Mt, N, 2, 2, 2).contiguous())
    cost = None
    grad = None
    for fi in range(xs.shape[0]):
        c, g = ops.lbfgs_cost_grad(xs[fi], cohs_f[fi], J, chunk_off,
                                   nchunks, bb, T, Nbase,
                                   robust_nu=robust_nu)
        cost = c if cost is None else cost + c
        grad = g if grad is None else grad + g
    if consensus is not None:
        y, bz, rho = consensus
        d = v - bz
        cost = cost + (y * d).sum() + 0.5 * (rho * d * d).sum()
        grad = grad + y + rho * d
    return cost, grad


def bfgsfit_minibatch(band, x, cohs, bb, T, Nbase, chunk_off, nchunks,
                      lbfgs_iters=6, m=7, robust_nu=None, consensus=None):
    """One minibatch LBFGS fit with persistent memory. Returns final cost."""
    Mt, N = band.J.shape[0], band.J.shape[1]
    cdtype = band.J.dtype

    def fg(v):
        return _cost_grad_band(v, x, cohs, chunk_off, nchunks, bb, T, Nbase,
                               Mt, N, cdtype, robust_nu, consensus)

    v0 = torch.view_as_real(band.J).reshape(-1).clone()
    v1, band.mem, info = lbfgs_mod.lbfgs_fit(
        fg, v0, maxiter=lbfgs_iters, m=m, mem=band.mem, stochastic=True)
    band.J = torch.view_as_complex(
        v1.reshape(Mt, N, 2, 2, 2).contiguous())
    return info


def split_minibatches(T, nmb):
    """Contiguous time-slot minibatches (loadDataMinibatch semantics)."""
    sizes = [(T + nmb - 1 - i) // nmb for i in range(nmb)]
    out = []
    t0 = 0
    for s in sizes:
        if s > 0:
            out.append((t0, t0 + s))
            t0 += s
    return out


class MinibatchConsensusCalibration:
    """Single-node bandpass consensus (minibatch_consensus_mode.cpp):
    channels -> `nsolbw` mini-bands, each with persistent LBFGS state;
    per-epoch minibatch loop; polynomial consensus across the bands.

    With `fed_world/fed_rank` set (distributed federated mode,
    sagecal_stochastic_*.cpp): after each consensus update, per-node Z is
    manifold-averaged across ranks and local z-updates get `+alpha(Zavg-X)`
    (find_prod_inverse_full_fed with alpha)."""

    def __init__(self, pack, N, freqs, nsolbw=2, Npoly=2, poly_type=0,
                 multifreq=False,
                 rho=1.0, device='cpu', dtype=torch.complex128,
                 fed_alpha=0.0, dist_group=None, world=1, rank=0):
        self.pack = pack
        self.N = N
        self.M = pack.M
        self.nchunks = [int(c) for c in pack.nchunk]
        self.chunk_off = np.cumsum([0] + self.nchunks[:-1]).tolist()
        self.Mt = sum(self.nchunks)
        self.device = device
        self.dtype = dtype
        # mini-band channel ranges
        Nchan = len(freqs)
        nsolbw = min(nsolbw, Nchan)
        edges = np.linspace(0, Nchan, nsolbw + 1).astype(int)
        self.bands = [(int(a), int(b)) for a, b in zip(edges[:-1], edges[1:])
                      if b > a]
        self.band_freqs = np.array([np.mean(freqs[a:b])
                                    for a, b in self.bands])
        self.freq0 = float(np.mean(freqs))
        self.B = poly_mod.setup_polynomials(self.band_freqs, self.freq0,
                                            Npoly, poly_type)
        self.Npoly = Npoly
        rho_t = torch.full((self.M, len(self.bands)), float(rho)).double()
        self.rho = rho_t
        self.Bii = poly_mod.find_prod_inverse(self.B, rho_t,
                                              alpha=fed_alpha)
        self.states = [BandState(self.Mt, N, device=device, dtype=dtype)
                       for _ in self.bands]
        P = self.Mt * N * 8
        self.Z = torch.zeros(self.M, Npoly, N, 2, 2, dtype=dtype,
                             device=device)
        self.rho_scalar = float(rho)
        self.fed_alpha = fed_alpha
        self.dist_group = dist_group
        self.world, self.rank = world, rank
        self.Xlag = torch.zeros_like(self.Z)    # federated Lagrange X
        # multifreq: per-channel gradient accumulation across each
        # mini-band (lbfgs_multifreq.cu semantics) instead of fitting
        # the band's channel average — exact for steep bandpasses,
        # costs one predict per channel
        self.multifreq = multifreq

    def _consensus_vec(self, bi):
        """(y, bz) flattened for band bi from current Z and duals."""
        Zb = poly_mod.eval_poly_jones(self.Z, self.B[bi])   # [M,N,2,2]
        # expand clusters to chunks
        bz = torch.cat([Zb[ci:ci + 1].expand(self.nchunks[ci], -1, -1, -1)
                        for ci in range(self.M)])
        return torch.view_as_real(bz.contiguous()).reshape(-1)

    def epoch(self, tile, bb, nmb=2, lbfgs_iters=6, robust_nu=5.0):
        """One epoch over time minibatches for every mini-band."""
        T, Nbase = tile.tilesz, tile.Nbase
        mbs = split_minibatches(T, nmb)
        fdelta_ch = tile.fdelta / len(tile.freqs)
        for (t0, t1) in mbs:
            rows = slice(t0 * Nbase, t1 * Nbase)
            u, v, w = tile.u[rows], tile.v[rows], tile.w[rows]
            bbm = bb[rows]
            Tm = t1 - t0
            for bi, (a, b) in enumerate(self.bands):
                band = self.states[bi]
                bz = self._consensus_vec(bi)
                cons = (band.Y, bz, self.rho_scalar)
                if self.multifreq:
                    self._fit_band_multifreq(band, tile, a, b, rows, bbm,
                                             Tm, Nbase, fdelta_ch,
                                             lbfgs_iters, robust_nu, cons)
                    continue
                f = float(self.band_freqs[bi])
                cohs = ops.predict_coh(self.pack, u, v, w, f, tile.freq0,
                                       fdelta_ch * (b - a), tile.tdelta,
                                       tile.dec0)
                if cohs.dtype != self.dtype:
                    cohs = cohs.to(self.dtype)
                xb = tile.xo[a:b, rows].mean(dim=0).to(self.dtype)
                bfgsfit_minibatch(band, xb, cohs, bbm, Tm, Nbase,
                                  self.chunk_off, self.nchunks,
                                  lbfgs_iters=lbfgs_iters,
                                  robust_nu=robust_nu, consensus=cons)
            self.consensus_update()

    def _fit_band_multifreq(self, band, tile, a, b, rows, bbm, Tm,
                            Nbase, fdelta_ch, lbfgs_iters, robust_nu,
                            cons):
        from . import lbfgs as lbfgs_mod
        u, v, w = tile.u[rows], tile.v[rows], tile.w[rows]
        cohs_f = []
        for fi in range(a, b):
            c = ops.predict_coh(self.pack, u, v, w,
                                float(tile.freqs[fi]), tile.freq0,
                                fdelta_ch, tile.tdelta, tile.dec0)
            cohs_f.append(c.to(self.dtype))
        xs = tile.xo[a:b, rows].to(self.dtype)
        Mt, N = band.J.shape[0], band.J.shape[1]

        def fg(vv):
            return _cost_grad_band_multifreq(
                vv, xs, cohs_f, self.chunk_off, self.nchunks, bbm, Tm,
                Nbase, Mt, N, self.dtype, robust_nu, consensus=cons)
        v0 = torch.view_as_real(band.J).reshape(-1).clone()
        v1, band.mem, info = lbfgs_mod.lbfgs_fit(
            fg, v0, maxiter=lbfgs_iters, m=7, mem=band.mem,
            stochastic=True)
        band.J = torch.view_as_complex(
            v1.reshape(Mt, N, 2, 2, 2).contiguous())

    def consensus_update(self):
        """Global Z from all mini-bands + dual updates
        (update_global_z_multi, minibatch_consensus_mode.cpp:581)."""
        acc = torch.zeros_like(self.Z)
        for bi in range(len(self.bands)):
            band = self.states[bi]
            Yj = torch.view_as_complex(
                band.Y.reshape(self.Mt, self.N, 2, 2, 2).contiguous())
            contrib_c = []
            for ci in range(self.M):
                o = self.chunk_off[ci]
                nc = self.nchunks[ci]
                contrib_c.append((Yj[o:o + nc].mean(dim=0)
                                  + self.rho_scalar
                                  * band.J[o:o + nc].mean(dim=0)))
            contrib = torch.stack(contrib_c)          # [M,N,2,2]
            for p in range(self.Npoly):
                acc[:, p] += float(self.B[bi, p]) * contrib
        if self.fed_alpha > 0:
            acc = acc + self.fed_alpha * (self._fed_avg() - self.Xlag)
        self.Z = poly_mod.update_global_z(acc, self.Bii)
        # dual update per band
        for bi in range(len(self.bands)):
            bz = self._consensus_vec(bi)
            band = self.states[bi]
            j = torch.view_as_real(band.J).reshape(-1)
            band.Y = band.Y + self.rho_scalar * (j - bz)
        if self.fed_alpha > 0:
            self.Xlag = self.Xlag + self.fed_alpha * (self.Z
                                                      - self._fed_avg())

    def _fed_avg(self):
        """Federated manifold average of Z across ranks
        (sagecal_stochastic_master.cpp:340-350). Z coefficient 0 (the DC
        poly term) is averaged on the manifold; higher orders linearly."""
        import torch.distributed as dist
        from . import __init__  # noqa
        if self.world <= 1 or not dist.is_initialized():
            return self.Z
        from ..consensus import manifold
        gathered = [torch.zeros_like(torch.view_as_real(self.Z))
                    for _ in range(self.world)]
        dist.all_gather(gathered, torch.view_as_real(self.Z).contiguous(),
                        group=self.dist_group)
        Zs = [torch.view_as_complex(g) for g in gathered]
        out = torch.zeros_like(self.Z)
        for ci in range(self.M):
            J_bands = torch.stack([Zs[r][ci, 0] for r in range(self.world)])
            proj, _ = manifold.manifold_average_projectback(J_bands)
            out[ci, 0] = proj[self.rank]
            for p in range(1, self.Npoly):
                out[ci, p] = torch.stack(
                    [Zs[r][ci, p] for r in range(self.world)]).mean(dim=0)
        return out

    def global_band_J(self, bi):
        """Consensus solution for mini-band bi: J = sum_p B[bi,p] Z_p
        (the -U use-global path, minibatch_consensus_mode.cpp)."""
        flat = poly_mod.eval_poly_jones(self.Z, self.B[bi]).to(self.dtype)
        out = torch.zeros(self.Mt, self.N, 2, 2, dtype=self.dtype,
                          device=self.device)
        for ci in range(self.M):
            o = self.chunk_off[ci]
            out[o:o + self.nchunks[ci]] = flat[ci]
        return out

    def residuals(self, tile, bb, use_global=False):
        """Per-channel residuals using each channel's mini-band solution
        (or the consensus polynomial solution when use_global)."""
        out = torch.empty_like(tile.xo)
        T, Nbase = tile.tilesz, tile.Nbase
        fdelta_ch = tile.fdelta / len(tile.freqs)
        for fi, f in enumerate(tile.freqs):
            bi = next(i for i, (a, b) in enumerate(self.bands)
                      if a <= fi < b)
            cohs = ops.predict_coh(self.pack, tile.u, tile.v, tile.w,
                                   float(f), tile.freq0, fdelta_ch,
                                   tile.tdelta, tile.dec0)
            if cohs.dtype != self.dtype:
                cohs = cohs.to(self.dtype)
            V = torch.zeros(tile.x.shape, dtype=self.dtype,
                            device=self.device)
            J = self.global_band_J(bi) if use_global \
                else self.states[bi].J
            for ci in range(self.M):
                rows = R.chunk_rows_for(ci, self.nchunks, T, Nbase,
                                        tile.x.shape[0], self.device)
                o = self.chunk_off[ci]
                V += R.apply_jones(cohs[ci],
                                   J[o:o + self.nchunks[ci]], bb, rows)
            out[fi] = tile.xo[fi] - V
        return out
