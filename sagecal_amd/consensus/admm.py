"""Multi-band consensus ADMM — the sagecal-mpi path, MI355X-native.

Re-implements the semantics of /root/reference/src/MPI/sagecal_master.cpp
(per-timeslot ADMM loop :694-1060) and sagecal_slave.cpp (:643-979) with a
REPLICATED-MASTER design over collectives: one rank per
GPU/sub-band on RCCL over xGMI (gloo on CPU for tests); the hub's
TAG_YDATA gather + Z-update + TAG_CONSENSUS scatter collapse into ONE
fused all-reduce of the B-weighted accumulator (Npoly x 8NM values), with
the per-cluster Npoly x Npoly solve replicated on every rank. Control tags
disappear (SPMD loop); Zold/Yhat are rank-local; BB rho updates piggyback
on a tiny all-reduce.
"""
import numpy as np
import torch
import torch.distributed as dist

from . import poly
from ..solvers import sage


class ConsensusADMM:
    """Per-rank consensus ADMM driver.

    Parameters
    ----------
    state : sage.CalState        this rank's calibration state
    freqs_all : [W] float        every rank's band centre frequency
    rank, world : ints           this rank / world size
    Npoly, poly_type             frequency basis config (-P / -Q)
    rho : [M] tensor             per-cluster regularization (-r / -G)
    use_bb : bool                Barzilai-Borwein adaptive rho (-a analog)
    group : torch.distributed group or None (single-process: world==1)
    """

    def __init__(self, state, freqs_all, freq0, rank, world, Npoly=2,
                 poly_type=0, rho=None, use_bb=False, rho_upper=1e3,
                 group=None, federated_alpha=0.0, spatial=None,
                 spatial_alpha=0.0, centroids=None):
        self.state = state
        self.rank, self.world = rank, world
        self.group = group
        self.Npoly = Npoly
        M, N, Mt = state.M, state.N, state.Mt
        dev = state.J.device
        self.dev = dev
        self.cdtype = state.J.dtype
        self.B = poly.setup_polynomials(freqs_all, freq0, Npoly, poly_type)
        self.Bf = self.B[rank]               # [Npoly] this rank's row
        if rho is None:
            rho = torch.full((M,), 5.0)
        self.rho = rho.clone().double()      # [M]
        self.rho_upper = rho_upper
        self.use_bb = use_bb
        self.alpha = federated_alpha
        self.fratio = torch.ones(world)      # per-band flag-ratio weights
        self._update_bii()
        # consensus state (per rank): Y dual, Z poly coefficients.
        # Z is indexed by EFFECTIVE cluster (Mt, hybrid chunks expanded),
        # matching the reference master where iodata.M = worker Mt and
        # the Z file has Mt columns (sagecal_master.cpp:250,517,1168).
        self.Y = torch.zeros(Mt, N, 2, 2, dtype=self.cdtype, device=dev)
        self.Z = torch.zeros(Mt, Npoly, N, 2, 2, dtype=self.cdtype,
                             device=dev)
        self.Zold = None
        self.Yhat_prev = None
        self.J_prev = None
        # spatial regularization (-X / -u, sagecal_master.cpp:293-423 +
        # :887-985): spatial = (lam, mu, order, fista_iters, cadence)
        self.spatial = spatial
        self.spatial_alpha = spatial_alpha
        self.centroids = centroids          # (ll[M], mm[M])
        self.Zspat = None
        self.Xs = torch.zeros_like(self.Z)  # Lagrange multiplier
        self._Phi = None
        if spatial is not None and centroids is not None:
            from . import fista as fista_mod
            lam, mu_l1, order, fiters, cadence = spatial
            # centroids come per user cluster; the spatial basis rows are
            # per EFFECTIVE cluster (ll/mm are length iodata.M = Mt in
            # sagecal_master.cpp:523-529) — chunk-expand
            nch = state.nchunks
            ll = np.concatenate([[centroids[0][ci]] * int(nch[ci])
                                 for ci in range(M)])
            mm = np.concatenate([[centroids[1][ci]] * int(nch[ci])
                                 for ci in range(M)])
            self.centroids = (ll, mm)
            beta = float(max(np.max(np.abs(ll)), np.max(np.abs(mm)), 1e-3))
            self._Phi = fista_mod.spatial_basis(ll, mm, order, beta)

    def _rho_eff(self):
        """rho expanded to effective clusters [Mt] (double, CPU) — the
        master's arho expansion via chunkvec (sagecal_master.cpp:598)."""
        st = self.state
        if st.Mt == st.M:
            return self.rho
        return torch.cat([
            torch.full((int(st.nchunks[ci]),), float(self.rho[ci]),
                       dtype=torch.float64) for ci in range(st.M)])

    def _update_bii(self):
        # per-band effective rho = rho * fratio_f (sagecal_master.cpp:720:
        # bands with more flagged data weigh less in the Z fit);
        # Bii per EFFECTIVE cluster [Mt, Npoly, Npoly]
        rho_mf = self._rho_eff()[:, None] * self.fratio[None, :].double()
        self.Bii = poly.find_prod_inverse(self.B, rho_mf, self.alpha)

    def set_fratio(self, unflagged_fraction):
        """Exchange per-band unflagged-data fractions (TAG_FRATIO) and
        rebuild the Bii inverses with fratio-scaled rho."""
        # collectives must run on the backend's device (NCCL: cuda)
        fr = torch.zeros(self.world, device=self.dev)
        fr[self.rank] = float(unflagged_fraction)
        if self.world > 1 and dist.is_initialized():
            dist.all_reduce(fr, group=self.group)
        else:
            fr[:] = float(unflagged_fraction)
        self.fratio = fr.cpu().clamp_min(1e-3)
        self._update_bii()

    def _rho_chunk(self):
        st = self.state
        if st.Mt == st.M:
            return self.rho.to(device=self.dev, dtype=torch.float32
                               if self.cdtype == torch.complex64
                               else torch.float64)
        return torch.cat([
            torch.full((st.nchunks[ci],), float(self.rho[ci]))
            for ci in range(st.M)]).to(self.dev)

    def _allreduce(self, t):
        if self.world > 1 and dist.is_initialized():
            dist.all_reduce(t, group=self.group)
        return t

    def z_update(self, first=False):
        """Global Z from all bands: fused all-reduce of B_f (x) (Y + rho J)
        per cluster (collapsed TAG_YDATA + update_global_z_multi +
        TAG_CONSENSUS, sagecal_master.cpp:813-877)."""
        st = self.state
        # per effective cluster [Mt]: no chunk-averaging — each hybrid
        # chunk is its own consensus variable, like the reference master
        Jm = st.J
        Ym = self.Y
        rho_m = (self._rho_eff() * float(self.fratio[self.rank])).to(
            device=self.dev).to(Jm.real.dtype)
        if first:
            # admm==0: Y already holds rho * (gauge-unified J); the Z fit
            # uses it alone (sagecal_master.cpp:843-877 at admm==0)
            contrib = Ym
        else:
            contrib = Ym + rho_m[:, None, None, None].to(self.cdtype) * Jm
        Bfd = self.Bf.to(device=self.dev, dtype=Jm.real.dtype)
        acc = (Bfd[None, :, None, None, None].to(self.cdtype)
               * contrib[:, None]).contiguous()
        self._allreduce(torch.view_as_real(acc))
        if self.spatial is not None and self.Zspat is not None:
            # augmented spatial constraint: z += alpha (Zspat - Xs); the
            # Bii inverses already carry +alpha I (federated machinery)
            acc = acc + self.spatial_alpha * (self.Zspat - self.Xs)
        self.Z = poly.update_global_z(acc, self.Bii)
        return self.Z

    def bz(self):
        """This band's consensus target B_f Z, per effective cluster
        [Mt, N, 2, 2]."""
        return poly.eval_poly_jones(self.Z, self.Bf).to(self.cdtype)

    def y_update(self, BZ, first=False):
        """Y <- Y + rho (J - BZ) (sagecal_slave.cpp:870-888). At admm==0
        Y already holds rho*Jbar (the unified gauge), so the update is
        Y <- Y - rho BZ (slave :856-874: 'we already have Y + rho J')."""
        st = self.state
        rho_chunk = self._rho_chunk()
        rc = rho_chunk[:, None, None, None].to(self.cdtype)
        if first:
            self.Y = self.Y - rc * BZ
        else:
            self.Y = self.Y + rc * (st.J - BZ)

    def bb_update(self, BZ_old):
        """Barzilai-Borwein rho (sagecal_slave.cpp:899-904 +
        consensus_poly.c:928): Yhat = Y + rho (J - B Zold); deltas vs the
        previous iteration, averaged across bands by all-reduce."""
        st = self.state
        rho_chunk = self._rho_chunk()
        Yhat = self.Y + rho_chunk[:, None, None, None].to(self.cdtype) * \
            (st.J - BZ_old)
        if self.Yhat_prev is not None:
            M = st.M
            # inner products over the CONCATENATED per-cluster chunk
            # deltas (rho_bb_threadfn walks 8*N*nchunk contiguous values
            # per cluster, consensus_poly.c:934-977)
            dYt = torch.view_as_real(Yhat - self.Yhat_prev).reshape(
                st.Mt, -1).double().cpu()
            dJt = torch.view_as_real(st.J - self.J_prev).reshape(
                st.Mt, -1).double().cpu()
            ip11c = (dYt * dYt).sum(dim=1)
            ip12c = (dYt * dJt).sum(dim=1)
            ip22c = (dJt * dJt).sum(dim=1)
            ip11 = torch.zeros(M, dtype=torch.float64)
            ip12 = torch.zeros(M, dtype=torch.float64)
            ip22 = torch.zeros(M, dtype=torch.float64)
            for ci in range(M):
                o, nc = st.chunk_off[ci], int(st.nchunks[ci])
                ip11[ci] = ip11c[o:o + nc].sum()
                ip12[ci] = ip12c[o:o + nc].sum()
                ip22[ci] = ip22c[o:o + nc].sum()
            new_rho = poly.update_rho_bb_ip(self.rho, self.rho_upper,
                                            ip11, ip12, ip22)
            # keep rho consistent across ranks (mean); reduce on the
            # backend's device (NCCL: cuda)
            if self.world > 1 and dist.is_initialized():
                nr = new_rho.to(self.dev)
                dist.all_reduce(nr, group=self.group)
                new_rho = (nr / self.world).cpu()
            self.rho = new_rho
            self._update_bii()
        self.Yhat_prev = Yhat.clone()
        self.J_prev = self.state.J.clone()

    def run(self, cohs, tile, bb, opts, n_admm=10, flags=None,
            verbose=False, diffuse_hook=None):
        """The per-tile ADMM loop (sagecal_master.cpp:731-1060 semantics).
        Returns (res0, res1) of the final local solve.

        diffuse_hook(adm, cohs): called after each spatial-model update
        so the designated diffuse cluster's coherencies are re-predicted
        with the spatial model applied (recalculate_diffuse_coherencies
        at the admm cadence, sagecal_slave.cpp:669-694)."""
        res0 = res1 = None
        st = self.state
        for it in range(n_admm):
            BZ_old = self.bz() if it > 0 else None
            admm_terms = None
            if it > 0:
                admm_terms = (self.rho.to(self.dev), self.Y, BZ_old)
            r0, r1 = sage.sagefit(st, cohs, tile, bb, opts, flags=flags,
                                  admm_terms=admm_terms)
            if res0 is None:
                res0 = r0
            res1 = r1
            # pre-consensus divergence guard (sagecal_slave.cpp:798-803):
            # a blown-up local solve must not contaminate the global Z
            if not (r1 == r1) or (r0 > 0 and r1 > 5.0 * r0):
                st.reset()
            if it == 0:
                # gauge unification (sagecal_master.cpp:827 at admm==0):
                # manifold-average J across bands and seed Y = rho Jbar so
                # the first Z fit sees one common unitary frame
                self._unify_gauge()
            self.z_update(first=(it == 0))
            BZ = self.bz()
            self.y_update(BZ, first=(it == 0))
            if self.use_bb and BZ_old is not None:
                self.bb_update(BZ_old)
            if self.spatial is not None and self._Phi is not None:
                lam, mu_l1, order, fiters, cadence = self.spatial
                if (it + 1) % max(cadence, 1) == 0:
                    self.spatial_update(lam, mu_l1, fiters)
                    if diffuse_hook is not None:
                        diffuse_hook(self, cohs)
            if verbose:
                # primal ||J - BZ|| and dual ||Z - Zold|| residual norms
                # (sagecal_master.cpp:881-885 / sagecal_slave.cpp:911-919)
                primal = float((st.J - BZ).abs().norm()) / max(st.Mt, 1)
                dual = 0.0 if BZ_old is None else \
                    float((BZ - BZ_old).abs().norm()) / max(st.Mt, 1)
                print(f"ADMM {it}: res {r1:.6f} primal {primal:.3e} "
                      f"dual {dual:.3e} rho[0] {float(self.rho[0]):.2f}")
        return res0, res1

    def _unify_gauge(self):
        """Replicated analog of calculate_manifold_average at admm==0:
        all-gather every band's J, average each cluster-chunk over bands
        up to the common unitary, and seed Y = rho * Jbar (the slaves'
        received 'unified Y', sagecal_slave.cpp:836-841). For world==1
        the average is exactly J (polar of a PSD Gram is identity), so
        seed directly — no gather, no sync-heavy averaging.

        The multi-band average is fully batched over (band, chunk):
        A_{f,t} = sum_s J_{f,t,s}^H Jbar_{t,s}; U = polar(A); aligned
        mean — no per-station python loops (manifold_average.c:204
        vectorized)."""
        st = self.state
        rho_chunk = self._rho_chunk().to(self.dev)
        if self.world == 1 or not dist.is_initialized():
            self.Y = rho_chunk[:, None, None, None].to(self.cdtype) * st.J
            return
        from .manifold import polar_unitary
        # gather on the backend's device (NCCL: cuda), average on CPU
        Jr = torch.view_as_real(st.J).contiguous()
        gath = [torch.zeros_like(Jr) for _ in range(self.world)]
        dist.all_gather(gath, Jr, group=self.group)
        Jall = torch.stack([torch.view_as_complex(g).cpu()
                            .to(torch.complex128)
                            for g in gath])            # [F, T, N, 2, 2]
        Jbar = Jall[0].clone()
        for _ in range(5):
            A = torch.einsum('ftsji,tsjk->ftik', Jall.conj(), Jbar)
            U = polar_unitary(A.reshape(-1, 2, 2)).reshape(
                self.world, st.Mt, 2, 2)
            aligned = torch.einsum('ftsij,ftjk->ftsik', Jall, U)
            Jbar = aligned.mean(dim=0)
        self.Y = (rho_chunk.cpu().double()[:, None, None, None]
                  * Jbar).to(device=self.dev, dtype=self.cdtype)

    def spatial_update(self, lam, mu_l1, fiters):
        """Fit the spatial (elastic-net shapelet) model to the per-cluster
        Z and form the smoothed constraint Zspat + multiplier update
        (sagecal_master.cpp:887-985)."""
        from . import fista as fista_mod
        M = self.state.Mt            # spatial rows per effective cluster
        Zb = torch.view_as_real(self.Z).reshape(M, -1)
        Zb = torch.complex(Zb[:, 0::2], Zb[:, 1::2])   # [M, K]
        Zsp = fista_mod.update_spatialreg_fista(
            Zb.cpu(), self._Phi, lam=lam, mu=mu_l1, maxiter=fiters)
        Zhat = (self._Phi.to(Zsp.dtype) @ Zsp.T)       # [M, K]
        flat = torch.empty(M, Zhat.shape[1] * 2, dtype=torch.float64)
        flat[:, 0::2] = Zhat.real
        flat[:, 1::2] = Zhat.imag
        self.Zspat = torch.view_as_complex(
            flat.reshape(*self.Z.shape, 2).contiguous()).to(
                device=self.dev, dtype=self.cdtype)
        if self.spatial_alpha > 0:
            self.Xs = self.Xs + self.spatial_alpha * (self.Z - self.Zspat)
            # rebuild Bii with the +alpha I term
            self.alpha = self.spatial_alpha
            self._update_bii()

    def spatial_coefficients(self):
        """FISTA spatial-model coefficient matrix Zsp [P, G] (P = the
        flattened Npoly*N*4 complex parameters, G = basis modes) — the
        payload of the master's spatial_<solfile> write
        (sagecal_master.cpp:1176-1186)."""
        if self.Zspat is None or self._Phi is None:
            return None
        from . import fista as fista_mod
        M = self.state.Mt
        lam, mu_l1, order, fiters, cadence = self.spatial
        Zb = torch.view_as_real(self.Z).reshape(M, -1)
        Zb = torch.complex(Zb[:, 0::2], Zb[:, 1::2])
        return fista_mod.update_spatialreg_fista(
            Zb.cpu(), self._Phi, lam=lam, mu=mu_l1, maxiter=fiters)

    def diffuse_station_series(self):
        """Per-station Jones-valued spatial model as a shapelet series
        [N, G, 2, 2] with its scale (for this band): the FISTA model Zsp
        evaluated with this band's polynomial basis. Coefficients come in
        the image_basis normalization (4 pi^2/beta^2 prefactor, argument
        2 pi l / beta), so the raw-phi series scale is beta/(2 pi)
        (sagecal_master.cpp spatial model -> diffuse_predict.c hand-off)."""
        Zsp = self.spatial_coefficients()
        if Zsp is None:
            return None, None
        N = self.Z.shape[2]
        Npoly = self.Npoly
        G = Zsp.shape[1]
        Zsp = Zsp.reshape(Npoly, N, 2, 2, G)
        Bf = self.Bf.to(Zsp.dtype)
        Zband = torch.einsum('p,pnijg->ngij', Bf, Zsp)   # [N, G, 2, 2]
        beta = float(max(np.max(np.abs(self.centroids[0])),
                         np.max(np.abs(self.centroids[1])), 1e-3))
        C0 = 4.0 * np.pi ** 2 / beta ** 2
        return Zband * C0, beta / (2.0 * np.pi)

    def diffuse_coherencies(self, u, v, w, bb, Cm, beta_c, lmn, freq,
                            fdelta):
        """Diffuse-cluster coherencies with the spatial model applied
        (recalculate_diffuse_coherencies, diffuse_predict.c:295):
        Zp C Zq^H per baseline via Jones shapelet products."""
        from .. import shapelet as shmod
        Z, bz = self.diffuse_station_series()
        if Z is None:
            return None
        ll, mm, nn1 = lmn
        return shmod.recalculate_diffuse_coherencies(
            u.cpu().double(), v.cpu().double(), w.cpu().double(),
            bb.cpu(), Z, bz, Cm, beta_c, ll, mm, nn1, freq, fdelta)

    def global_solution(self):
        """J = B_f Z (use_global_solution path, sagecal_master:1064)."""
        return self.bz()


class MultiplexedADMM:
    """Consensus ADMM with MORE frequency bands than ranks: each rank
    owns a slice of the MS list and rotates through them, one local
    solve per ADMM iteration per rank (sagecal_master.cpp Scurrent
    rotation :1055-1060; slave mmid multiplexing). The Z fit always sums
    over ALL bands — non-current bands contribute their cached
    (Y + rho J) — mirroring the master's persistent per-MS Y buffers.

    band_sets: list of per-band dicts owned by THIS rank, each with keys
    'state' (CalState), 'freq0'; freqs_all covers every band globally;
    my_band_ids are this rank's global band indices.
    """

    def __init__(self, band_sets, my_band_ids, freqs_all, freq0, rank,
                 world, Npoly=2, poly_type=0, rho=None, group=None):
        self.bands = band_sets
        self.ids = list(my_band_ids)
        self.rank, self.world = rank, world
        self.group = group
        self.F = len(freqs_all)
        self.Npoly = min(Npoly, self.F)
        st0 = band_sets[0]['state']
        self.M, self.N, self.Mt = st0.M, st0.N, st0.Mt
        self.dev = st0.J.device
        self.cdtype = st0.J.dtype
        self.B = poly.setup_polynomials(freqs_all, freq0, self.Npoly,
                                        poly_type)
        if rho is None:
            rho = torch.full((self.M,), 5.0)
        self.rho = rho.clone().double()
        # Z/Bii per EFFECTIVE cluster (reference master iodata.M = Mt)
        rho_mf = self._rho_eff()[:, None].expand(-1, self.F)
        self.Bii = poly.find_prod_inverse(self.B, rho_mf)
        for b in self.bands:
            b['Y'] = torch.zeros(self.Mt, self.N, 2, 2,
                                 dtype=self.cdtype, device=self.dev)
        self.Z = torch.zeros(self.Mt, self.Npoly, self.N, 2, 2,
                             dtype=self.cdtype, device=self.dev)
        self.cur = 0

    def _rho_eff(self):
        st = self.bands[0]['state']
        if st.Mt == st.M:
            return self.rho
        return torch.cat([
            torch.full((int(st.nchunks[ci]),), float(self.rho[ci]),
                       dtype=torch.float64) for ci in range(st.M)])

    def _rho_chunk(self, st):
        return self._rho_eff().to(self.dev)

    def bz(self, bi, st):
        return poly.eval_poly_jones(self.Z, self.B[bi]).to(self.cdtype)

    def z_update(self):
        """Allreduce of sum over ALL owned bands of B_b (x) (Y_b +
        rho J_b) — stale contributions for the bands not solved this
        iteration, like the master's persistent Y (sagecal_master.cpp)."""
        acc = torch.zeros(self.Mt, self.Npoly, self.N, 2, 2,
                          dtype=self.cdtype, device=self.dev)
        rho_m = self._rho_eff().to(self.dev).to(torch.float64)
        for bi, b in zip(self.ids, self.bands):
            st = b['state']
            Jm = st.J
            Ym = b['Y']
            contrib = Ym + rho_m[:, None, None, None].to(self.cdtype) * Jm
            Bb = self.B[bi].to(torch.float64)
            acc = acc + (Bb[None, :, None, None, None].to(self.cdtype)
                         * contrib[:, None])
        if self.world > 1 and dist.is_initialized():
            dist.all_reduce(torch.view_as_real(acc), group=self.group)
        self.Z = poly.update_global_z(acc, self.Bii)

    def run(self, tiles, opts, n_admm=10):
        """tiles: list aligned with band_sets of dicts holding 'cohs',
        'tile', 'bb'. One rotating local solve per ADMM iteration."""
        res = {}
        for it in range(n_admm):
            k = self.cur
            b = self.bands[k]
            bi = self.ids[k]
            st = b['state']
            t = tiles[k]
            admm_terms = None
            if it >= len(self.bands):    # every band solved once already
                BZ = self.bz(bi, st)
                admm_terms = (self.rho.to(self.dev), b['Y'], BZ)
            r0, r1 = sage.sagefit(st, t['cohs'], t['tile'], t['bb'],
                                  opts, admm_terms=admm_terms)
            res.setdefault(bi, [r0, r1])[1] = r1
            self.z_update()
            BZ = self.bz(bi, st)
            rc = self._rho_chunk(st)[:, None, None, None].to(self.cdtype)
            b['Y'] = b['Y'] + rc * (st.J - BZ)
            # rotate to the next owned band (Scurrent increment)
            if len(self.bands) > 1:
                self.cur = (self.cur + 1) % len(self.bands)
        return res
