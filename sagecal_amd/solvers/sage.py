"""SAGE (Space-Alternating Generalized EM) calibration driver.

Re-implements /root/reference/src/lib/Dirac/lmfit.c sagefit_visibilities
(lmfit.c:777-1053): the EM loop over direction clusters — per cluster "add
own model back to residual, solve, subtract new model" — with hybrid
time-chunk solutions, robust (Student's-t) IRLS wrapping, and a final joint
LBFGS polish.

MI355X-first re-architecture: two modes
  * 'sequential' — faithful Gauss-Seidel EM (cluster ci sees updated
    models of clusters < ci), like the reference CPU path;
  * 'batched'    — ALL clusters' solves proceed simultaneously against the
    shared residual (Jacobi-style EM). This replaces the reference's
    2-GPU/2-cluster pthread pipeline (lmfit_cuda.c:451-575) with one
    batched solver that fills a 256-CU GPU; with per-EM-iteration residual
    refresh it converges to the same fixed point (tests/test_sage.py).
"""
import torch

from ..ops import dispatch as ops
from ..ops import reference as R
from . import lm as lm_mod
from ..constants import (ROBUST_MODES, SM_RLM_RLBFGS, SM_RTR_OSLM_LBFGS,
                         SM_RTR_OSRLM_RLBFGS, SM_NSD_RLBFGS, NU_LOW,
                         NU_HIGH, NU_GRID, LMCUT)


class SageSolveOptions:
    """Solver options. `em_group` sets how many clusters solve together as
    one batched LM problem per EM step: 1 = pure sequential Gauss-Seidel EM
    (reference CPU path), M = pure Jacobi (all clusters at once); groups are
    Gauss-Seidel across, Jacobi within — the generalization of the
    reference's pairwise 2-GPU pipeline (lmfit_cuda.c:803-887, which is
    em_group=2). Larger groups fill the GPU better but converge slightly
    slower per sweep; group size is a throughput/convergence dial."""

    def __init__(self, max_emiter=3, max_iter=15, solver_mode=SM_RLM_RLBFGS,
                 robust_nulow=NU_LOW, robust_nuhigh=NU_HIGH,
                 robust_outer=3, lbfgs_iters=0, lbfgs_minibatch=0,
                 mode='batched', em_group=None, linsolv=0, nsubsets=0,
                 joint_iters=0, randomize=False):
        self.max_emiter = max_emiter
        self.max_iter = max_iter
        self.solver_mode = solver_mode
        self.robust = solver_mode in ROBUST_MODES
        self.robust_nulow = robust_nulow
        self.robust_nuhigh = robust_nuhigh
        self.robust_outer = robust_outer
        self.lbfgs_iters = lbfgs_iters
        self.mode = mode
        # group size 2 matches the reference's 2-GPU pairwise pipeline and
        # measures equal to sequential in convergence per sweep; large
        # groups (Jacobi) can diverge on strongly-overlapping clusters
        # (guarded in _solve_group).
        self.em_group = em_group or (1 if mode == 'sequential' else 2)
        self.linsolv = linsolv
        self.nsubsets = nsubsets  # >0: ordered-subsets acceleration
        # -R: alternate EM sweeps reallocate LM iterations toward the
        # groups whose cost dropped most (lmfit.c:871-879 weighted_iter;
        # off by default — the batched design converges without it, and
        # constant iteration counts keep hipGraph reuse perfect)
        self.randomize = randomize
        # OS solver modes imply subsets (oslevmar_*, reference modes 1/2)
        from ..constants import SM_OSLM_LBFGS, SM_OSLM_OSRLM_RLBFGS
        if nsubsets == 0 and solver_mode in (SM_OSLM_LBFGS,
                                             SM_OSLM_OSRLM_RLBFGS):
            self.nsubsets = 2
        # joint cross-cluster LM refinement iterations after the EM sweeps
        # (quadratic local convergence; not in the reference, which only has
        # the LBFGS polish)
        self.joint_iters = joint_iters


class CalState:
    """Per-MS persistent calibration state: packed Jones params + chunk
    bookkeeping (plays the role of the reference's p/pinit arrays and
    ptoclus map, fullbatch_mode.cpp:146-231)."""

    def __init__(self, pack, N, device='cpu', dtype=torch.complex128):
        self.N = N
        self.M = pack.M
        self.nchunks = [int(c) for c in pack.nchunk]
        self.chunk_off = []
        off = 0
        for nc in self.nchunks:
            self.chunk_off.append(off)
            off += nc
        self.Mt = off
        # init J = identity per station per chunk (fullbatch_mode.cpp:207)
        eye = torch.eye(2, dtype=dtype, device=device)
        self.J = eye.expand(self.Mt, N, 2, 2).clone()
        self.nu = torch.full((self.M,), 2.0)

    def cluster_J(self, ci):
        o = self.chunk_off[ci]
        return self.J[o:o + self.nchunks[ci]]

    def set_cluster_J(self, ci, Jc):
        o = self.chunk_off[ci]
        self.J[o:o + self.nchunks[ci]] = Jc

    def reset(self):
        eye = torch.eye(2, dtype=self.J.dtype, device=self.J.device)
        self.J = eye.expand(self.Mt, self.N, 2, 2).clone()


def precalc_coherencies(pack, tile, device=None, discrete=True):
    """Channel-averaged coherencies for the solve.

    discrete=True (default): the EXACT model of channel-averaged data —
    the mean of per-channel coherencies over the tile's actual channel
    grid (each with its own channel-width smearing). The reference instead
    predicts once at freq0 with a continuous |sinc| bandwidth-smearing
    factor (precalculate_coherencies, predict.c:503) — an approximation
    whose Dirichlet-vs-sinc mismatch puts a floor under the residuals of
    wide sub-bands; set discrete=False for reference-equivalent behavior.
    Predict cost is x Nchan, amortized once per tile."""
    u, v, w = tile.u, tile.v, tile.w
    freqs = getattr(tile, 'freqs', None)
    if not discrete or freqs is None or len(freqs) <= 1:
        return ops.predict_coh(pack, u, v, w, tile.freq0, tile.freq0,
                               tile.fdelta, tile.tdelta, tile.dec0)
    fdelta_ch = tile.fdelta / len(freqs)
    acc = None
    for f in freqs:
        c = ops.predict_coh(pack, u, v, w, float(f), tile.freq0,
                            fdelta_ch, tile.tdelta, tile.dec0)
        acc = c if acc is None else acc + c
    return acc / len(freqs)


def _layout_for(state, bb, T, Nbase, nseg, device):
    """BaselineLayout for GPU kernels (None on CPU). Cached per nseg."""
    if not (hasattr(bb, 'is_cuda') and bb.is_cuda):
        return None
    import os
    if os.environ.get('SAGECAL_FORCE_REFERENCE') == '1':
        return None
    from ..ops.hip_host import BaselineLayout
    cache = getattr(state, '_lay_cache', None)
    if cache is None:
        cache = state._lay_cache = {}
    key = (nseg, T, Nbase)
    if key not in cache:
        cache[key] = BaselineLayout(bb, Nbase, T, nseg, state.N, device)
    return cache[key]


def _model_cluster(state, ci, coh_ci, bb, T, Nbase, B, lay=None):
    rows = R.chunk_rows_for(ci, state.nchunks, T, Nbase, B, coh_ci.device)
    if lay is None:
        lay = _layout_for(state, bb, T, Nbase, 1, coh_ci.device)
    # per-cluster chunk rows are LOCAL here; offset to the cluster's global
    # chunk range for the packed-J kernels
    rows_g = rows
    Jc = state.cluster_J(ci)
    return ops.apply_jones(coh_ci, Jc, bb, rows_g, lay), rows


def total_model(state, cohs, bb, T, Nbase, skip=()):
    """Sum of J_p C J_q^H over clusters; `skip` indices are left out
    (negative-cluster-id semantics: residual.c:74 subtracts only
    clusters with id >= 0 — target-field clusters keep their flux in
    the residual)."""
    B = cohs.shape[1]
    V = torch.zeros_like(cohs[0])
    for ci in range(state.M):
        if ci in skip:
            continue
        Vc, _ = _model_cluster(state, ci, cohs[ci], bb, T, Nbase, B)
        V = V + Vc
    return V


def _inner_solve(prob, J, opts, maxiter):
    """Dispatch one deterministic inner solve by solver mode, honoring the
    reference's LMCUT heuristic (sagecalmain.h:24: N <= 40 -> RTR/NSD
    replaced by LM) and ordered-subsets acceleration for the OS modes
    (oslevmar_*; the graphed robust path solves full-batch instead —
    a strictly more accurate iteration at the same cost there)."""
    mode = opts.solver_mode
    if prob.N > LMCUT and mode in (SM_RTR_OSLM_LBFGS,
                                   SM_RTR_OSRLM_RLBFGS):
        from . import rtr as rtr_mod
        return rtr_mod.rtr_solve(prob, J, maxiter=maxiter)[0]
    if prob.N > LMCUT and mode == SM_NSD_RLBFGS:
        from . import rtr as rtr_mod
        return rtr_mod.nsd_solve(prob, J, maxiter=2 * maxiter)[0]
    if getattr(opts, 'nsubsets', 0) > 1:
        return lm_mod.os_lm_solve(prob, J, maxiter=maxiter,
                                  nsubsets=opts.nsubsets)[0]
    return lm_mod.lm_solve(prob, J, maxiter=maxiter)[0]


def robust_lm(prob, J0, nu0, opts):
    """IRLS Student's-t wrapper (robustlm.c rlevmar_der_single_* /
    rtr_solve_robust.c): alternate the weighted inner solver with weight +
    nu AECM updates."""
    import os as _os
    nu = nu0 if nu0 > 0 else 2.0
    inner = max(3, opts.max_iter // max(1, opts.robust_outer))
    graph_ok = (opts.robust_outer == 1 and prob.layout is not None
                and prob.x.is_cuda
                and (opts.solver_mode != 5 or prob.N <= 40)
                and _os.environ.get('SAGECAL_NO_GRAPH') != '1')
    if graph_ok:
        # single-IRLS-pass schedule: whole solve as one graph replay; the
        # nu grid update runs on the host from the captured mean(log w - w)
        J, logsumw, _ = lm_mod.robust_lm_graphed(prob, J0, nu, inner)
        import math as _math
        dgm = float(torch.special.digamma(torch.tensor((nu + 8) * 0.5)))
        dgm -= _math.log((nu + 8) * 0.5)
        grid = torch.linspace(opts.robust_nulow, opts.robust_nuhigh,
                              NU_GRID)
        q = (-torch.special.digamma(grid * 0.5) + torch.log(grid * 0.5)
             + logsumw + dgm + 1.0)
        nu = float(grid[torch.argmin(q.abs())])
        return J, nu
    J = J0
    for outer in range(opts.robust_outer):
        J = _inner_solve(prob, J, opts, inner)
        V = ops.apply_jones(prob.coh, J, prob.bb, prob.chunk_rows,
                            prob.layout)
        r = prob.x - V
        w = ops.update_weights(r, nu, p=8)
        nu = ops.update_nu_aecm(w, nu, nulow=opts.robust_nulow,
                                nuhigh=opts.robust_nuhigh, Nd=NU_GRID, p=8)
        prob.weights = w
    # final solve with last weights
    J = _inner_solve(prob, J, opts, inner)
    return J, nu


def _solve_group(state, group, res, cohs, bb, T, Nbase, B, opts,
                 admm_terms=None, itermax=None):
    """Solve a group of clusters as ONE batched LM problem (block-diagonal
    across clusters via the chunk axis), then update the running residual
    incrementally: res += sum(V_old - V_new) over the group."""
    dev = res.device
    # static per-(group, cohs) concatenations cached across EM sweeps:
    # the cluster coherencies, pair table and chunk rows never change
    # within a tile — only xsub does
    cache = getattr(state, '_grp_cache', None)
    if cache is None:
        cache = state._grp_cache = {}
    # token-based key: id() is recycled by the allocator across tiles
    # (ops.obj_token); a fresh token means a new tile -> drop old buffers
    new_tile = getattr(cohs, '_sagecal_token', None) is None
    tok = ops.obj_token(cohs)
    if new_tile or len(cache) > 64:
        cache.clear()
    key = (tuple(group), tok, B)
    ent = cache.get(key)
    if ent is None:
        cs, bbs, rows_all, chunk_counts = [], [], [], []
        for ci in group:
            rows = R.chunk_rows_for(ci, state.nchunks, T, Nbase, B, dev)
            r = rows if rows is not None else torch.zeros(
                B, dtype=torch.long, device=dev)
            cs.append(cohs[ci])
            bbs.append(bb)
            rows_all.append(r + sum(chunk_counts))
            chunk_counts.append(state.nchunks[ci])
        ent = {
            'ccat': torch.cat(cs), 'bbcat': torch.cat(bbs),
            'rcat': torch.cat(rows_all), 'counts': chunk_counts,
            'xbuf': torch.empty(len(group) * B, 2, 2, dtype=res.dtype,
                                device=dev),
        }
        cache[key] = ent
    chunk_counts = ent['counts']
    nch_tot = sum(chunk_counts)
    ccat, bbcat, rcat = ent['ccat'], ent['bbcat'], ent['rcat']
    Vold = []
    xcat = ent['xbuf']
    for gi, ci in enumerate(group):
        Vc, _ = _model_cluster(state, ci, cohs[ci], bb, T, Nbase, B)
        Vold.append(Vc)
        torch.add(res, Vc, out=xcat[gi * B:(gi + 1) * B])
    J0 = torch.cat([state.cluster_J(ci) for ci in group])
    lay = _layout_for(state, bb, T, Nbase, len(group), xcat.device)
    admm = None
    if admm_terms is not None:
        rho_m, Y, BZ = admm_terms
        sel = torch.cat([torch.arange(state.chunk_off[ci],
                                      state.chunk_off[ci]
                                      + state.nchunks[ci])
                         for ci in group]).to(Y.device)
        rho_c = torch.cat([
            torch.full((state.nchunks[ci],), float(rho_m[ci]))
            for ci in group]).to(device=Y.device,
                                 dtype=torch.float32 if Y.dtype == torch.complex64 else torch.float64)
        admm = (rho_c, Y[sel], BZ[sel])
    prob = lm_mod.LMProblem(xcat, ccat, bbcat, state.N, nch_tot, rcat,
                            layout=lay, admm=admm)
    mi = itermax if itermax is not None else opts.max_iter
    if opts.robust:
        nus = float(torch.stack([state.nu[ci] for ci in group]).mean())
        if itermax is not None:
            import copy as _copy
            opts = _copy.copy(opts)
            opts.max_iter = mi
        Jn, nu_new = robust_lm(prob, J0, nus, opts)
        for ci in group:
            state.nu[ci] = nu_new
    else:
        if opts.nsubsets > 1:
            Jn, _ = lm_mod.os_lm_solve(prob, J0, maxiter=mi,
                                       nsubsets=opts.nsubsets)
        else:
            Jn = _inner_solve(prob, J0, opts, mi)
    Jprev = [state.cluster_J(ci).clone() for ci in group]
    off = 0
    for gi, ci in enumerate(group):
        nc = chunk_counts[gi]
        state.set_cluster_J(ci, Jn[off:off + nc])
        off += nc
    # incremental residual update
    res_new = res
    for gi, ci in enumerate(group):
        Vnew, _ = _model_cluster(state, ci, cohs[ci], bb, T, Nbase, B)
        res_new = res_new + Vold[gi] - Vnew
    # group divergence guard: a Jacobi group whose combined update grew the
    # residual is reverted (parallel solves can overshoot when clusters
    # overlap strongly; the reference guards per-tile, fullbatch:622)
    if len(group) > 1 and float((res_new.abs() ** 2).sum()) > \
            float((res.abs() ** 2).sum()):
        for gi, ci in enumerate(group):
            state.set_cluster_J(ci, Jprev[gi])
        # redo the group sequentially
        for ci in group:
            res = _solve_group(state, [ci], res, cohs, bb, T, Nbase, B,
                               opts, admm_terms)
        return res
    return res_new


def sagefit(state, cohs, tile, bb, opts, flags=None, admm_terms=None):
    """The SAGE EM loop (lmfit.c:777-1053). Returns (res_0, res_1): initial
    and final residual scales. NOTE on normalization: we report the
    per-visibility RMS sqrt(sum|r|^2 / n_real); the reference prints
    ||r||_2 / n_real (lmfit.c:869) — a factor sqrt(n_real) smaller. Both
    are monotone in the same quantity; ratios res_1/res_0 agree exactly.

    cohs: [M, B, 2, 2] cluster coherencies (channel-averaged).
    """
    x = tile.x
    T, Nbase = tile.tilesz, tile.Nbase
    B = x.shape[0]
    valid = (~tile.flags) if flags is None else ~flags
    nvalid = max(int(valid.sum()), 1)
    if nvalid < B:
        # zero flagged rows in data and coherencies so they contribute
        # nothing to any solve (preset_flags_and_data,
        # baseline_utils.c semantics)
        vm = valid[:, None, None].to(x.dtype)
        x = x * vm
        cohs = cohs * valid[None, :, None, None].to(cohs.dtype)

    def resnorm(res):
        return float((res[valid].abs() ** 2).sum().sqrt() / (8.0 * nvalid) ** 0.5)

    V = total_model(state, cohs, bb, T, Nbase)
    res = x - V
    res_0 = resnorm(res)

    G = opts.em_group
    groups = [list(range(g, min(g + G, state.M)))
              for g in range(0, state.M, G)]
    nerr = [1.0 / len(groups)] * len(groups)
    weighted = False
    import os as _os
    import time as _time
    trace = _os.environ.get('SAGECAL_TRACE') == '1'
    for em in range(opts.max_emiter):
        t_em = _time.perf_counter() if trace else 0.0
        red = []
        for gi, group in enumerate(groups):
            itermax = None
            if weighted:
                # lmfit.c:880 formula, quantized to a few buckets so the
                # captured-graph cache stays small
                raw = (0.2 * nerr[gi] * len(groups) * opts.max_iter
                       + 0.8 * opts.max_iter)
                buckets = [max(2, int(0.8 * opts.max_iter)),
                           opts.max_iter,
                           int(1.5 * opts.max_iter)]
                itermax = min(buckets, key=lambda b: abs(b - raw))
            if opts.randomize:
                # per-group cost tracking syncs the device; only pay for
                # it when -R weighted allocation is on
                r_before = float((res[valid].abs() ** 2).sum())
            res = _solve_group(state, group, res, cohs, bb, T, Nbase, B,
                               opts, admm_terms, itermax=itermax)
            if opts.randomize:
                red.append(max(r_before
                               - float((res[valid].abs() ** 2).sum()),
                               0.0))
        if opts.randomize:
            tot = sum(red)
            if tot > 0:
                nerr = [r / tot for r in red]
            weighted = not weighted
        # divergence guard (fullbatch_mode.cpp:622-632 resets on blow-up):
        rn = resnorm(res)
        if not (rn == rn) or rn > 5.0 * res_0:
            state.reset()
            V = total_model(state, cohs, bb, T, Nbase)
            res = x - V
        if trace:
            if x.is_cuda:
                torch.cuda.synchronize()
            dt_ms = 1e3 * (_time.perf_counter() - t_em)
            print(f"[trace] EM {em}: {dt_ms:.1f} ms, res {rn:.6f}")

    if opts.joint_iters > 0:
        weights = None
        if opts.robust:
            nu = float(state.nu.mean())
            weights = ops.update_weights(res, nu, p=8)
        Jn, _ = lm_mod.joint_lm_solve(
            x, cohs, state.J, state.chunk_off, state.nchunks, bb, T, Nbase,
            maxiter=opts.joint_iters, weights=weights)
        state.J = Jn
        V = total_model(state, cohs, bb, T, Nbase)
        res = x - V

    if opts.lbfgs_iters > 0:
        from . import lbfgs as lbfgs_mod
        lbfgs_mod.polish(state, cohs, tile, bb, opts)
        V = total_model(state, cohs, bb, T, Nbase)
        res = x - V

    res_1 = resnorm(res)
    return res_0, res_1


def calculate_residuals_multifreq(state, pack, tile, bb, ccid=None, rho=0.0,
                                  device=None, coh_fn=None):
    """Per-channel residuals with the solved gains
    (residual.c calculate_residuals_multifreq:940): re-predict each channel's
    coherencies at its own frequency (spectral-index flux scaling + per
    channel smearing) and subtract J_p C J_q^H.

    Optionally correct residuals by the inverted solution of cluster `ccid`
    (mat_invert(J + rho I), residual.c:125-196). coh_fn(freq) overrides
    the per-channel coherency predict — the beamed-residual hook
    (calculate_residuals_multifreq_withbeam, Dirac_radio.h:495-497).
    Returns xres [F, B, 2, 2]."""
    T, Nbase = tile.tilesz, tile.Nbase
    B = tile.x.shape[0]
    fdelta_ch = tile.fdelta / len(tile.freqs)
    out = torch.empty_like(tile.xo)
    # negative cluster ids: solved during calibration but NOT subtracted
    # from the residual (residual.c:74-75 target-field convention)
    ids = getattr(pack, 'cluster_ids', list(range(state.M)))
    skip = {i for i, c in enumerate(ids) if c < 0}
    for fi, f in enumerate(tile.freqs):
        if coh_fn is not None:
            cohs = coh_fn(float(f)).to(tile.xo.dtype)
        else:
            cohs = ops.predict_coh(pack, tile.u, tile.v, tile.w, float(f),
                                   tile.freq0, fdelta_ch, tile.tdelta,
                                   tile.dec0)
        V = total_model(state, cohs, bb, T, Nbase, skip=skip)
        out[fi] = tile.xo[fi] - V
    if ccid is not None:
        ids = getattr(pack, 'cluster_ids', list(range(state.M)))
        match = [i for i, c in enumerate(ids) if c == ccid]
        if match:
            ci = match[0]
            rows = R.chunk_rows_for(ci, state.nchunks, T, Nbase, B,
                                    tile.x.device)
            if rows is None:
                rows = torch.zeros(B, dtype=torch.long, device=tile.x.device)
            Jc = state.cluster_J(ci)
            Jp = Jc[rows, bb[:, 0]]
            Jq = Jc[rows, bb[:, 1]]
            out = correct_residuals(out, Jp, Jq, rho)
    return out


def _mmse_inv(J, rho):
    """Robust correction inverse, matching the reference's mat_invert
    (residual.c:163-196) exactly: the direct 2x2 inverse of (J + rho I)
    with a determinant guard (det += rho when sqrt(|det|) <= rho) —
    oracle-verified; NOT a true MMSE left-inverse."""
    eye = torch.eye(2, dtype=J.dtype, device=J.device)
    A = J + rho * eye
    det = A[..., 0, 0] * A[..., 1, 1] - A[..., 0, 1] * A[..., 1, 0]
    det = torch.where(det.abs().sqrt() <= rho, det + rho, det)
    inv = torch.empty_like(A)
    inv[..., 0, 0] = A[..., 1, 1]
    inv[..., 0, 1] = -A[..., 0, 1]
    inv[..., 1, 0] = -A[..., 1, 0]
    inv[..., 1, 1] = A[..., 0, 0]
    return inv / det[..., None, None]


def correct_residuals(xres, Jp, Jq, rho=0.0):
    """Apply inverted solutions of one cluster to residuals:
    x <- Jp^+ x (Jq^+)^H per baseline row
    (kernel_correct_residuals predict_model.cu:1825). xres: [F,B,2,2]."""
    Gp = _mmse_inv(Jp, rho)          # [B,2,2]
    Gq = _mmse_inv(Jq, rho)
    return Gp.unsqueeze(0) @ xres @ Gq.conj().transpose(-1, -2).unsqueeze(0)
