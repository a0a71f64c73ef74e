"""Riemannian Trust-Region (RTR) and Nesterov SD solvers on the product
manifold of per-station 2x2 Jones matrices with a common unitary/linear
ambiguity.

Re-implements /root/reference/src/lib/Dirac/rtr_solve.c (and the robust /
ADMM variants rtr_solve_robust.c, rtr_solve_robust_admm.c):
  - the quotient geometry: X = stacked [2N, 2] complex; vertical space
    {X Om}; projection solves (I (x) X^H X + (X^H X)^T (x) I) om =
    vec(X^H Z - Z^H X) (fns_proj, rtr_solve.c:340-410);
  - identity retraction X + eta (fns_R :419);
  - metric g(eta,gamma) = 2 Re tr(eta^H gamma) (fns_g :323);
  - per-station 1/baseline-count gradient scaling (fns_fscale);
  - truncated-CG trust-region subproblem (tcg_solve :887-1150) with the
    standard radius/rho schedule, RSD warmup;
  - robust Student's-t weighting via the same IRLS wrapping as LM.

MI355X-first: gradients and Gauss-Newton Hessian-vector products come from
the SAME fused JtJ/Jtr kernels as the LM path (grad = -2 Jtr; Hv = 2 JtJ v)
— batched over cluster-chunks, dense [nchunk, 8N, 8N] matvecs on device.
"""
import torch

from ..ops import dispatch as ops
from ..ops.reference import vecR
from . import lm as lm_mod


def _to_X(J):
    """[nchunk, N, 2, 2] -> [nchunk, 2N, 2] stacked (stations x rows)."""
    n, N = J.shape[0], J.shape[1]
    return J.permute(0, 1, 2, 3).reshape(n, 2 * N, 2)


def _herm2_eig(S):
    """Closed-form eigendecomposition of Hermitian 2x2 batch:
    S = U diag(l1,l2) U^H. Returns (l [n,2], U [n,2,2])."""
    a = S[:, 0, 0].real
    d = S[:, 1, 1].real
    b = S[:, 0, 1]
    tr = a + d
    disc = torch.sqrt(((a - d) * 0.5) ** 2 + (b.abs() ** 2)).clamp_min(0)
    l1 = 0.5 * tr + disc
    l2 = 0.5 * tr - disc
    # eigenvector for l1: (b, l1 - a) unless b ~ 0
    bz = b.abs() < 1e-30
    v1a = torch.where(bz, torch.ones_like(b), b)
    v1b = torch.where(bz, torch.zeros_like(b), (l1 - a).to(b.dtype))
    nrm = torch.sqrt(v1a.abs() ** 2 + v1b.abs() ** 2).clamp_min(1e-30)
    v1a, v1b = v1a / nrm.to(b.dtype), v1b / nrm.to(b.dtype)
    # orthogonal second eigenvector
    v2a = -v1b.conj()
    v2b = v1a.conj()
    U = torch.stack([torch.stack([v1a, v2a], -1),
                     torch.stack([v1b, v2b], -1)], -2)
    lam = torch.stack([l1, l2], -1)
    return lam, U


def _proj(J, Z):
    """Project ambient direction Z onto the horizontal space at J
    (fns_proj): Z - J Om where Om solves the Sylvester equation
    (X^H X) Om + Om (X^H X) = X^H Z - Z^H X.

    Solved in CLOSED FORM via the 2x2 Hermitian eigendecomposition of
    S = X^H X (no batched LAPACK: S = U L U^H => Om' = R'/(l_i + l_j) in
    the eigenbasis) — the reference's 4x4 zgels (rtr_solve.c:340-410)
    is equivalent but launch/LAPACK-bound on GPU. J, Z: [n, N, 2, 2]."""
    X = _to_X(J)
    Zt = _to_X(Z)
    S = X.conj().transpose(-1, -2) @ X           # [n,2,2] Hermitian
    XZ = X.conj().transpose(-1, -2) @ Zt
    Rr = XZ - XZ.conj().transpose(-1, -2)        # skew-Hermitian RHS
    lam, U = _herm2_eig(S)
    Rp = U.conj().transpose(-1, -2) @ Rr @ U
    denom = (lam[:, :, None] + lam[:, None, :]).clamp_min(1e-30)
    Omp = Rp / denom.to(Rp.dtype)
    Om = U @ Omp @ U.conj().transpose(-1, -2)
    return Z - torch.einsum('nsij,njk->nsik', J, Om)


def _inner(a, b):
    """g(eta,gamma) = 2 Re tr(eta^H gamma), batched over chunks."""
    return 2.0 * (a.conj() * b).real.sum(dim=(-1, -2, -3))


class _Objective:
    """Cost/grad/Hessian-vector oracle built on the fused kernels."""

    def __init__(self, prob, iw):
        self.prob = prob
        self.iw = iw                      # [N] per-station 1/count scale

    def refresh(self, J):
        p = self.prob
        JtJ, Jtr, _ = ops.jtj_jtr(p.x, p.coh, J, p.bb, p.N, p.weights,
                                  p.chunk_rows, p.nchunk, p.layout)
        if p.admm is not None:
            JtJ, Jtr = lm_mod._apply_admm_terms(JtJ, Jtr, J, p, p.N,
                                                p.nchunk)
        self.JtJ = JtJ
        self.Jtr = Jtr

    def cost(self, J):
        return lm_mod._total_cost(self.prob.x, self.prob.coh, J,
                                  self.prob.bb, self.prob)

    def grad(self, J):
        """Riemannian gradient: -2 Jtr scaled per station, projected."""
        n, N = J.shape[0], J.shape[1]
        g = -2.0 * self.Jtr.reshape(n, N, 8)
        g = g * self.iw.reshape(1, N, 1).to(g.dtype)
        gc = torch.view_as_complex(
            g.reshape(n, N, 2, 2, 2).contiguous())
        return _proj(J, gc)

    def hess_vec(self, J, eta):
        """Gauss-Newton Hessian-vector: 2 JtJ eta (scaled, projected)."""
        n, N = J.shape[0], J.shape[1]
        v = vecR(eta).reshape(n, 8 * N)
        Hv = 2.0 * torch.bmm(self.JtJ, v.unsqueeze(-1)).squeeze(-1)
        Hv = (Hv.reshape(n, N, 8)
              * self.iw.reshape(1, N, 1).to(Hv.dtype))
        Hc = torch.view_as_complex(Hv.reshape(n, N, 2, 2, 2).contiguous())
        return _proj(J, Hc)


def _tcg(obj, J, grad, delta, maxit=30, kappa=0.01, theta=1.0):
    """Truncated CG (tcg_solve, rtr_solve.c:887-1150), batched."""
    n = J.shape[0]
    eta = torch.zeros_like(J)
    r = grad.clone()
    d = -r
    r0 = _inner(r, r)
    rnorm0 = r0.sqrt()
    done = torch.zeros(n, dtype=torch.bool, device=J.device)
    tol = rnorm0 * torch.minimum(rnorm0 ** theta,
                                 torch.full_like(rnorm0, kappa))
    for it in range(maxit):
        Hd = obj.hess_vec(J, d)
        dHd = _inner(d, Hd)
        alpha = _inner(r, r) / dHd.clamp_min(1e-30)
        eta_new = eta + alpha[:, None, None, None] * d
        enorm = _inner(eta_new, eta_new).sqrt()
        neg = dHd <= 0
        big = enorm > delta
        hit = (neg | big) & ~done
        # boundary step along d for chunks that hit
        ee = _inner(eta, eta)
        ed = _inner(eta, d)
        dd = _inner(d, d)
        tau = (-ed + (ed ** 2 + dd * (delta ** 2 - ee)).clamp_min(0)
               .sqrt()) / dd.clamp_min(1e-30)
        bnd = eta + tau[:, None, None, None] * d
        eta_new = torch.where(hit[:, None, None, None], bnd, eta_new)
        eta = torch.where(done[:, None, None, None], eta, eta_new)
        done = done | hit
        r = r + alpha[:, None, None, None] * Hd
        rn = _inner(r, r)
        done = done | (rn.sqrt() <= tol)
        beta = rn / r0.clamp_min(1e-30)
        d = -r + beta[:, None, None, None] * d
        r0 = rn
        # poll every 8 iters only (host-sync hygiene)
        if (it & 7) == 7 and bool(done.all()):
            break
    return eta


def rtr_solve(prob, J0, maxiter=20, rsd_iters=2, delta0=None,
              verbose=False):
    """Batched RTR (rtr_solve_nocuda, rtr_solve.c:1208). Returns (J, info).

    prob: lm.LMProblem (weights -> robust RTR; admm -> RTR-ADMM)."""
    import os as _os
    # graph-captured path is the CUDA default (GPU-validated round 2:
    # test_rtr_graphed_matches_eager); opt out with SAGECAL_RTR_GRAPH=0
    if (_os.environ.get('SAGECAL_RTR_GRAPH', '1') != '0'
            and _os.environ.get('SAGECAL_NO_GRAPH') != '1'
            and J0.is_cuda and prob.layout is not None):
        return rtr_solve_graphed(prob, J0, maxiter=maxiter,
                                 rsd_iters=rsd_iters)
    N = prob.N
    n = prob.nchunk
    dev = prob.x.device
    # per-station inverse baseline counts (fns_fcount, Dirac.h:1119)
    counts = torch.zeros(N, dtype=torch.float64)
    bbc = prob.bb.cpu()
    import numpy as np
    cnt = np.bincount(bbc.reshape(-1).numpy(), minlength=N)
    counts = torch.tensor(np.maximum(cnt, 1), dtype=torch.float64)
    iw = (1.0 / counts).to(device=dev, dtype=prob.x.real.dtype)

    J = J0.clone()
    obj = _Objective(prob, iw)
    obj.refresh(J)
    cost = obj.cost(J)
    init_cost = cost.clone()
    # RSD warmup (rtr_solve.c:1348): a few projected steepest-descent steps
    for _ in range(rsd_iters):
        g = obj.grad(J)
        gn = _inner(g, g)
        Hg = obj.hess_vec(J, g)
        step = (gn / _inner(g, Hg).clamp_min(1e-30))
        Jn = J - step[:, None, None, None].to(J.real.dtype) * g
        cn = obj.cost(Jn)
        better = cn < cost
        J = torch.where(better[:, None, None, None], Jn, J)
        cost = torch.where(better, cn, cost)
        obj.refresh(J)
    g = obj.grad(J)
    gnorm = _inner(g, g).sqrt()
    if delta0 is None:
        delta = 0.1 * _inner(J, J).sqrt().clamp_min(1.0)
    else:
        delta = torch.full((n,), delta0, dtype=gnorm.dtype, device=dev)
    delta_bar = delta * 8
    for it in range(maxiter):
        eta = _tcg(obj, J, g, delta)
        Jn = J + eta          # identity retraction (fns_R)
        cn = obj.cost(Jn)
        # model decrease m(0) - m(eta)
        Heta = obj.hess_vec(J, eta)
        mdec = -(_inner(g, eta) + 0.5 * _inner(eta, Heta))
        rho = (cost - cn) / mdec.clamp_min(1e-30)
        accept = (rho > 0.1) & (cn < cost)
        J = torch.where(accept[:, None, None, None], Jn, J)
        cost = torch.where(accept, cn, cost)
        delta = torch.where(rho < 0.25, delta * 0.25,
                            torch.where(rho > 0.75,
                                        torch.minimum(delta * 2.0,
                                                      delta_bar), delta))
        obj.refresh(J)
        g = obj.grad(J)
        if (it & 3) == 3 and float(_inner(g, g).sqrt().max()) < 1e-9:
            break
    return J, {'init_cost': init_cost, 'final_cost': cost,
               'niter': maxiter}


def nsd_solve(prob, J0, maxiter=40, lr=None):
    """Nesterov accelerated projected SD (nsd_solve_nocuda_robust,
    rtr_solve_robust.c API Dirac.h:1161-1175), batched."""
    N = prob.N
    import numpy as np
    bbc = prob.bb.cpu()
    cnt = np.bincount(bbc.reshape(-1).numpy(), minlength=N)
    iw = torch.tensor(1.0 / np.maximum(cnt, 1),
                      device=prob.x.device).to(prob.x.real.dtype)
    obj = _Objective(prob, iw)
    J = J0.clone()
    Yk = J.clone()
    t = 1.0
    obj.refresh(J)
    cost = obj.cost(J)
    init_cost = cost.clone()
    for it in range(maxiter):
        obj.refresh(Yk)
        g = obj.grad(Yk)
        Hg = obj.hess_vec(Yk, g)
        step = (_inner(g, g) / _inner(g, Hg).clamp_min(1e-30))
        Jn = Yk - step[:, None, None, None].to(J.real.dtype) * g
        t_new = 0.5 * (1 + (1 + 4 * t * t) ** 0.5)
        Yk = Jn + ((t - 1) / t_new) * (Jn - J)
        J = Jn
        t = t_new
    obj.refresh(J)
    cost = obj.cost(J)
    return J, {'init_cost': init_cost, 'final_cost': cost,
               'niter': maxiter}


# ---------------------------------------------------------------------------
# hipGraph capture of the RTR loop (rtr_solve body with the host polls
# removed): fixed RSD warmup + fixed outer iterations + fixed-length tCG.
# Default ON for CUDA since round 2 (GPU-validated:
# tests/test_gpu.py::test_rtr_graphed_matches_eager); disable with
# SAGECAL_RTR_GRAPH=0. The poll-free body is also validated on CPU in
# tests/test_rtr.py::test_rtr_body_matches_eager.
# ---------------------------------------------------------------------------

def _tcg_fixed(obj, J, grad, delta, maxit):
    """_tcg without the host-side early-exit poll (capture-safe)."""
    n = J.shape[0]
    eta = torch.zeros_like(J)
    r = grad.clone()
    d = -r
    r0 = _inner(r, r)
    rnorm0 = r0.sqrt()
    done = torch.zeros(n, dtype=torch.bool, device=J.device)
    tol = rnorm0 * torch.minimum(rnorm0, torch.full_like(rnorm0, 0.01))
    for it in range(maxit):
        Hd = obj.hess_vec(J, d)
        dHd = _inner(d, Hd)
        alpha = _inner(r, r) / dHd.clamp_min(1e-30)
        eta_new = eta + alpha[:, None, None, None] * d
        enorm = _inner(eta_new, eta_new).sqrt()
        hit = ((dHd <= 0) | (enorm > delta)) & ~done
        ee = _inner(eta, eta)
        ed = _inner(eta, d)
        dd = _inner(d, d)
        tau = (-ed + (ed ** 2 + dd * (delta ** 2 - ee)).clamp_min(0)
               .sqrt()) / dd.clamp_min(1e-30)
        bnd = eta + tau[:, None, None, None] * d
        eta_new = torch.where(hit[:, None, None, None], bnd, eta_new)
        eta = torch.where(done[:, None, None, None], eta, eta_new)
        done = done | hit
        r = r + alpha[:, None, None, None] * Hd
        rn = _inner(r, r)
        done = done | (rn.sqrt() <= tol)
        beta = rn / r0.clamp_min(1e-30)
        d = -r + beta[:, None, None, None] * d
        r0 = rn
    return eta


def _rtr_body(x, coh, bb, N, nchunk, chunk_rows, w, layout, J0, iw,
              maxiter, rsd_iters, tcg_maxit, prob_admm):
    """Capture-safe RTR: no host reads, fixed iteration counts. Converged
    chunks keep iterating (masked no-ops), like the graphed LM body."""

    class _P:
        pass
    p = _P()
    p.x, p.coh, p.bb, p.N = x, coh, bb, N
    p.nchunk, p.chunk_rows, p.layout = nchunk, chunk_rows, layout
    p.weights = w
    p.admm = prob_admm
    J = J0.clone()
    obj = _Objective(p, iw)
    obj.refresh(J)
    cost = obj.cost(J)
    init_cost = cost.clone()
    for _ in range(rsd_iters):
        g = obj.grad(J)
        gn = _inner(g, g)
        Hg = obj.hess_vec(J, g)
        step = (gn / _inner(g, Hg).clamp_min(1e-30))
        Jn = J - step[:, None, None, None].to(J.real.dtype) * g
        cn = obj.cost(Jn)
        better = cn < cost
        J = torch.where(better[:, None, None, None], Jn, J)
        cost = torch.where(better, cn, cost)
        obj.refresh(J)
    g = obj.grad(J)
    delta = 0.1 * _inner(J, J).sqrt().clamp_min(1.0)
    delta_bar = delta * 8
    for _ in range(maxiter):
        eta = _tcg_fixed(obj, J, g, delta, tcg_maxit)
        Jn = J + eta
        cn = obj.cost(Jn)
        Heta = obj.hess_vec(J, eta)
        mdec = -(_inner(g, eta) + 0.5 * _inner(eta, Heta))
        rho = (cost - cn) / mdec.clamp_min(1e-30)
        accept = (rho > 0.1) & (cn < cost)
        J = torch.where(accept[:, None, None, None], Jn, J)
        cost = torch.where(accept, cn, cost)
        delta = torch.where(rho < 0.25, delta * 0.25,
                            torch.where(rho > 0.75,
                                        torch.minimum(delta * 2.0,
                                                      delta_bar), delta))
        obj.refresh(J)
        g = obj.grad(J)
    return J, cost, init_cost


_rtr_graph_cache = {}


def _station_iw(bb, N, device, rdtype):
    import numpy as np
    cnt = np.bincount(bb.cpu().reshape(-1).numpy(), minlength=N)
    return torch.tensor(1.0 / np.maximum(cnt, 1)).to(device=device,
                                                     dtype=rdtype)


def rtr_solve_graphed(prob, J0, maxiter=20, rsd_iters=2, tcg_maxit=12):
    """Graph-captured RTR (one replay per solve). Same caching scheme as
    lm._lm_solve_graphed: static input buffers keyed by shape+layout."""
    from ..ops import dispatch as ops_d
    x, coh, bb, N = prob.x, prob.coh, prob.bb, prob.N
    nchunk = prob.nchunk
    dev = x.device
    B = x.shape[0]
    key = (B, nchunk, N, maxiter, rsd_iters, tcg_maxit,
           ops_d.obj_token(prob.layout), prob.admm is not None)
    ent = _rtr_graph_cache.get(key)
    if ent is None:
        ent = {'x': torch.empty_like(x), 'coh': torch.empty_like(coh),
               'J0': torch.empty_like(J0),
               'w': torch.ones(B, dtype=torch.float32, device=dev),
               'rows': torch.zeros(B, dtype=torch.long, device=dev),
               'bb': bb,
               'iw': _station_iw(bb, N, dev, x.real.dtype)}
        admm_s = None
        if prob.admm is not None:
            rho_a, Y_a, BZ_a = prob.admm
            ent['admm'] = (torch.empty_like(rho_a), torch.empty_like(Y_a),
                           torch.empty_like(BZ_a))
            admm_s = ent['admm']
        ent['admm_s'] = admm_s
        ent['x'].copy_(x)
        ent['coh'].copy_(coh)
        ent['J0'].copy_(J0)
        if prob.weights is not None:
            ent['w'].copy_(prob.weights.to(torch.float32))
        if prob.chunk_rows is not None:
            ent['rows'].copy_(prob.chunk_rows)
        if admm_s is not None:
            for d_, s_ in zip(admm_s, prob.admm):
                d_.copy_(s_)
        st = torch.cuda.Stream()
        st.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(st):
            for _ in range(2):
                _rtr_body(ent['x'], ent['coh'], ent['bb'], N, nchunk,
                          ent['rows'], ent['w'], prob.layout, ent['J0'],
                          ent['iw'], maxiter, rsd_iters, tcg_maxit, admm_s)
        torch.cuda.current_stream().wait_stream(st)
        gobj = torch.cuda.CUDAGraph()
        with torch.cuda.graph(gobj):
            outs = _rtr_body(ent['x'], ent['coh'], ent['bb'], N, nchunk,
                             ent['rows'], ent['w'], prob.layout, ent['J0'],
                             ent['iw'], maxiter, rsd_iters, tcg_maxit,
                             admm_s)
        ent['graph'] = gobj
        ent['outs'] = outs
        _rtr_graph_cache[key] = ent
    ent['x'].copy_(x)
    ent['coh'].copy_(coh)
    ent['J0'].copy_(J0)
    if prob.weights is not None:
        ent['w'].copy_(prob.weights.to(torch.float32))
    else:
        ent['w'].fill_(1.0)
    if prob.chunk_rows is not None:
        ent['rows'].copy_(prob.chunk_rows)
    else:
        ent['rows'].zero_()
    if ent['admm_s'] is not None:
        for d_, s_ in zip(ent['admm_s'], prob.admm):
            d_.copy_(s_)
    ent['graph'].replay()
    J, cost, init_cost = ent['outs']
    return J.clone(), {'init_cost': init_cost.clone(),
                       'final_cost': cost.clone(), 'niter': maxiter}
