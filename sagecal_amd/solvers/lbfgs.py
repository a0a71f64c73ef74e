"""LBFGS: full-batch and minibatch (stochastic, persistent-memory).

Re-implements /root/reference/src/lib/Dirac/lbfgs.c: two-loop recursion
(mult_hessian, lbfgs.c:33), line searches (Fletcher w/ cubic interpolation
lbfgs.c:115-441; backtracking for stochastic mode lbfgs.c:443), and the
cyclic y/s curvature memory persisted across minibatches
(persistent_data_t, Dirac.h:84-110). Works on a generic cost/grad closure,
so it is radio-agnostic like the reference library (test/Dirac/demo.c
minimizes Rosenbrock with it — mirrored in tests/test_lbfgs.py).
"""
import torch


class LBFGSMemory:
    """Cyclic s/y curvature store (persistent_data_t analog)."""

    def __init__(self, m, n, dtype=torch.float64, device='cpu'):
        self.m = m
        self.S = torch.zeros(m, n, dtype=dtype, device=device)
        self.Y = torch.zeros(m, n, dtype=dtype, device=device)
        self.rho = torch.zeros(m, dtype=dtype, device=device)
        self.count = 0          # total updates pushed
        self.running_avg = None  # gradient-variance tracking (online mode)
        self.running_avg_sq = None
        self.niter = 0

    def reset(self):
        self.S.zero_(); self.Y.zero_(); self.rho.zero_()
        self.count = 0
        self.running_avg = None
        self.running_avg_sq = None

    def push(self, s, y):
        sy = torch.dot(s, y)
        if float(sy) <= 1e-300:
            return  # curvature condition failed; skip update
        idx = self.count % self.m
        self.S[idx] = s
        self.Y[idx] = y
        self.rho[idx] = 1.0 / sy
        self.count += 1

    def two_loop(self, g):
        """H0-scaled two-loop recursion (mult_hessian lbfgs.c:33)."""
        k = min(self.count, self.m)
        if k == 0:
            return g.clone()
        order = [(self.count - 1 - i) % self.m for i in range(k)]
        q = g.clone()
        alphas = []
        for idx in order:
            a = self.rho[idx] * torch.dot(self.S[idx], q)
            alphas.append(a)
            q = q - a * self.Y[idx]
        last = order[0]
        gamma = torch.dot(self.S[last], self.Y[last]) / \
            torch.dot(self.Y[last], self.Y[last]).clamp_min(1e-300)
        q = q * gamma
        for idx, a in zip(reversed(order), reversed(alphas)):
            b = self.rho[idx] * torch.dot(self.Y[idx], q)
            q = q + (a - b) * self.S[idx]
        return q


def _cubic_linesearch(fg, p, f0, g0, d, alpha0=1.0, c1=1e-4, c2=0.9,
                      max_ls=20):
    """Strong-Wolfe line search with cubic interpolation (the role of
    lbfgs.c:115-441's Fletcher search). fg(p) -> (f, g)."""
    dg0 = torch.dot(g0, d)
    if float(dg0) >= 0:
        return None, None, None, 0.0
    alpha = alpha0
    alpha_lo, alpha_hi = 0.0, float('inf')
    f_lo = f0
    best = None
    for _ in range(max_ls):
        f1, g1 = fg(p + alpha * d)
        if float(f1) > float(f0 + c1 * alpha * dg0) or \
           (best is not None and float(f1) >= float(f_lo)):
            alpha_hi = alpha
        else:
            dg1 = torch.dot(g1, d)
            best = (f1, g1, alpha)
            if abs(float(dg1)) <= -c2 * float(dg0):
                return f1, g1, p + alpha * d, alpha
            if float(dg1) >= 0:
                alpha_hi = alpha
            else:
                alpha_lo, f_lo = alpha, f1
        if alpha_hi < float('inf'):
            alpha = 0.5 * (alpha_lo + alpha_hi)
        else:
            alpha *= 2.0
    if best is not None:
        f1, g1, a = best
        return f1, g1, p + a * d, a
    return None, None, None, 0.0


def _backtracking(fg_cost, p, f0, g0, d, alpha0=1.0, c1=1e-4, beta=0.5,
                  max_ls=25):
    """Armijo backtracking (stochastic mode, lbfgs.c:443)."""
    dg0 = torch.dot(g0, d)
    alpha = alpha0
    for _ in range(max_ls):
        f1 = fg_cost(p + alpha * d)
        if float(f1) <= float(f0 + c1 * alpha * dg0):
            return alpha
        alpha *= beta
    return 0.0


def lbfgs_fit(fg, p0, maxiter=50, m=7, gtol=1e-9, mem=None,
              stochastic=False, alpha0=1.0):
    """Minimize via LBFGS. fg(p) -> (cost, grad). Returns (p, mem, info).

    With `mem` given, curvature is warm-started and persisted (minibatch
    mode: lbfgs_fit_minibatch lbfgs.c:717). stochastic=True switches to
    Armijo backtracking with a decaying initial step."""
    p = p0.clone()
    if mem is None:
        mem = LBFGSMemory(m, p.numel(), dtype=p.dtype, device=p.device)
    f, g = fg(p)
    info = {'f0': float(f), 'niter': 0}
    for it in range(maxiter):
        if float(g.abs().max()) < gtol:
            break
        d = -mem.two_loop(g)
        if stochastic:
            a0 = alpha0 / (1.0 + 0.1 * mem.niter)
            alpha = _backtracking(lambda q: fg(q)[0], p, f, g, d, alpha0=a0)
            if alpha == 0.0:
                break
            pn = p + alpha * d
            fn, gn = fg(pn)
        else:
            fn, gn, pn, alpha = _cubic_linesearch(fg, p, f, g, d,
                                                  alpha0=1.0 if mem.count else min(1.0, 1.0 / max(float(g.norm()), 1e-12)))
            if pn is None:
                break
        mem.push(pn - p, gn - g)
        mem.niter += 1
        p, f, g = pn, fn, gn
        info['niter'] = it + 1
    info['f1'] = float(f)
    return p, mem, info


# ---------------------------------------------------------------------------
# Calibration wrappers (robust_lbfgs.c lbfgs_fit_wrapper:738 analogs)
# ---------------------------------------------------------------------------

def _pack_params(J):
    return torch.view_as_real(J).reshape(-1).clone()


def _unpack_params(v, Mt, N, cdtype):
    return torch.view_as_complex(v.reshape(Mt, N, 2, 2, 2).contiguous())


def polish(state, cohs, tile, bb, opts):
    """Final joint LBFGS over the full 8NMt parameter vector
    (lmfit.c:1019-1037), Gaussian or Student's-t cost per solver mode."""
    from ..ops import dispatch as ops
    Mt, N = state.Mt, state.N
    T, Nbase = tile.tilesz, tile.Nbase
    chunk_off = state.chunk_off
    nchunks = state.nchunks
    nu = float(state.nu.mean()) if opts.robust else None

    def fg(v):
        J = _unpack_params(v, Mt, N, state.J.dtype)
        c, g = ops.lbfgs_cost_grad(tile.x, cohs, J, chunk_off, nchunks, bb,
                                   T, Nbase, robust_nu=nu)
        return c, g

    v0 = _pack_params(state.J)
    v1, _, _ = lbfgs_fit(fg, v0, maxiter=opts.lbfgs_iters,
                         m=getattr(opts, 'lbfgs_m', 7))
    state.J = _unpack_params(v1, Mt, N, state.J.dtype)


def lbfgs_fit_bounded(fg, p0, lbound, ubound, maxiter=50, m=7, gtol=1e-9,
                      mem=None):
    """Bound-constrained LBFGS (the role of lbfgsb.c, Dirac.h:1797-1846),
    via gradient projection: iterates are projected onto [lbound, ubound],
    the search direction is the two-loop direction with bound-active
    coordinates zeroed, and the line search walks the projected path."""
    lb = torch.as_tensor(lbound, dtype=p0.dtype, device=p0.device)
    ub = torch.as_tensor(ubound, dtype=p0.dtype, device=p0.device)

    def proj(p):
        return torch.minimum(torch.maximum(p, lb), ub)

    p = proj(p0.clone())
    if mem is None:
        mem = LBFGSMemory(m, p.numel(), dtype=p.dtype, device=p.device)
    f, g = fg(p)
    for it in range(maxiter):
        active = ((p <= lb) & (g > 0)) | ((p >= ub) & (g < 0))
        gf = torch.where(active, torch.zeros_like(g), g)
        if float(gf.abs().max()) < gtol:
            break
        d = -mem.two_loop(gf)
        d = torch.where(active, torch.zeros_like(d), d)
        alpha = _backtracking(lambda q: fg(proj(q))[0], p, f, g, d)
        if alpha == 0.0:
            break
        pn = proj(p + alpha * d)
        fn, gn = fg(pn)
        mem.push(pn - p, gn - g)
        p, f, g = pn, fn, gn
    return p, mem, {'f1': float(f)}


def lbfgsb_fit(fg, p0, lbound, ubound, maxiter=100, m=7, gtol=1e-9,
               ftol=1e-12):
    """Full Byrd–Lu–Nocedal–Zhu L-BFGS-B (the reference's lbfgsb.c):
    generalized Cauchy point along the projected-gradient path + direct
    primal subspace minimization over the free variables using the
    compact representation B = theta*I - W M W^T, then a projected Armijo
    line search. `lbfgs_fit_bounded` remains as the cheap
    projected-gradient variant for well-conditioned problems.

    fg(p) -> (f, grad); returns (p, info) with info['f1'], info['iters'].
    Validated against scipy.optimize L-BFGS-B in tests/test_lbfgsb.py.
    """
    dt, dev = p0.dtype, p0.device
    lb = torch.as_tensor(lbound, dtype=dt, device=dev).expand_as(p0).clone()
    ub = torch.as_tensor(ubound, dtype=dt, device=dev).expand_as(p0).clone()
    n = p0.numel()

    def proj(x):
        return torch.minimum(torch.maximum(x, lb), ub)

    x = proj(p0.clone())
    f, g = fg(x)
    Slist, Ylist = [], []
    theta = 1.0
    eps = torch.finfo(dt).eps

    def compact():
        """W [n,2k], M [2k,2k] with B = theta I - W M W^T."""
        if not Slist:
            return None, None
        S = torch.stack(Slist, dim=1)          # [n, k]
        Y = torch.stack(Ylist, dim=1)
        SY = S.T @ Y                           # [k, k]
        D = torch.diag(torch.diagonal(SY))
        L = torch.tril(SY, diagonal=-1)
        StS = S.T @ S
        k = S.shape[1]
        Mi = torch.zeros(2 * k, 2 * k, dtype=dt, device=dev)
        Mi[:k, :k] = -D
        Mi[:k, k:] = L.T
        Mi[k:, :k] = L
        Mi[k:, k:] = theta * StS
        W = torch.cat([Y, theta * S], dim=1)   # [n, 2k]
        return W, torch.linalg.inv(Mi)

    info = {'f1': float(f), 'iters': 0}
    for it in range(maxiter):
        pg = x - proj(x - g)                   # projected gradient
        if float(pg.abs().max()) < gtol:
            break
        W, M = compact()

        # ---- generalized Cauchy point (Alg. CP) ----
        t_brk = torch.full_like(x, float('inf'))
        neg, pos = g < 0, g > 0
        t_brk[neg] = (x[neg] - ub[neg]) / g[neg]
        t_brk[pos] = (x[pos] - lb[pos]) / g[pos]
        d = torch.where(t_brk > 0, -g, torch.zeros_like(g))
        xcp = x.clone()
        if W is not None:
            p = W.T @ d
            c = torch.zeros_like(p)
        fp = -float(d @ d)
        if W is not None:
            fpp = -theta * fp - float(p @ (M @ p))
        else:
            fpp = -theta * fp
        fpp = max(fpp, eps)
        dt_min = -fp / fpp
        t_old = 0.0
        order = torch.argsort(t_brk)
        for bi in order.tolist():
            t_b = float(t_brk[bi])
            if t_b <= 0 or not np_isfinite(t_b):
                if t_b <= 0:
                    continue
                break
            delta = t_b - t_old
            if dt_min < delta:
                break
            # hit breakpoint bi: fix at its bound
            t_old = t_b
            xcp[bi] = ub[bi] if g[bi] < 0 else lb[bi]
            z_b = float(xcp[bi] - x[bi])
            g_b = float(g[bi])
            if W is not None:
                c = c + delta * p
                w_b = W[bi]
                wMc = float(w_b @ (M @ c))
                wMp = float(w_b @ (M @ p))
                wMw = float(w_b @ (M @ w_b))
            else:
                wMc = wMp = wMw = 0.0
            fp += delta * fpp + g_b * g_b + theta * g_b * z_b - g_b * wMc
            fpp += -theta * g_b * g_b - 2.0 * g_b * wMp - g_b * g_b * wMw
            fpp = max(fpp, eps)
            if W is not None:
                p = p + g_b * w_b
            d[bi] = 0.0
            dt_min = -fp / fpp
            if fp >= 0:
                dt_min = 0.0
                break
        dt_min = max(dt_min, 0.0)
        t_final = t_old + dt_min
        move = t_brk > t_old                   # still-moving coordinates
        xcp[move] = proj(x + t_final * d)[move]
        if W is not None:
            c = c + dt_min * p

        # ---- subspace minimization over free variables at xcp ----
        free = (xcp > lb) & (xcp < ub)
        xn_target = xcp
        if bool(free.any()):
            r = (g + theta * (xcp - x))
            if W is not None:
                r = r - W @ (M @ c)
            rf = r[free]
            if W is not None:
                Wf = W[free]
                k2 = W.shape[1]
                K = torch.eye(k2, dtype=dt, device=dev) \
                    - (M @ (Wf.T @ Wf)) / theta
                v = torch.linalg.solve(K, M @ (Wf.T @ rf))
                du = -(rf / theta + (Wf @ v) / theta ** 2)
            else:
                du = -rf / theta
            # truncate to the box
            xf = xcp[free]
            lbf, ubf = lb[free], ub[free]
            with torch.no_grad():
                pos_d = du > 0
                neg_d = du < 0
                amax = torch.ones_like(du)
                amax[pos_d] = (ubf[pos_d] - xf[pos_d]) / du[pos_d]
                amax[neg_d] = (lbf[neg_d] - xf[neg_d]) / du[neg_d]
                alpha_max = float(amax.clamp(min=0.0).min()) \
                    if du.numel() else 1.0
            xn_target = xcp.clone()
            xn_target[free] = xf + min(1.0, alpha_max) * du

        # ---- projected Armijo backtracking from x toward xn_target ----
        dstep = xn_target - x
        gTd = float(g @ dstep)
        if gTd > 0:                            # not a descent dir: fall back
            dstep = -pg
            gTd = float(g @ dstep)
        alpha, fn, xn, gn = 1.0, None, None, None
        for _ in range(30):
            xt = proj(x + alpha * dstep)
            ft, gt = fg(xt)
            if float(ft) <= float(f) + 1e-4 * alpha * gTd or \
                    float(ft) < float(f) - abs(float(f)) * 1e-16:
                fn, xn, gn = ft, xt, gt
                break
            alpha *= 0.5
        if xn is None:
            break
        s, y = xn - x, gn - g
        sy = float(s @ y)
        if sy > 1e-12 * float(s.norm()) * float(y.norm()) + 1e-300:
            Slist.append(s)
            Ylist.append(y)
            if len(Slist) > m:
                Slist.pop(0)
                Ylist.pop(0)
            theta = float(y @ y) / sy
        fprev = float(f)
        x, f, g = xn, fn, gn
        info['iters'] = it + 1
        if abs(fprev - float(f)) <= ftol * max(1.0, abs(float(f))):
            break
    info['f1'] = float(f)
    return x, info


def np_isfinite(v):
    return v == v and v not in (float('inf'), float('-inf'))
