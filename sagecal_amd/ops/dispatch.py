"""Op dispatch: CUDA tensors -> the HIP/CDNA4 extension, CPU -> the PyTorch
reference implementation.

The HIP extension is MANDATORY on GPU: if a CUDA tensor reaches an op and
the extension is missing, we raise (no silent eager fallback — the driver's
round-end check verifies the native .so is actually loaded).
Set SAGECAL_FORCE_REFERENCE=1 to explicitly run the torch reference on GPU
(used only for kernel A/B numerics tests).
"""
import os
import torch

from . import reference as R

_ext = None
_ext_err = None


def _load_ext():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        import importlib
        _ext = importlib.import_module('sagecal_amd.ops.hip.dirac_hip')
    except Exception as e:  # noqa: BLE001
        _ext_err = e
        _ext = None
    return _ext


_token_counter = [0]


def obj_token(obj):
    """Stable per-object cache token. id() is NOT a safe cache key — CPython
    recycles addresses after GC, so an id()-keyed cache can serve STALE
    entries (seen in practice: sage._solve_group served a freed tile's
    coherency buffers). The token is stored on the object and dies with it."""
    tok = getattr(obj, '_sagecal_token', None)
    if tok is None:
        _token_counter[0] += 1
        tok = _token_counter[0]
        try:
            obj._sagecal_token = tok
        except AttributeError:   # __slots__/tensor subclass without attr
            return id(obj)
    return tok


def have_ext():
    return _load_ext() is not None


def _use_hip(t):
    if not t.is_cuda:
        return False
    if os.environ.get('SAGECAL_FORCE_REFERENCE') == '1':
        return False
    if _load_ext() is None:
        raise RuntimeError(
            f"sagecal_amd HIP extension not available on GPU tensor "
            f"(build with python setup.py build_ext --inplace): {_ext_err}")
    return True


# ---------------------------------------------------------------------------

def predict_coh(pack, u, v, w, freq, freq0, fdelta, tdelta, dec0, **kw):
    if _use_hip(u):
        from . import hip_host
        return hip_host.predict_coh(pack, u, v, w, freq, freq0, fdelta,
                                    tdelta, dec0, **kw)
    return R.predict_coh(pack, u, v, w, freq, freq0, fdelta, tdelta, dec0,
                         **kw)


def jtj_jtr(x, coh, J, bb, N, weights=None, chunk_rows=None, nchunk=1,
            layout=None):
    if _use_hip(x):
        from . import hip_host
        return hip_host.jtj_jtr(x, coh, J, bb, N, weights, chunk_rows,
                                nchunk, layout)
    return R.jtj_jtr(x, coh, J, bb, N, weights, chunk_rows, nchunk)


def apply_jones(coh, J, bb, chunk_rows=None, layout=None):
    if _use_hip(coh):
        from . import hip_host
        return hip_host.apply_jones(coh, J, bb, chunk_rows, layout)
    return R.apply_jones(coh, J, bb, chunk_rows)


def model_cost_per_chunk(x, coh, J, bb, N, weights=None, chunk_rows=None,
                         nchunk=1, layout=None):
    """Per-chunk weighted cost [nchunk] (kernel_fcost analog)."""
    if _use_hip(x) and layout is not None:
        from . import hip_host
        return hip_host.model_cost_per_chunk(x, coh, J, bb, N, weights,
                                             chunk_rows, nchunk, layout)
    V = R.apply_jones(coh, J, bb, chunk_rows)
    r = x - V
    e2 = (r.abs() ** 2).sum(dim=(-1, -2))
    if weights is not None:
        e2 = e2 * weights
    if chunk_rows is None:
        return e2.sum().unsqueeze(0)
    out = torch.zeros(nchunk, dtype=e2.dtype, device=e2.device)
    out.index_add_(0, chunk_rows, e2)
    return out


def model_and_cost(x, coh, J, bb, weights=None, chunk_rows=None):
    return R.model_and_cost(x, coh, J, bb, weights, chunk_rows)


def update_weights(r, nu, p=8):
    return R.update_weights(r, nu, p)


def update_nu_aecm(w, nu_old, **kw):
    return R.update_nu_aecm(w, nu_old, **kw)


def lbfgs_cost_grad(x, cohs, J_packed, chunk_off, nchunks, bb, T, Nbase,
                    robust_nu=None, weights=None):
    if _use_hip(x):
        from . import hip_host
        return hip_host.lbfgs_cost_grad(x, cohs, J_packed, chunk_off,
                                        nchunks, bb, T, Nbase, robust_nu,
                                        weights)
    return R.lbfgs_cost_grad(x, cohs, J_packed, chunk_off, nchunks, bb, T,
                             Nbase, robust_nu, weights)
