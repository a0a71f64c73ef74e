"""PPM image output for spatial models.

Re-implements /root/reference/src/lib/Dirac/pngoutput.c
convert_tensor_to_image (Dirac.h:1593): render a 2-D amplitude map to a
binary PPM (P6) with a simple thermal colormap.
"""
import numpy as np


def _colormap(t):
    r = np.clip(3 * t - 1.0, 0, 1)
    g = np.clip(3 * t - 0.5, 0, 1) - np.clip(3 * t - 2.0, 0, 1)
    b = np.clip(1.5 * t, 0, 1) - np.clip(3 * t - 1.5, 0, 1)
    return np.stack([r, g, b], axis=-1)


def write_ppm(path, amp):
    """amp: 2-D array; normalized and colormapped to a P6 PPM."""
    a = np.asarray(amp, dtype=np.float64)
    lo, hi = float(a.min()), float(a.max())
    t = (a - lo) / (hi - lo) if hi > lo else np.zeros_like(a)
    rgb = (_colormap(t) * 255).astype(np.uint8)
    with open(path, 'wb') as f:
        f.write(b"P6\n%d %d\n255\n" % (a.shape[1], a.shape[0]))
        f.write(rgb.tobytes())


def plot_spatial_model(path, Z, basis_fn, grid=64, extent=1.0):
    """Render a spatial model's amplitude over an (l, m) grid and save as
    PPM (plot_spatial_model, Dirac_radio.h:432)."""
    l = np.linspace(-extent, extent, grid)
    ll, mm = np.meshgrid(l, l, indexing='ij')
    import torch
    Phi = basis_fn(ll.ravel(), mm.ravel())        # [grid^2, G]
    A = (torch.as_tensor(Phi, dtype=Z.real.dtype) @ Z.T.to(Phi.dtype)
         if not torch.is_complex(torch.as_tensor(Phi)) else Phi @ Z.T)
    amp = np.abs(np.asarray(A)).sum(axis=-1).reshape(grid, grid)
    write_ppm(path, amp)
    return amp
