"""uv-distance data tapers.

Re-implements /root/reference/src/lib/Dirac/updatenu.c whiten_data
(Dirac.h:841) and ncp_weight (updatenu.c:343-349): down-weight short
baselines with 1/(1 + 1.8 exp(-0.05 |uv|_lambda)) (no effect beyond
~400 lambda), used to suppress the north-celestial-pole / large-scale
contamination before calibration.
"""
import torch


def ncp_weight(uvdist_lambda):
    w = 1.0 / (1.0 + 1.8 * torch.exp(-0.05 * uvdist_lambda))
    return torch.where(uvdist_lambda > 400.0, torch.ones_like(w), w)


def whiten_data(x, u, v, freq0):
    """Scale visibilities in place by the uv taper (whiten_data semantics;
    u, v in seconds). Returns the weights used."""
    ud = torch.sqrt((u * freq0) ** 2 + (v * freq0) ** 2)
    w = ncp_weight(ud)
    return x * w[:, None, None].to(x.dtype), w
