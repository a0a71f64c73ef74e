"""sagecal_amd: an MI355X-native radio-interferometric calibration engine.

A from-scratch re-design of nlesc-dirac/sagecal's capabilities for AMD
Instinct MI355X (gfx950): PyTorch-ROCm host plumbing + hand-written CDNA4
HIP kernels for the hot path + RCCL over xGMI for multi-band consensus
ADMM. See SURVEY.md for the reference blueprint.
"""
__version__ = '0.1.0'
