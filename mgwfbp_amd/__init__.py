"""mgwfbp_amd — MI355X-native merged-gradient WFBP training framework.

A from-scratch re-design of the capabilities of HKBU-HPML/MG-WFBP for AMD
Instinct MI355X (gfx950): PyTorch-ROCm compute, hand-written HIP/CDNA4
kernels for the gradient pack/unpack + fused SGD hot path, and RCCL over
xGMI for data-parallel collectives (no Horovod, no MPI, no CUDA).
"""
__version__ = '0.1.0'
