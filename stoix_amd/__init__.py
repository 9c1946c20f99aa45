"""stoix_amd — MI355X-native single-agent RL engine.

Brand-new framework with the capabilities of EdanToledo/Stoix,
built for AMD MI355X (gfx950/CDNA4): PyTorch-ROCm host layer, hand-written
HIP kernels for the hot path, RCCL over xGMI for data parallelism.
"""
__version__ = "0.1.0"
