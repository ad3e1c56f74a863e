"""perceiver_amd — MI355X-native Perceiver / Perceiver IO / Perceiver AR framework.

A from-scratch reimplementation of the capabilities of krasserm/perceiver-io for AMD
Instinct MI355X (gfx950): PyTorch-ROCm model layer, hand-written CDNA4 HIP kernels for
the attention/fusion hot paths, RCCL over xGMI for data parallelism, and a native
training loop replacing PyTorch Lightning.
"""
__version__ = "0.1.0"
