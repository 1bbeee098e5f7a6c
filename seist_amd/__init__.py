"""seist_amd — MI355X-native seismic-waveform deep-learning framework.

A from-scratch rebuild of the capabilities of senli1073/SeisT (reference at
/root/reference) designed for AMD Instinct MI355X (gfx950): PyTorch-ROCm
with hand-written CDNA4 HIP kernels for the hot ops and RCCL-over-xGMI data
parallelism. See SURVEY.md for the capability map.
"""

__version__ = "0.1.0"
