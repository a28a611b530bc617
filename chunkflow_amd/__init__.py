"""chunkflow_amd: MI355X-native chunkflow `inference` hot path.

A from-scratch implementation of chunkflow's convnet inference operator —
overlapping-patch tiler, convnet forward, bump-weighted blend, chunk-mask
normalize, margin crop — behind the reference's own CLI/operator/plugin
surface (reference: seung-lab/chunkflow v1.1.7). Host code is Python +
PyTorch-ROCm; the hot kernels are hand-written HIP for gfx950 behind the
C-ABI in include/chunkflow_amd.h.
"""
__version__ = '0.1.0'

from .cartesian import BoundingBox, BoundingBoxes, Cartesian
from .chunk import Chunk
