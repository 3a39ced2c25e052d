"""Low-level op surface.

The compute path is two HIP kernels (csrc/hip/engine_gpu.hip):
  k_move    - fused relocation + segment walk + track-length tally
  k_locate  - grid-based point-in-mesh localization
exposed at the Python level through TallyEngine.  This module re-exports
the engine for symmetry with the package layout.
"""
from .. import TallyEngine

__all__ = ["TallyEngine"]
