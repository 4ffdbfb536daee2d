"""bigstitcher_spark_amd — MI355X-native implementation of the
BigStitcher-Spark pairwise phase-correlation stitching and block-wise
affine-fusion hot path (BASELINE.json north_star).

Layers:
  libbigstitch.so (csrc/)  — hand-written HIP/CDNA4 kernels + C ABI
                             (include/bigstitch.h); the product compute
                             path. No CPU fallback: import/use raises
                             without the built extension + a GPU.
  _native                  — ctypes binding of the C ABI.
  host                     — host-side work-unit planning: overlap
                             intervals from registrations, Grid.create
                             output-block decomposition, view culling
                             (the reference's driver-layer semantics).
"""

from . import host  # noqa: F401
from ._native import (  # noqa: F401
    Context,
    FUSION_AVG,
    FUSION_AVG_BLEND,
    FUSION_MAX_INTENSITY,
    NativeUnavailable,
)
