"""dbeel_amd — MI355X-native SSTable compaction engine for dbeel.

Product scope: the `LSMTree::compact` hot path
(reference lsm_tree.rs:950-1156), rebuilt from scratch as HIP/CDNA4 kernels
behind a C ABI (include/dbeel_gpu.h). Host-side mirrors of the reference's
format and trigger logic live beside it; everything else in dbeel is out of
scope by contract.
"""
from . import format  # noqa: F401
from .engine import Job, DbeelGpuError, compact  # noqa: F401

__all__ = ["compact", "Job", "DbeelGpuError", "format"]
