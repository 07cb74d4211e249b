"""dbeel_amd — MI355X-native SSTable compaction engine for dbeel.

Product scope: the `LSMTree::compact` hot path
(reference lsm_tree.rs:950-1156), rebuilt from scratch as HIP/CDNA4 kernels
behind a C ABI (include/dbeel_gpu.h). Host-side mirrors of the reference's
format and trigger logic live beside it; everything else in dbeel is out of
scope by contract.
"""
from . import format  # noqa: F401
from .engine import (  # noqa: F401
    BatchJob,
    DbeelGpuError,
    Job,
    compact,
    compact_sliced,
    encode_run,
    lookup,
    pin_host,
    scan,
    unpin_host,
)

__all__ = [
    "compact", "compact_sliced", "scan", "lookup", "encode_run",
    "Job", "BatchJob", "pin_host", "unpin_host", "DbeelGpuError", "format",
]
