"""curvine_amd — an MI355X-native distributed cache engine.

A from-scratch re-design of the capabilities of CurvineIO/curvine
(reference: /root/reference, Rust) for AMD Instinct MI355X nodes:

* the worker's multi-tier block store promotes HBM3E (288 GB per GPU) to the
  hot tier above host-DRAM and NVMe/file tiers,
* the byte-moving data pipeline (CRC32C, scatter/gather coalesce, zero-fill,
  block copy) is hand-written HIP for CDNA4 (gfx950),
* intra-node block distribution uses RCCL/xGMI (torch.distributed "nccl"
  backend on ROCm) instead of TCP,
* a libfuse-free FUSE server speaks the raw /dev/fuse protocol, with the hot
  read/write loop in C++ feeding from pinned staging buffers filled by
  hipMemcpyAsync from the HBM arena.

The control plane (master metadata service, RPC, journal) mirrors the
reference's architecture but is a new implementation.
"""

__version__ = "0.1.0"

from curvine_amd.errors import FsError, ErrorCode  # noqa: F401
from curvine_amd.conf import ClusterConf  # noqa: F401
