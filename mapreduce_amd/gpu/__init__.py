"""GPU tier: HBM-resident MapReduce data plane.

Replaces the reference's GridFS shuffle files (C5-C8) with partitioned
tensors exchanged by RCCL all-to-all over xGMI, and the Lua hot loops
(K1-K8) with the CDNA4 kernels in mapreduce_amd/ops/hip/."""

from . import dist as dist_utils  # noqa: F401
from .wordcount import WordCountJob  # noqa: F401
