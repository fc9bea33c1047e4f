"""adaptdl_amd: an MI355X-native elastic data-parallel training framework.

A ground-up rebuild of the capabilities of petuum/adaptdl (Pollux, OSDI'21)
for AMD Instinct MI355X (gfx950, CDNA4) nodes: adaptive batch sizing driven
by the gradient noise scale, a fitted goodput/performance model, elastic
checkpoint-restart, and an in-process Pollux-style allocator — with the hot
path (gradient bucket all-reduce + GNS statistics + optimizer step) running
on hand-written HIP kernels and RCCL over xGMI.
"""

__version__ = "0.1.0"

from adaptdl_amd import env  # noqa: F401
