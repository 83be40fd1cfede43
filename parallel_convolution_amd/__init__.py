"""parallel_convolution_amd — MI355X-native iterative 2D convolution engine.

A from-scratch CDNA4/gfx950 stencil engine with the capabilities of the
MPI/OpenMP C reference `jimouris/parallel-convolution` (BASELINE.json):
2D block decomposition across the GPUs of one node, ghost halo exchange on
RCCL over xGMI, hand-written HIP convolution kernels (LDS-staged, fp32
accumulation), raw-image I/O and the reference CLI surface.

Layers:
  spec.py            problem + arithmetic contract
  parallel/decomp.py grid/halo/chunk geometry (shared by all backends)
  ops/reference.py   fp32 oracle
  cpu_engine.py      CPU protocol twin (config 1, tests)
  hip_engine.py      the product: C++/HIP/RCCL engine (csrc/)
  models/filters.py  filter definitions + filter-file format
  utils/rawio.py     raw-image block I/O
  cli.py             reference-compatible CLI
"""

__version__ = "0.1.0"
