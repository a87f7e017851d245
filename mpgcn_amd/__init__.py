"""MPGCN-MI355X: an AMD MI355X-native origin-destination flow forecasting framework.

Re-implements the capabilities of underdoc-wang/MPGCN (ICDE'20) from scratch for
CDNA4 (gfx950): hand-written HIP/MFMA kernels for the 2-D graph convolution, a fused
HIP LSTM cell, on-device graph-support builders, and RCCL-over-xGMI data parallelism.
PyTorch-ROCm provides the framework layer (autograd, optimizer, host runtime) only.

Layering (physical, unlike the reference's flat files — see SURVEY.md §7):
  mpgcn_amd.ops      — HIP extension + autograd wrappers (eager fallback on CPU)
  mpgcn_amd.graph    — graph-support builders (adjacency preprocessing, dynamic graphs)
  mpgcn_amd.models   — BDGCN / MPGCN modules (checkpoint-compatible state_dict keys)
  mpgcn_amd.data     — OD data containers: npz loader, synthetic generator, windowing
  mpgcn_amd.parallel — process-group bootstrap, bucketed DP grad all-reduce, region partition
  mpgcn_amd.train    — trainer (reference-compatible loop/checkpoint/scores) + metrics
"""

__version__ = "0.1.0"

from mpgcn_amd import graph, ops  # noqa: F401
