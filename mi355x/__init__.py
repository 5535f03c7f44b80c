"""mi355x — an MI355X-native data-parallel training harness.

Built from scratch with the same capabilities and two-script surface as
rensortino/DDP-Tutorial (see SURVEY.md for the full reference analysis):
a torchrun-style launcher, a from-scratch DDP engine (flat-bucket gradient
reducer over RCCL/xGMI, overlapped with backward), a CIFAR-10 data pipeline
with our own DistributedSampler, and a hand-written HIP/CDNA4 (gfx950)
kernel library for the CNN hot path (implicit-GEMM conv, fused BN+ReLU,
maxpool, MFMA linear, cross-entropy, fused flat SGD).

Layout:
  mi355x.models    Net (reference-parity LeNet), ResNet-18/50
  mi355x.ops       autograd ops: HIP kernels on GPU, torch fp32 reference on CPU
  mi355x.data      CIFAR-10 pickle reader, synthetic data, DistributedSampler, loader
  mi355x.parallel  comm layer, flat-param/flat-grad management, DDP reducer, SyncBN
  mi355x.launcher  fork-per-GPU launcher with the torchrun env contract
  mi355x.csrc      HIP/CDNA4 kernels (gfx950 only, no dual paths)
"""

__version__ = "0.1.0"

from . import ops  # noqa: F401
