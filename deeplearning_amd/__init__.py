"""deeplearning_amd — an MI355X-native computer-vision training framework.

A from-scratch re-design of the capabilities of the KKKSQJ/DeepLearning zoo
(reference: /root/reference) as ONE shared framework: PyTorch-ROCm core,
hand-written HIP/CDNA4 (gfx950) kernels for the hot ops, RCCL over xGMI for
data-parallel training.

Layout (maps to reference layers):
  core/      device/process runtime, config, logging, checkpointing   (L1/L7)
  data/      datasets, samplers, prefetcher                            (L2)
  ops/       HIP-kernel-backed ops with eager CPU references           (L0)
  models/    model zoos: classification/detection/segmentation/...     (L3)
  engine/    training engines + evaluation                             (L4/L5)
  parallel/  DDP-semantics bucketed RCCL data parallel, SyncBN         (§2.3)
"""

__version__ = "0.1.0"
