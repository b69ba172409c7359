"""improved_body_parts_amd — an MI355X-native bottom-up multi-person 2D pose
estimation framework.

A from-scratch re-design of the capabilities of hellojialee/Improved-Body-Parts
("SimplePose", AAAI-2020) for AMD Instinct MI355X (gfx950 / CDNA4):

  * PyTorch-ROCm front-end with the reference's `PoseNet` API and checkpoint format
  * hand-written CDNA4 HIP kernels for the hot ops (MFMA implicit-GEMM conv,
    fused BN+LeakyReLU, fused focal-L2 loss, on-device GT generation,
    on-device keypoint post-processing) — no CUDA shims, no Apex, native bf16
  * one-process-per-GPU data parallelism over RCCL/xGMI with bucketed
    all-reduce overlapped with backward

Layout:
  config/    canonical skeleton + training + inference configuration
  models/    PoseNet (stacked IMHN hourglass), losses
  ops/       operator dispatch + HIP kernels (csrc/)
  data/      GT heatmapper (CPU oracle + device kernels), augmentation, datasets
  parallel/  RCCL DDP engine, SyncBN
  engine/    training / evaluation drivers, checkpointing
  utils/     padding, NMS, meters, logging
"""
__version__ = "0.1.0"
