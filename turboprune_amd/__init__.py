"""turboprune_amd — an MI355X-native lottery-ticket pruning + DDP training framework.

A from-scratch rebuild of the capabilities of TurboPrune (reference:
nelaturuharsha/TurboPrune) designed MI355X-first:

- PyTorch-ROCm orchestration, one process per GPU over RCCL/xGMI.
- Hand-written HIP/CDNA4 (gfx950) kernels for the hot ops: fused mask
  application, fused SGD-momentum + mask-reapply, radix-select kth-value
  thresholds, Philox Bernoulli mask generation, GPU data augmentation,
  fused cross-entropy, and MFMA masked GEMM.
- Masks are stored as fp32 ``*.mask`` buffers in checkpoints for format
  compatibility with the reference (reference: utils/mask_layers.py:21),
  while the compute path keeps a cached masked weight maintained by the
  fused optimizer so the forward pass never re-multiplies mask*weight.

Public surface mirrors the reference's hydra config groups
(dataset_params / optimizer_params / pruning_params / experiment_params /
model_params / cyclic_training) and its experiment-directory / checkpoint
formats (reference: utils/harness_utils.py:49-94,354-365).
"""

__version__ = "0.1.0"

from turboprune_amd.config import compose, Config  # noqa: F401
