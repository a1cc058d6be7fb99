"""mgproto_amd — an MI355X-native Mixture-of-Gaussian-Prototypes framework.

A from-scratch rebuild of the capabilities of cwangrun/MGProto (TPAMI 2025,
reference layout documented in SURVEY.md) designed MI355X-first:

* backbones (ResNet / DenseNet / VGG feature extractors) run on PyTorch-ROCm
  (MIOpen convolutions, bf16 autocast),
* the per-patch GMM prototype math (log-likelihood, top-T mining, memory-bank
  enqueue, EM, push projection) runs in hand-written CDNA4 HIP kernels
  (gfx950 MFMA + LDS tiling) exposed through ``mgproto_amd.ops``,
* multi-GPU training is one process per GPU over RCCL/xGMI
  (``mgproto_amd.parallel``), with DP-correct memory-bank / EM / push
  semantics (the reference's single-process DataParallel is not reproduced).

Public API mirrors the reference's ``model.py`` surface
(``construct_MGProto``, ``MGProto.forward(x, gt)``, ``push_forward``,
``update_GMM``, ``prune_prototypes_topM``) so checkpoints and drivers
translate 1:1.
"""

__version__ = "0.1.0"

from .model import MGProto, NonNegLinear, construct_MGProto, base_architecture_to_features  # noqa: F401
