"""Fused BatchNorm(+Add)(+ReLU) — autograd wrapper over the gfx950 kernels.

``bn_act(x, bn, relu=..., residual=...)`` runs the fused NHWC bf16 HIP path
when applicable (CUDA + bf16 + channels-last + C%8==0 + extension built)
and falls back to the exact-equivalent unfused torch ops otherwise (CPU
path / oracle). ``FusedBatchNorm2d`` is a drop-in nn.BatchNorm2d whose
forward applies the fused op (module state dict unchanged).

rocprof motivation: MIOpen spatial-BN + the separate add/ReLU elementwise
kernels are ~28 ms of the 48 ms flagship step (profiles/ in repo).
"""

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import hip_loader


def _ext():
    try:
        return hip_loader.load()
    except Exception:  # noqa: BLE001
        return None


def _usable(x: torch.Tensor, residual) -> bool:
    if os.environ.get('MGPROTO_NO_FUSED_BN') == '1':
        return False
    if not (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4):
        return False
    if x.shape[1] % 8 != 0:
        return False
    if not x.is_contiguous(memory_format=torch.channels_last):
        return False
    if residual is not None:
        if not (residual.is_cuda and residual.dtype == torch.bfloat16
                and residual.is_contiguous(memory_format=torch.channels_last)):
            return False
    return _ext() is not None


def _use_mask() -> bool:
    """Forward emits a 1-bit relu mask per element so the backward never
    re-reads y (saves ~2 bf16 activation passes of backward traffic).

    DEFAULT ON since round 2: parity-green on MI355X (20/20
    tests/test_fused_bn_gpu.py with MGPROTO_BN_MASK=1) and faster
    end-to-end (1737.9 vs 1690.6 img/s A/B, gpurun_out/r2). Set
    MGPROTO_BN_MASK=0 to fall back to the y>0 re-read backward."""
    return os.environ.get('MGPROTO_BN_MASK', '1') == '1'


class _FusedBN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2, weight, bias, running_mean, running_var,
                training, momentum, eps, relu, residual2):
        use_mask = relu and _use_mask()
        y, mean, rstd, mask = _ext().bn_fwd(
            x2, weight.float().contiguous(), bias.float().contiguous(),
            running_mean, running_var, training, momentum, eps, relu,
            residual2, use_mask)
        # with the mask, y need not be saved at all (less activation memory)
        ctx.save_for_backward(x2, mask if use_mask else y, weight, mean, rstd)
        ctx.flags = (training, relu, residual2 is not None, use_mask)
        return y

    @staticmethod
    def backward(ctx, dy):
        x2, y_or_mask, weight, mean, rstd = ctx.saved_tensors
        training, relu, has_res, use_mask = ctx.flags
        dx, dw, db, dres = _ext().bn_bwd(
            dy.contiguous(), y_or_mask, x2, weight.float().contiguous(),
            mean, rstd, training, relu, has_res, use_mask)
        return (dx, dw.to(weight.dtype), db.to(weight.dtype), None, None,
                None, None, None, None, dres if has_res else None)


def bn_act(x: torch.Tensor, bn: nn.BatchNorm2d, relu: bool = False,
           residual=None) -> torch.Tensor:
    """y = [relu](batch_norm(x) [+ residual]) — fused on GPU, exact torch
    fallback elsewhere."""
    training = bn.training
    if training and bn.track_running_stats and bn.num_batches_tracked is not None:
        bn.num_batches_tracked.add_(1)
    momentum = bn.momentum if bn.momentum is not None else 0.1

    if _usable(x, residual):
        N, C, H, W = x.shape
        x2 = x.permute(0, 2, 3, 1).reshape(-1, C)
        res2 = (residual.permute(0, 2, 3, 1).reshape(-1, C)
                if residual is not None else None)
        y2 = _FusedBN.apply(x2, bn.weight, bn.bias, bn.running_mean,
                            bn.running_var, training, momentum, bn.eps,
                            relu, res2)
        return y2.view(N, H, W, C).permute(0, 3, 1, 2)

    y = F.batch_norm(x, bn.running_mean, bn.running_var, bn.weight, bn.bias,
                     training, momentum, bn.eps)
    if residual is not None:
        y = y + residual
    return F.relu(y) if relu else y


class FusedBatchNorm2d(nn.BatchNorm2d):
    """Drop-in BatchNorm2d with an optional fused trailing ReLU.

    Used by the DenseNet/VGG trunks (their norm->relu pairs become
    FusedBatchNorm2d(fused_relu=True) + nn.Identity()); the ResNet blocks
    fuse explicitly in their forward to also absorb the residual add.
    """

    def __init__(self, *args, fused_relu: bool = False, **kwargs):
        super().__init__(*args, **kwargs)
        self.fused_relu = fused_relu

    def forward(self, x):
        return bn_act(x, self, relu=self.fused_relu)
