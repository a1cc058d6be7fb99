"""1x1 convolutions as hipBLASLt GEMMs.

A stride-1 1x1 conv on an NHWC tensor IS the GEMM ``[N*H*W, Cin] x
[Cin, Cout]``. MIOpen routes these through CK batched-GEMM kernels that
measure ~45 TF effective on the flagship step (profiles/ — ~11 ms/step);
``torch.matmul`` on the same shapes hits hipBLASLt's tuned bf16 GEMMs.
Per the MI355X design rules, plain library GEMMs belong to
hipBLASLt/rocBLAS — hand-written MFMA kernels are reserved for fused ops.

``conv1x1`` falls back to F.conv2d off the fast path (CPU, fp32,
non-channels-last, strided, or grouped), which is numerically the
reference behavior.
"""

import os

import torch
import torch.nn as nn
import torch.nn.functional as F


def _usable(x: torch.Tensor, conv: nn.Conv2d) -> bool:
    # Measured on the flagship step: plain hipBLASLt matmul on these
    # skinny-K shapes is ~35% SLOWER than MIOpen's CK batched-GEMM picks
    # (1236 vs 1666 img/s, profiles/bench_history_r01.md). Round 2 closed
    # the question: TunableOp-tuned hipBLASLt still loses end-to-end
    # (1646.9 vs 1994.2 img/s) and per shape CK wins 12 of 14 — a hybrid
    # would save 0.04 ms of 4.32 (profiles/bench_r02/conv1x1_per_shape.log).
    # The GEMM path stays opt-in for A/B only.
    if os.environ.get('MGPROTO_GEMM_CONV1X1') != '1':
        return False
    return (x.is_cuda and x.dim() == 4
            and x.dtype in (torch.bfloat16, torch.float16)
            and conv.kernel_size == (1, 1)
            and conv.stride == (1, 1)
            and conv.padding == (0, 0)
            and conv.groups == 1
            and x.is_contiguous(memory_format=torch.channels_last))


def conv1x1(x: torch.Tensor, conv: nn.Conv2d) -> torch.Tensor:
    """Apply a 1x1 nn.Conv2d, as a GEMM when the layout allows."""
    if not _usable(x, conv):
        return F.conv2d(x, conv.weight, conv.bias, conv.stride,
                        conv.padding, conv.dilation, conv.groups)
    N, C, H, W = x.shape
    Co = conv.out_channels
    x2 = x.permute(0, 2, 3, 1).reshape(-1, C)          # free view (NHWC)
    w = conv.weight.view(Co, C)
    if conv.bias is not None:
        y2 = torch.addmm(conv.bias.to(x2.dtype), x2, w.t().to(x2.dtype))
    else:
        y2 = x2 @ w.t().to(x2.dtype)
    return y2.view(N, H, W, Co).permute(0, 3, 1, 2)    # channels_last NCHW


class GemmConv2d(nn.Conv2d):
    """Drop-in nn.Conv2d that takes the hipBLASLt GEMM path for stride-1
    1x1 convs on channels-last bf16 inputs (state dict unchanged)."""

    def forward(self, x):
        return conv1x1(x, self)
