"""bf16 MFMA 1x1-conv GEMM (round-3 lever) — parity vs the fp32 oracle.

Gated on MGPROTO_GEMM1X1_HIP=1 (opt-in until it earns a default via the
fused-BN-stats dispatch, docs/DESIGN_ROUND3_CONV.md)."""

import os

import pytest
import torch

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(os.environ.get('MGPROTO_GEMM1X1_HIP') != '1',
                       reason='hand-written 1x1 GEMM is opt-in '
                              '(MGPROTO_GEMM1X1_HIP=1)'),
]


def _ext():
    from mgproto_amd.ops import hip_loader
    return hip_loader.load()


def _mk(M, K, N, seed=0, bias=False):
    g = torch.Generator().manual_seed(seed)
    x = (torch.randn(M, K, generator=g) / (K ** 0.5)).bfloat16().cuda()
    w = (torch.randn(N, K, generator=g) / (K ** 0.5)).bfloat16().cuda()
    b = torch.randn(N, generator=g).float().cuda() if bias else None
    return x, w, b


def _oracle(x, w, b):
    want = x.float() @ w.float().t()
    if b is not None:
        want = want + b.unsqueeze(0)
    return want.bfloat16().float()


# both tile configs (N<=256 -> 256x64; N>=512 -> 128x128), M%BM!=0 edges,
# every K in the flagship set
@pytest.mark.parametrize('M,K,N', [
    (62720, 256, 1024), (62720, 1024, 256), (15680, 512, 2048),
    (15680, 2048, 512), (1000, 64, 64), (250880, 128, 512),
    (777, 192, 80)])
def test_gemm1x1_fwd_parity(M, K, N):
    x, w, b = _mk(M, K, N, seed=M % 97, bias=True)
    y, _ = _ext().gemm1x1_fwd(x, w, b, False)
    want = _oracle(x, w, b)
    # bf16-rounded output vs bf16-rounded fp32 oracle: the only gap is
    # fp32 accumulation order
    assert torch.allclose(y.float(), want, rtol=2e-2, atol=1e-2), \
        (y.float() - want).abs().max().item()


def test_gemm1x1_identity_catches_transpose():
    """A = I with an ASYMMETRIC B (guide rule): y must equal B^T rows."""
    K = 64
    x = torch.eye(K).bfloat16().cuda()
    w = (torch.arange(K * 128, dtype=torch.float32).view(128, K)
         / (K * 128)).bfloat16().cuda()
    y, _ = _ext().gemm1x1_fwd(x, w, None, False)
    want = w.float().t().bfloat16().float()
    assert torch.allclose(y.float(), want, rtol=1e-2, atol=1e-3), \
        (y.float() - want).abs().max().item()


def test_gemm1x1_bn_partials():
    """Per-block partials merge to the exact per-channel sum / sumsq of
    the ROUNDED output (what the fused-BN pipeline consumes)."""
    M, K, N = 15680, 256, 128
    x, w, b = _mk(M, K, N, seed=3, bias=False)
    y, partials = _ext().gemm1x1_fwd(x, w, None, True)
    assert partials.numel() > 0
    merged = partials.sum(dim=0)                      # [2N]
    yf = y.float()
    want_sum = yf.sum(dim=0)
    want_sq = (yf * yf).sum(dim=0)
    assert torch.allclose(merged[:N], want_sum, rtol=1e-3, atol=1e-2), \
        (merged[:N] - want_sum).abs().max().item()
    assert torch.allclose(merged[N:], want_sq, rtol=1e-3, atol=1e-2), \
        (merged[N:] - want_sq).abs().max().item()
