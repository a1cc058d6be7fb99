"""CPU semantics of the fused/drop-in modules: off the HIP fast path,
FusedBatchNorm2d and GemmConv2d must be bit-compatible with the stock
modules (state dict, running stats, numerics), and PhaseTimer must
aggregate when enabled."""

import os

import torch
import torch.nn as nn


def test_fused_bn_cpu_matches_stock():
    from mgproto_amd.models.fused_bn import FusedBatchNorm2d
    torch.manual_seed(0)
    ref = nn.BatchNorm2d(8)
    fbn = FusedBatchNorm2d(8)
    fbn.load_state_dict(ref.state_dict())
    x = torch.randn(4, 8, 5, 5)
    ref.train(), fbn.train()
    for _ in range(3):
        a, b = ref(x), fbn(x)
    assert torch.allclose(a, b, atol=1e-6)
    assert torch.allclose(ref.running_mean, fbn.running_mean, atol=1e-6)
    assert torch.allclose(ref.running_var, fbn.running_var, atol=1e-6)
    ref.eval(), fbn.eval()
    assert torch.allclose(ref(x), fbn(x), atol=1e-6)
    # fused_relu clamps
    frelu = FusedBatchNorm2d(8, fused_relu=True)
    frelu.load_state_dict(ref.state_dict())
    frelu.eval()
    assert torch.allclose(frelu(x), ref(x).relu(), atol=1e-6)


def test_gemm_conv1x1_cpu_matches_stock():
    from mgproto_amd.models.conv1x1 import GemmConv2d
    torch.manual_seed(0)
    ref = nn.Conv2d(6, 10, kernel_size=1)
    g = GemmConv2d(6, 10, kernel_size=1)
    g.load_state_dict(ref.state_dict())
    x = torch.randn(2, 6, 7, 7)
    assert torch.allclose(ref(x), g(x), atol=1e-6)
    # gradient path identical
    (g(x).square().mean()).backward()
    assert g.weight.grad is not None


def test_phase_timer_enabled(monkeypatch):
    monkeypatch.setenv('MGPROTO_TIMING', '1')
    from mgproto_amd.utils.timing import PhaseTimer
    t = PhaseTimer(device=torch.device('cpu'))
    assert t.enabled
    with t.phase('fwd'):
        torch.randn(64, 64) @ torch.randn(64, 64)
    with t.phase('fwd'):
        pass
    s = t.summary()
    assert 'fwd' in s and s['fwd'] >= 0.0


def test_phase_timer_disabled_is_noop():
    os.environ.pop('MGPROTO_TIMING', None)
    from mgproto_amd.utils.timing import PhaseTimer
    t = PhaseTimer(device=torch.device('cpu'))
    assert not t.enabled
    with t.phase('x'):
        pass
    assert t.summary() == {}
