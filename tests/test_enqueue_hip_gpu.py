"""HIP enqueue/bank-push kernels (K5) vs the torch path — gated on
MGPROTO_HIP_ENQUEUE=1 (opt-in until validated on hardware)."""

import os

import pytest
import torch
import torch.nn.functional as F

from mgproto_amd.ops import reference as R
from mgproto_amd.utils.memory import MemoryBank

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(os.environ.get('MGPROTO_HIP_ENQUEUE') != '1',
                       reason='HIP enqueue path is opt-in '
                              '(MGPROTO_HIP_ENQUEUE=1)'),
]


@pytest.mark.parametrize('B,C,K,HW', [(80, 200, 10, 784), (7, 5, 3, 16),
                                      (1, 2, 1, 4), (64, 37, 10, 196)])
def test_enqueue_rows_matches_reference(B, C, K, HW):
    import mgproto_amd.ops as O
    dev = torch.device('cuda', 0)
    g = torch.Generator().manual_seed(B + C)
    feat = torch.randn(B * HW, 16, generator=g).to(dev)
    top1 = torch.randint(0, HW, (B, C * K), generator=g).to(dev)
    gt = torch.randint(0, C, (B,), generator=g).to(dev)
    f_hip, l_hip = O.enqueue_candidates(feat, top1, gt, C, K, HW)
    f_ref, l_ref = R.enqueue_candidates(feat, top1, gt, C, K, HW)
    assert torch.equal(l_hip, l_ref)
    assert torch.equal(f_hip, f_ref)


@pytest.mark.parametrize('seed', [0, 1, 2])
def test_bank_push_matches_torch_path(seed):
    """Random multi-step push streams (incl. sentinels and oversized
    pushes): bit-identical bank state vs the torch scatter path."""
    dev = torch.device('cuda', 0)
    C, d, cap = 6, 16, 4
    hip_bank = MemoryBank(C, d, capacity=C * cap).to(dev)
    ref_bank = MemoryBank(C, d, capacity=C * cap).to(dev)
    g = torch.Generator().manual_seed(seed)
    for step in range(15):
        M = int(torch.randint(0, 40, (1,), generator=g))
        feats = torch.randn(M, d, generator=g).to(dev)
        labels = torch.randint(0, C + 1, (M,), generator=g).to(dev)  # incl sentinel
        hip_bank.push(feats, labels)                  # env on -> HIP kernel
        os.environ['MGPROTO_HIP_ENQUEUE'] = '0'
        try:
            ref_bank.push(feats, labels)              # torch path
        finally:
            os.environ['MGPROTO_HIP_ENQUEUE'] = '1'
        assert torch.equal(hip_bank.mem, ref_bank.mem), step
        assert torch.equal(hip_bank.mem_len, ref_bank.mem_len), step
        assert torch.equal(hip_bank.head, ref_bank.head), step


def test_forward_enqueue_matches_torch_path():
    """In-situ parity: drive BOTH enqueue paths from the SAME forward's
    tensors (the real strided top-1 slice, real features) and compare the
    resulting bank states bit-for-bit.

    Round-2 note: the original form of this test built two models and
    compared their banks after separate forwards; the round-2 debug run
    (gpurun_out/r2b/enqueue_debug.log) showed the FEATURES already differ
    between two same-seed model instances in one process (MIOpen algo
    timing jitter upstream), so cross-run bitwise bank equality was an
    invalid oracle — the kernels themselves are bit-identical on the same
    inputs, which is what this asserts."""
    import mgproto_amd.ops as O
    from mgproto_amd.model import construct_MGProto

    os.environ['MGPROTO_HIP_ENQUEUE'] = '1'
    torch.manual_seed(0)
    C, K, d = 5, 3, 16
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(C * K, d, 1, 1),
                          num_classes=C, add_on_layers_type='regular',
                          sz_embedding=8, mem_capacity=8,
                          mine_K=2).cuda()
    captured = []
    orig = O.enqueue_candidates

    def spy(feat, top1, gt, C_, K_, HW_):
        captured.append((feat, top1, gt, C_, K_, HW_))
        return orig(feat, top1, gt, C_, K_, HW_)

    import mgproto_amd.model as MM
    MM.ops.enqueue_candidates = spy
    try:
        g = torch.Generator().manual_seed(3)
        x = torch.randn(6, 3, 64, 64, generator=g).cuda()
        y = (torch.arange(6) % C).cuda()
        with torch.no_grad():
            m(x, y)
            m(x.flip(0), y.flip(0))
    finally:
        MM.ops.enqueue_candidates = orig
    assert len(captured) == 2

    # replay both paths over the captured in-situ tensors
    banks = {}
    for env in ('1', '0'):
        os.environ['MGPROTO_HIP_ENQUEUE'] = env
        bank = MemoryBank(C, d, capacity=C * 8).cuda()
        for (feat, top1, gt, C_, K_, HW_) in captured:
            f, lab = O.enqueue_candidates(feat, top1, gt, C_, K_, HW_)
            bank.push(f, lab)
        banks[env] = (bank.mem.cpu(), bank.mem_len.cpu(), bank.head.cpu())
    os.environ['MGPROTO_HIP_ENQUEUE'] = '1'
    assert torch.equal(banks['1'][0], banks['0'][0])
    assert torch.equal(banks['1'][1], banks['0'][1])
    assert torch.equal(banks['1'][2], banks['0'][2])
