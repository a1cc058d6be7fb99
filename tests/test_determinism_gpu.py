"""Determinism on GPU: identical inputs must give bitwise-identical results
for every path that feeds distributed state (SURVEY.md §5 race detection —
the reference has an unguarded DataParallel buffer race; here determinism
is by construction and these tests pin it)."""

import math

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu


def test_fused_bn_deterministic():
    from mgproto_amd.models.fused_bn import bn_act
    torch.manual_seed(0)
    bn = torch.nn.BatchNorm2d(64).cuda().train()
    x = torch.randn(8, 64, 28, 28).cuda().to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    r = torch.randn_like(x).contiguous(memory_format=torch.channels_last)

    outs, grads = [], []
    for _ in range(3):
        torch.manual_seed(1)
        bn2 = torch.nn.BatchNorm2d(64).cuda().train()
        bn2.load_state_dict(bn.state_dict())
        x1 = x.clone().requires_grad_(True)
        y = bn_act(x1, bn2, relu=True, residual=r)
        y.float().sum().backward()
        outs.append(y.detach().clone())
        grads.append(x1.grad.clone())
    assert torch.equal(outs[0], outs[1]) and torch.equal(outs[1], outs[2])
    assert torch.equal(grads[0], grads[1]) and torch.equal(grads[1], grads[2])


def test_gmm_kernels_deterministic():
    from mgproto_amd.ops import hip_loader, reference as R
    ext = hip_loader.load()
    g = torch.Generator().manual_seed(0)
    feat = F.normalize(torch.randn(4096, 64, generator=g), dim=1).cuda()
    means = F.normalize(torch.rand(2000, 64, generator=g), dim=1).cuda()
    covs = torch.full((2000, 64), 1 / math.sqrt(2 * math.pi)).cuda()
    Wt, bias = R.gmm_expand_params(means, covs)
    o1 = ext.gmm_fwd(feat, Wt.contiguous(), bias.contiguous(), True)
    o2 = ext.gmm_fwd(feat, Wt.contiguous(), bias.contiguous(), True)
    assert torch.equal(o1, o2)
    gr = torch.randn(4096, 2000, device='cuda')
    b1 = ext.gmm_bwd(gr.contiguous(), feat, Wt.t().contiguous())
    b2 = ext.gmm_bwd(gr.contiguous(), feat, Wt.t().contiguous())
    assert torch.equal(b1, b2)


def test_em_update_deterministic():
    from mgproto_amd.model import construct_MGProto
    results = []
    for _ in range(2):
        torch.manual_seed(0)
        m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(40, 64, 1, 1), num_classes=10,
                              add_on_layers_type='regular', sz_embedding=8,
                              mem_capacity=16, mine_K=3).cuda()
        g = torch.Generator().manual_seed(7)
        feats = F.normalize(torch.randn(160, 64, generator=g), dim=1).cuda()
        labels = torch.arange(10).repeat_interleave(16).cuda()
        m.queue.push(feats, labels)
        m.memory_updated_cls[:] = True
        m.update_GMM()
        torch.cuda.synchronize()
        results.append((m.prototype_means.data.clone(),
                        m.last_layer.weight.data.clone()))
    assert torch.equal(results[0][0], results[1][0])
    assert torch.equal(results[0][1], results[1][1])


def test_enqueue_deterministic():
    from mgproto_amd.ops import reference as R
    g = torch.Generator().manual_seed(3)
    B, C, K, HW, d = 32, 10, 4, 196, 64
    feat = torch.randn(B * HW, d, generator=g).cuda()
    top1 = torch.randint(0, HW, (B, C * K), generator=g).cuda()
    gt = torch.randint(0, C, (B,), generator=g).cuda()
    f1, l1 = R.enqueue_candidates(feat, top1, gt, C, K, HW)
    f2, l2 = R.enqueue_candidates(feat, top1, gt, C, K, HW)
    assert torch.equal(f1, f2) and torch.equal(l1, l2)
