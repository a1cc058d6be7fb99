"""HIP EM kernels (K6/K7) vs the torch oracle path — gated on
MGPROTO_HIP_EM=1 (the kernels are opt-in until measured; flip the env and
run this file on an MI355X to validate them)."""

import math
import os

import pytest
import torch
import torch.nn.functional as F

from mgproto_amd.ops import reference as R

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(os.environ.get('MGPROTO_HIP_EM') != '1',
                       reason='HIP EM path is opt-in (MGPROTO_HIP_EM=1)'),
]


def _ext():
    from mgproto_amd.ops import hip_loader
    return hip_loader.load()


def _mk(G, N, K, d, seed=0, uniform=True):
    g = torch.Generator().manual_seed(seed)
    dev = torch.device('cuda', 0)
    x = F.normalize(torch.randn(G, N, d, generator=g), dim=2).to(dev)
    means = F.normalize(torch.rand(G, K, d, generator=g), dim=2).to(dev)
    covs = torch.full((G, K, d), 1 / math.sqrt(2 * math.pi), device=dev)
    if not uniform:
        covs = covs * (0.5 + torch.rand(G, K, d, generator=g).to(dev))
    pi = torch.softmax(torch.rand(G, K, generator=g), dim=1).to(dev)
    return x, means, covs, pi


@pytest.mark.parametrize('G,N,K,d', [(8, 800, 10, 64), (3, 100, 32, 128),
                                     (1, 50, 1, 8), (5, 257, 7, 40)])
def test_estep_matches_torch_path(G, N, K, d):
    import mgproto_amd.ops as O
    x, means, covs, pi = _mk(G, N, K, d, seed=G + K)
    wlp_hip, lr_hip = O.em_e_step(x, means, covs, pi)
    wlp_ref, lr_ref = R.em_e_step(x, means, covs, pi)
    assert torch.allclose(wlp_hip, wlp_ref, atol=2e-4, rtol=1e-4), \
        (wlp_hip - wlp_ref).abs().max().item()
    assert torch.allclose(lr_hip, lr_ref, atol=2e-4, rtol=1e-4)


@pytest.mark.parametrize('G,N,K,d', [(8, 800, 10, 64), (3, 100, 32, 128),
                                     (1, 50, 1, 8), (5, 257, 7, 40)])
def test_mstep_matches_torch_path(G, N, K, d):
    import mgproto_amd.ops as O
    x, means, covs, pi = _mk(G, N, K, d, seed=2 * G + K, uniform=False)
    wlp, log_resp = R.em_e_step(x, means, covs, pi)
    g_hip, pi_hip = O.em_m_step_grads(x, log_resp, wlp, means, covs)
    g_ref, pi_ref = R.em_m_step_grads(x, log_resp, wlp, means, covs)
    assert torch.allclose(g_hip, g_ref, atol=2e-4, rtol=1e-3), \
        (g_hip - g_ref).abs().max().item()
    assert torch.allclose(pi_hip, pi_ref, atol=1e-5)


def test_update_gmm_matches_torch_path():
    """Full update_GMM with the HIP EM vs the default path: same means/pi."""
    from mgproto_amd.model import construct_MGProto

    def run(enable):
        os.environ['MGPROTO_HIP_EM'] = '1' if enable else '0'
        torch.manual_seed(0)
        C, K, d, cap = 6, 5, 64, 32
        m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1),
                              num_classes=C, add_on_layers_type='regular',
                              sz_embedding=8, mem_capacity=cap,
                              mine_K=2).cuda()
        g = torch.Generator().manual_seed(7)
        for c in range(C):
            m.queue.push(F.normalize(torch.randn(cap, d, generator=g),
                                     dim=1).cuda(),
                         torch.full((cap,), c, dtype=torch.long).cuda())
            m.memory_updated_cls[c] = True
        m.update_GMM()
        return (m.prototype_means.data.cpu(),
                m.last_layer.weight.data.cpu())
    try:
        means_hip, w_hip = run(True)
        means_ref, w_ref = run(False)
    finally:
        os.environ['MGPROTO_HIP_EM'] = '1'   # restore for this test session
    assert torch.allclose(means_hip, means_ref, atol=1e-4), \
        (means_hip - means_ref).abs().max().item()
    assert torch.allclose(w_hip, w_ref, atol=1e-5)
