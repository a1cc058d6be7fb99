"""HIP kernel parity vs the fp32 PyTorch oracle (GPU only).

Every kernel is compared against the plain-PyTorch reference implementation
of the same op (mgproto_amd.ops.reference), per the numerics-test contract.
"""

import math

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

from mgproto_amd.ops import reference as R


def _ext():
    from mgproto_amd.ops import hip_loader
    return hip_loader.load()


def make_gmm(N, P, d, device, seed=0, uniform_sigma=True):
    g = torch.Generator().manual_seed(seed)
    feat = F.normalize(torch.randn(N, d, generator=g), dim=1).to(device)
    means = F.normalize(torch.rand(P, d, generator=g), dim=1).to(device)
    covs = torch.full((P, d), 1 / math.sqrt(2 * math.pi), device=device)
    if not uniform_sigma:
        covs = covs * (0.5 + torch.rand(P, d, generator=g).to(device))
    return feat, means, covs


@pytest.mark.parametrize('N,P,d', [(15680, 2000, 64), (1024, 2000, 64),
                                   (62720, 2000, 64), (640, 2000, 128),
                                   (1000, 500, 64), (128, 96, 64),
                                   (512, 370, 64)])  # P%4!=0 (Pets C=37)
def test_gmm_fwd_parity(N, P, d):
    dev = torch.device('cuda')
    feat, means, covs = make_gmm(N, P, d, dev)
    Wt, bias = R.gmm_expand_params(means, covs)
    ext = _ext()
    out = ext.gmm_fwd(feat, Wt.contiguous(), bias.contiguous(), True)
    want = torch.exp(R.gmm_logprob(feat, means, covs))
    assert torch.allclose(out, want, atol=1e-4, rtol=1e-4), \
        (out - want).abs().max().item()
    # no-exp variant
    out_lp = ext.gmm_fwd(feat, Wt.contiguous(), bias.contiguous(), False)
    want_lp = R.gmm_logprob(feat, means, covs)
    assert torch.allclose(out_lp, want_lp, atol=1e-4, rtol=1e-4)


def test_gmm_fwd_vs_direct_oracle():
    """Against the literal (x-mu)^2/sigma^2 formula, non-uniform sigma."""
    dev = torch.device('cuda')
    feat, means, covs = make_gmm(4096, 2000, 64, dev, uniform_sigma=False)
    Wt, bias = R.gmm_expand_params(means, covs)
    out = _ext().gmm_fwd(feat, Wt.contiguous(), bias.contiguous(), False)
    want = R.gmm_logprob_direct(feat, means, covs)
    assert torch.allclose(out, want, atol=2e-4, rtol=1e-4), \
        (out - want).abs().max().item()


@pytest.mark.parametrize('N,P,d', [(15680, 2000, 64), (640, 2000, 128),
                                   (1000, 500, 64)])
def test_gmm_bwd_parity(N, P, d):
    dev = torch.device('cuda')
    feat, means, covs = make_gmm(N, P, d, dev, seed=3)
    Wt, bias = R.gmm_expand_params(means, covs)
    g = torch.randn(N, P, device=dev)
    got = _ext().gmm_bwd(g.contiguous(), feat, Wt.t().contiguous())
    gw = g @ Wt
    want = gw[:, :d] + 2.0 * feat * gw[:, d:]
    assert torch.allclose(got, want, atol=1e-3, rtol=1e-4), \
        (got - want).abs().max().item()


@pytest.mark.parametrize('N,P,d', [(15680, 2000, 64), (62720, 2000, 64),
                                   (640, 2000, 128), (512, 370, 64),
                                   (96, 15, 16)])  # d < KMAX tile (graph-test shape)
def test_gmm_fwd_uni_parity(N, P, d):
    """Uniform-sigma reduced kernel (half-K + cuni*||x||^2) vs oracle."""
    dev = torch.device('cuda')
    feat, means, covs = make_gmm(N, P, d, dev, seed=7)
    inv_var = float(1.0 / (covs.flatten()[0] ** 2))
    A = (means * inv_var).contiguous()
    bias = (-0.5 * d * math.log(2 * math.pi)
            + 0.5 * d * math.log(inv_var)
            - 0.5 * inv_var * (means * means).sum(dim=1)).contiguous()
    rn2 = (feat * feat).sum(dim=1).contiguous()
    out = _ext().gmm_fwd_uni(feat, A, bias, rn2, -0.5 * inv_var, True)
    want = torch.exp(R.gmm_logprob(feat, means, covs))
    assert torch.allclose(out, want, atol=1e-4, rtol=1e-4), \
        (out - want).abs().max().item()
    out_lp = _ext().gmm_fwd_uni(feat, A, bias, rn2, -0.5 * inv_var, False)
    want_lp = R.gmm_logprob(feat, means, covs)
    assert torch.allclose(out_lp, want_lp, atol=1e-4, rtol=1e-4)


@pytest.mark.parametrize('N,P,d', [(15680, 2000, 64), (640, 2000, 128),
                                   (1000, 500, 64),
                                   (96, 16, 16)])  # d < KMAX: w staging must
                                                   # not read past d rows
def test_gmm_bwd_uni_parity(N, P, d):
    dev = torch.device('cuda')
    feat, means, covs = make_gmm(N, P, d, dev, seed=9)
    inv_var = float(1.0 / (covs.flatten()[0] ** 2))
    cuni = -0.5 * inv_var
    A = (means * inv_var).contiguous()
    g = torch.randn(N, P, device=dev)
    rs = g.sum(dim=1).contiguous()
    got = _ext().gmm_bwd_uni(g.contiguous(), feat, A.t().contiguous(),
                             rs, cuni)
    want = g @ A + (2.0 * cuni) * feat * rs.unsqueeze(1)
    assert torch.allclose(got, want, atol=1e-3, rtol=1e-4), \
        (got - want).abs().max().item()


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
@pytest.mark.parametrize('B,C,H,W', [(80, 64, 14, 14), (3, 8, 5, 7),
                                     (1, 4, 1, 1), (16, 128, 28, 28)])
def test_up2x_parity(B, C, H, W, dtype):
    """HIP 2x bilinear upsample fwd/bwd vs torch F.interpolate autograd."""
    dev = torch.device('cuda')
    g = torch.Generator().manual_seed(B + C)
    x = torch.randn(B, C, H, W, generator=g).to(dev).to(dtype) \
        .contiguous(memory_format=torch.channels_last)

    got = _ext().up2x_fwd(x)
    want = F.interpolate(x.float(), scale_factor=2, mode='bilinear',
                         align_corners=False)
    atol = 1e-6 if dtype == torch.float32 else 3e-2
    assert torch.allclose(got.float(), want, atol=atol, rtol=1e-3), \
        (got.float() - want).abs().max().item()

    gout = torch.randn(B, C, 2 * H, 2 * W, generator=g).to(dev).to(dtype) \
        .contiguous(memory_format=torch.channels_last)
    gin = _ext().up2x_bwd(gout)
    xr = x.float().clone().requires_grad_(True)
    F.interpolate(xr, scale_factor=2, mode='bilinear',
                  align_corners=False).backward(gout.float())
    atol_b = 1e-5 if dtype == torch.float32 else 5e-2
    assert torch.allclose(gin.float(), xr.grad, atol=atol_b, rtol=1e-3), \
        (gin.float() - xr.grad).abs().max().item()


def test_upsample2x_module_autograd():
    """Through the Upsample2x module with autograd, vs the torch op."""
    from mgproto_amd.models.upsample import Upsample2x
    dev = torch.device('cuda')
    x1 = torch.randn(4, 16, 7, 9, device=dev) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    x2 = x1.detach().clone().requires_grad_(True)
    y1 = Upsample2x()(x1)
    y2 = F.interpolate(x2, scale_factor=2, mode='bilinear',
                       align_corners=False)
    assert torch.allclose(y1, y2, atol=1e-6)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert torch.allclose(x1.grad, x2.grad, atol=1e-5), \
        (x1.grad - x2.grad).abs().max().item()


def test_gmm_scores_strided_view_input():
    """B=1 push sweeps pass a non-copy permute+reshape VIEW as feat
    (caught by the GPU eval-driver run, round 2): the kernels need packed
    rows, so dispatch must make it contiguous."""
    from mgproto_amd import ops
    dev = torch.device('cuda')
    _, means, covs = make_gmm(16, 40, 16, dev, seed=13)
    base = torch.randn(1, 16, 4, 4, device=dev)
    feat = base.permute(0, 2, 3, 1).reshape(16, 16)    # strided view
    assert not feat.is_contiguous()
    out = ops.gmm_scores(feat, means, covs, apply_exp=False)
    want = R.gmm_logprob(feat.contiguous().cpu(), means.cpu(), covs.cpu())
    assert torch.allclose(out.cpu(), want, atol=1e-4, rtol=1e-4)


def test_gmm_scores_dispatch_paths():
    """Frozen-isotropic covs take the reduced path; general diag sigma
    keeps the full GEMM — both match the oracle."""
    from mgproto_amd import ops
    dev = torch.device('cuda')
    for uniform in (True, False):
        feat, means, covs = make_gmm(2048, 500, 64, dev, seed=11,
                                     uniform_sigma=uniform)
        out = ops.gmm_scores(feat, means, covs, apply_exp=False)
        want = R.gmm_logprob(feat.cpu(), means.cpu(), covs.cpu())
        assert torch.allclose(out.cpu(), want, atol=1e-4, rtol=1e-4), uniform


@pytest.mark.parametrize('P', [2000, 370])  # 370: exercises bwd padding
def test_gmm_autograd_end_to_end(P):
    """Through ops.gmm_scores (native path) vs the CPU reference autograd."""
    from mgproto_amd import ops
    dev = torch.device('cuda')
    feat, means, covs = make_gmm(1024, P, 64, dev, seed=5)
    f_gpu = feat.clone().requires_grad_(True)
    out = ops.gmm_scores(f_gpu, means, covs, apply_exp=True)
    gout = torch.randn_like(out)
    out.backward(gout)

    f_cpu = feat.cpu().clone().requires_grad_(True)
    out_cpu = R.gmm_probs(f_cpu, means.cpu(), covs.cpu())
    out_cpu.backward(gout.cpu())
    assert torch.allclose(out.cpu(), out_cpu, atol=1e-4, rtol=1e-4)
    assert torch.allclose(f_gpu.grad.cpu(), f_cpu.grad, atol=1e-3, rtol=1e-3)


@pytest.mark.parametrize('B,HW,P,T', [(8, 196, 2000, 20), (80, 784, 2000, 20),
                                      (2, 49, 100, 5), (3, 196, 2000, 1)])
def test_topk_hw_parity(B, HW, P, T):
    dev = torch.device('cuda')
    probs = torch.rand(B, HW, P, device=dev)
    vals, idx = _ext().topk_hw(probs, T)
    want_v, _ = torch.topk(probs.permute(0, 2, 1), T, dim=2)
    assert torch.equal(vals, want_v)
    # indices: must address the reported values (tie ORDER differs between
    # the kernel [lowest-index-wins, the reference CPU semantic] and CUDA
    # torch.topk [unspecified]) and be unique per row
    gathered = probs.permute(0, 2, 1).gather(2, idx.long())
    assert torch.equal(gathered, vals)
    sidx, _ = idx.long().sort(dim=2)
    assert (sidx[..., 1:] != sidx[..., :-1]).all() or T == 1


def test_topk_ties_lowest_index():
    dev = torch.device('cuda')
    probs = torch.ones(1, 32, 64, device=dev)
    vals, idx = _ext().topk_hw(probs, 4)
    assert torch.equal(idx[0, 0].cpu(), torch.tensor([0, 1, 2, 3], dtype=torch.int32))


def test_argmax_hw_parity():
    dev = torch.device('cuda')
    probs = torch.rand(7, 196, 2000, device=dev)
    vals, idx = _ext().argmax_hw(probs)
    want_v, want_i = probs.max(dim=1)
    assert torch.equal(vals, want_v)
    assert torch.equal(idx.long(), want_i)


def test_enqueue_and_em_on_gpu():
    """The batched enqueue + EM torch ops run correctly on device."""
    dev = torch.device('cuda')
    B, C, K, HW, d = 16, 10, 4, 49, 64
    P = C * K
    feat = F.normalize(torch.randn(B * HW, d, device=dev), dim=1)
    top1 = torch.randint(0, HW, (B, P), device=dev)
    gt = torch.randint(0, C, (B,), device=dev)
    feats, labs = R.enqueue_candidates(feat, top1, gt, C, K, HW)
    f_cpu, l_cpu = R.enqueue_candidates(feat.cpu(), top1.cpu(), gt.cpu(), C, K, HW)
    assert torch.equal(labs.cpu(), l_cpu)
    assert torch.allclose(feats.cpu(), f_cpu, atol=1e-6)

    G, N = 4, 32
    x = F.normalize(torch.randn(G, N, d, device=dev), dim=2)
    means = F.normalize(torch.rand(G, K, d, device=dev), dim=2)
    covs = torch.full((G, K, d), 1 / math.sqrt(2 * math.pi), device=dev)
    pi = torch.softmax(torch.rand(G, K, device=dev), dim=1)
    wlp, lr_ = R.em_e_step(x, means, covs, pi)
    wlp_c, lr_c = R.em_e_step(x.cpu(), means.cpu(), covs.cpu(), pi.cpu())
    assert torch.allclose(wlp.cpu(), wlp_c, atol=1e-4)
    assert torch.allclose(lr_.cpu(), lr_c, atol=1e-4)


@pytest.mark.gpu
@pytest.mark.parametrize('d', [144, 60])   # beyond the kernel envelope
def test_gmm_dispatch_fallback_out_of_envelope(d):
    """d > 128 or d % 8 != 0 routes to the GEMM fallback (hipBLASLt via
    addmm) transparently — same numerics, gradient flows."""
    import mgproto_amd.ops as O
    dev = torch.device('cuda', 0)
    feat, means, covs = make_gmm(256, 40, d, dev, seed=9)
    feat = feat.clone().requires_grad_(True)
    out = O.gmm_scores(feat, means, covs, apply_exp=True)
    ref = R.gmm_logprob_direct(feat.detach().float(),
                               means.float(), covs.float()).exp()
    assert torch.allclose(out, ref, atol=1e-4, rtol=1e-4)
    out.sum().backward()
    assert feat.grad is not None and torch.isfinite(feat.grad).all()


@pytest.mark.gpu
def test_opt_in_kernel_bindings_present():
    """The gated K5-K7 kernels are built and bound even when not enabled
    (their execution parity tests live in test_{em,enqueue}_hip_gpu.py,
    gated on MGPROTO_HIP_EM / MGPROTO_HIP_ENQUEUE)."""
    ext = _ext()
    for sym in ('em_estep', 'em_mstep', 'enqueue_rows', 'bank_push'):
        assert hasattr(ext, sym), sym
