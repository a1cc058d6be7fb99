"""MGProto model-level invariants (CPU; reference model.py behaviour)."""

import math

import pytest
import torch
import torch.nn.functional as F

from mgproto_amd.model import MGProto, construct_MGProto


@pytest.fixture(scope='module')
def small_model():
    torch.manual_seed(0)
    return construct_MGProto('resnet18', pretrained=False, img_size=64,
                             prototype_shape=(40, 32, 1, 1), num_classes=10,
                             add_on_layers_type='regular', sz_embedding=16,
                             mem_capacity=12, mine_K=4)


def test_forward_shapes(small_model):
    m = small_model
    x = torch.randn(3, 3, 64, 64)
    gt = torch.tensor([1, 5, 5])
    logits, emb = m(x, gt)
    assert logits.shape == (3, 10, 4)
    assert emb.shape == (3, 16)
    # embeddings are l2-normalized (model.py:184)
    assert torch.allclose(emb.norm(dim=1), torch.ones(3), atol=1e-5)


def test_forward_eval_no_enqueue(small_model):
    m = small_model
    before = m.queue.mem_len.clone()
    it_before = m.iteration_counter.clone()
    m(torch.randn(2, 3, 64, 64), None)
    assert torch.equal(m.queue.mem_len, before)
    assert torch.equal(m.iteration_counter, it_before)


def test_compute_log_prob_matches_direct(small_model):
    from mgproto_amd.ops import reference as R
    m = small_model
    feat = F.normalize(torch.randn(20, 32), dim=1)
    lp = m.compute_log_prob(feat)
    direct = R.gmm_logprob_direct(feat, m.prototype_means.data,
                                  m.prototype_covs.data)
    assert torch.allclose(lp.reshape(20, -1), direct, atol=1e-4, rtol=1e-5)


def test_push_forward_distances(small_model):
    m = small_model
    x = torch.randn(2, 3, 64, 64)
    bf, dist = m.push_forward(x)
    B, d, H, W = bf.shape
    assert dist.shape == (2, 40, H, W)
    # distances = -exp(logprob) <= 0 (model.py:437)
    assert (dist <= 0).all()
    # distances consistent with compute_log_prob on the packed features
    feat = bf.permute(0, 2, 3, 1).reshape(-1, d)
    lp = m.compute_log_prob(feat).reshape(B, H * W, 40)
    want = -lp.exp().permute(0, 2, 1).reshape(B, 40, H, W)
    assert torch.allclose(dist, want, atol=1e-5)


def test_wrong_class_mask_affects_only_levels_ge1():
    torch.manual_seed(0)
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(20, 16, 1, 1), num_classes=5,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=8, mine_K=3)
    m.eval()
    x = torch.randn(2, 3, 64, 64)
    gt = torch.tensor([0, 3])
    with torch.no_grad():
        lg_with, _ = m(x, gt)
        lg_none, _ = m(x, None)
    # level 0 is identical with and without gt (mask starts at level 1)
    assert torch.allclose(lg_with[:, :, 0], lg_none[:, :, 0], atol=1e-5)


def test_last_layer_mask_invariant(small_model):
    m = small_model
    neg = 1 - m.prototype_class_identity.t()
    assert float(m.last_layer.weight.data[neg == 1].abs().sum()) == 0.0
    # initial own-class weights are 1/K (model.py:445)
    pos = m.prototype_class_identity.t()
    K = m.num_prototypes_per_class
    init_val = m.last_layer.weight.data[pos == 1]
    # weights may have been EM-updated by other tests; just check mask holds
    assert init_val.shape[0] == m.num_prototypes


def test_update_gmm_marks_clean_and_preserves_mask():
    torch.manual_seed(0)
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(20, 16, 1, 1), num_classes=5,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=6, mine_K=3)
    # fill 3 classes, leave others partial
    for c in [0, 2, 4]:
        m.queue.push(F.normalize(torch.randn(6, 16), dim=1),
                     torch.full((6,), c, dtype=torch.long))
    m.queue.push(F.normalize(torch.randn(2, 16), dim=1),
                 torch.tensor([1, 1]))
    m.memory_updated_cls[[0, 1, 2]] = True  # 4 is full but clean
    means_before = m.prototype_means.data.clone()
    w_before = m.last_layer.weight.data.clone()
    m.update_GMM()
    assert not m.memory_updated_cls.any()  # all visited marked clean
    # dirty+full classes (0, 2) moved; clean/partial (1, 3, 4) untouched
    assert not torch.allclose(m.prototype_means.data[0], means_before[0])
    assert not torch.allclose(m.prototype_means.data[2], means_before[2])
    for c in [1, 3, 4]:
        assert torch.allclose(m.prototype_means.data[c], means_before[c])
        assert torch.allclose(m.last_layer.weight.data[c], w_before[c])
    neg = 1 - m.prototype_class_identity.t()
    assert float(m.last_layer.weight.data[neg == 1].abs().sum()) == 0.0
    # priors stay positive and roughly normalized (pi momentum of a simplex)
    own = m.last_layer.weight.data[m.prototype_class_identity.t() == 1]
    assert (own >= 0).all()


def test_prune_keeps_topM():
    torch.manual_seed(0)
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(20, 16, 1, 1), num_classes=5,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=6, mine_K=3)
    # perturb priors so topk is non-trivial
    w = m.last_layer.weight.data
    pos = m.prototype_class_identity.t()
    w += 0.01 * torch.rand_like(w) * pos
    m.prune_prototypes_topM(top_M=2)
    assert m.prototypes_to_keep.sum().item() == 5 * 2
    kept = m.last_layer.weight.data != 0
    assert kept.sum() <= 5 * 2


def test_checkpoint_roundtrip(tmp_path):
    torch.manual_seed(0)
    kw = dict(pretrained=False, img_size=64, prototype_shape=(20, 16, 1, 1),
              num_classes=5, add_on_layers_type='regular', sz_embedding=8,
              mem_capacity=6, mine_K=3)
    m1 = construct_MGProto('resnet18', **kw)
    m1(torch.randn(2, 3, 64, 64), torch.tensor([0, 3]))  # populate queue
    path = tmp_path / 'ckpt.pth'
    torch.save(m1.state_dict(), path)

    sd = torch.load(path, weights_only=False)
    # reference checkpoint layout present: queue.cls{i} keys
    assert 'queue.cls0' in sd and 'queue.mem_len' in sd
    assert 'prototype_means' in sd and 'prototype_covs' in sd
    assert 'last_layer.weight' in sd and 'iteration_counter' in sd

    torch.manual_seed(1)
    m2 = construct_MGProto('resnet18', **kw)
    m2.load_state_dict(sd)
    x = torch.randn(2, 3, 64, 64)
    with torch.no_grad():
        o1, e1 = m1(x, None)
        o2, e2 = m2(x, None)
    assert torch.allclose(o1, o2, atol=1e-6)
    assert torch.allclose(e1, e2, atol=1e-6)


def test_mining_grad_flows_to_backbone(small_model):
    m = small_model
    m.train()
    x = torch.randn(2, 3, 64, 64)
    gt = torch.tensor([0, 1])
    m.zero_grad(set_to_none=True)
    logits, emb = m(x, gt)
    loss = F.cross_entropy(logits[:, :, 0], gt)
    loss.backward()
    conv1_grad = m.features.conv1.weight.grad
    assert conv1_grad is not None and conv1_grad.abs().sum() > 0
    # prototype means get no grad from the CE path (detached, model.py:264)
    assert m.prototype_means.grad is None or m.prototype_means.grad.abs().sum() == 0


@pytest.mark.parametrize('C,K,d,T', [
    (5, 3, 16, 4),    # baseline shape
    (2, 1, 8, 1),     # minimal: one prototype/class, top-1
    (3, 4, 24, 12),   # T close to HW corner (HW=16 at img 64)
    (7, 2, 32, 3),    # odd class count
])
def test_forward_matches_manual_pipeline(C, K, d, T):
    """Full-forward differential oracle: model.forward(x, None) must equal
    the manually composed pipeline (direct log-prob formula -> exp -> top-T
    -> pi-weighted mixture -> log), i.e. the reference's semantics
    (model.py:208-254) assembled from first principles."""
    from mgproto_amd.ops import reference as R
    torch.manual_seed(0)
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(C * K, d, 1, 1), num_classes=C,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=4, mine_K=T)
    m.eval()
    x = torch.randn(2, 3, 64, 64)
    with torch.no_grad():
        got, _ = m(x, None)

        # manual pipeline from push_forward pieces + direct formula
        base, _ = m.conv_features(x)
        base = torch.nn.functional.normalize(base, dim=1)
        B, dd, H, W = base.shape
        feat = base.permute(0, 2, 3, 1).reshape(-1, dd)
        lp = R.gmm_logprob_direct(feat, m.prototype_means.data,
                                  m.prototype_covs.data)      # [N, P]
        probs = lp.exp().view(B, H * W, C * K)
        vals, _ = torch.topk(probs, T, dim=1)                 # [B, T, P]
        vals = vals.permute(0, 2, 1).view(B, C, K, T)
        pi = m.last_layer.weight.data.view(C, C, K)[torch.arange(C),
                                                    torch.arange(C)]
        want = torch.log(torch.einsum('bckt,ck->bct', vals, pi))
    assert torch.allclose(got, want, atol=1e-4, rtol=1e-4), \
        (got - want).abs().max().item()


def test_upsample_conv_commutation():
    """conv1x1(upsample(x)) == upsample(conv1x1(x)) exactly (the
    regular_upsample add-on runs the upsample LAST for 32x less traffic;
    this pins the mathematical equivalence)."""
    torch.manual_seed(0)
    Ci, d = 32, 8
    c1 = torch.nn.Conv2d(Ci, d, 1)
    c2 = torch.nn.Conv2d(d, d, 1)
    up = torch.nn.Upsample(scale_factor=2, mode='bilinear', align_corners=False)
    x = torch.randn(2, Ci, 7, 7)
    with torch.no_grad():
        ref = c2(c1(up(x)))          # the reference's commented-out order
        ours = up(c2(c1(x)))
    assert torch.allclose(ref, ours, atol=1e-5), \
        (ref - ours).abs().max().item()


def test_regular_upsample_grid():
    torch.manual_seed(0)
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(10, 16, 1, 1), num_classes=5,
                          add_on_layers_type='regular_upsample',
                          sz_embedding=8, mem_capacity=4, mine_K=2)
    with torch.no_grad():
        bf, dist = m.push_forward(torch.randn(1, 3, 64, 64))
    # 64/16 = 4 latent, x2 upsample -> 8x8 grid
    assert dist.shape[-2:] == (8, 8)


def test_update_gmm_matches_per_class_loop():
    """Batched masked update_GMM == an independent per-class EM loop using
    the (separately oracle-tested) e/m-step ops + manual Adam + pi
    momentum. Covers the masking/momentum/state bookkeeping composition:
    only dirty+full classes move, and exactly as the naive loop says."""
    import math

    from mgproto_amd.ops import reference as R

    torch.manual_seed(3)
    C, K, d, cap = 4, 3, 8, 6
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1),
                              num_classes=C, add_on_layers_type='regular',
                              sz_embedding=8, mem_capacity=cap, mine_K=2)
    # classes 0 and 2: full + dirty; class 1: full but clean; class 3:
    # dirty but not full -> only 0 and 2 may move
    for c in (0, 1, 2):
        model.queue.push(F.normalize(torch.randn(cap, d), dim=1),
                         torch.full((cap,), c, dtype=torch.long))
    model.queue.push(F.normalize(torch.randn(2, d), dim=1),
                     torch.full((2,), 3, dtype=torch.long))
    model.memory_updated_cls[[0, 2, 3]] = True

    w0 = model.last_layer.weight.data.clone()
    means0 = model.prototype_means.data.clone()
    covs = model.prototype_covs.data.clone()
    bank = model.queue.mem.clone()

    model.update_GMM()

    diag = torch.arange(C)
    pi_after = model.last_layer.weight.data.view(C, C, K)[diag, diag]
    b1, b2 = model.adam_betas

    for c in range(C):
        if c not in (0, 2):     # clean or not-full classes: untouched
            assert torch.equal(model.prototype_means.data[c], means0[c])
            assert torch.equal(pi_after[c], w0.view(C, C, K)[c, c])
            continue
        x = bank[c:c + 1]
        mu = means0[c:c + 1].clone()
        pi_old = w0.view(C, C, K)[c, c].view(1, K).clone()
        m = torch.zeros_like(mu)
        v = torch.zeros_like(mu)
        for step in range(1, model.num_em_loop + 1):
            wlp, log_resp = R.em_e_step(x, mu, covs[c:c + 1], pi_old)
            grad, pi_unnorm = R.em_m_step_grads(
                x, log_resp, wlp, mu, covs[c:c + 1],
                alpha=model.alpha, lamda=model.lamda)
            m = b1 * m + (1 - b1) * grad
            v = b2 * v + (1 - b2) * grad * grad
            mu = mu - model.prototype_lr * (m / (1 - b1 ** step)) \
                / ((v / (1 - b2 ** step)).sqrt() + model.adam_eps)
            pi = pi_unnorm / cap
            pi_old = model.tau * pi_old + (1 - model.tau) * pi
        assert torch.allclose(model.prototype_means.data[c], mu[0],
                              atol=1e-5), c
        assert torch.allclose(pi_after[c], pi_old[0], atol=1e-6), c

    # dirty flags consumed
    assert not model.memory_updated_cls.any()


def test_update_gmm_single_component_class():
    """K=1 per class: the diversity term has no pairs — EM must stay
    finite (the reference's 0/0 diversity division would NaN the means)."""
    C, K, d, cap = 3, 1, 8, 4
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1),
                              num_classes=C, add_on_layers_type='regular',
                              sz_embedding=8, mem_capacity=cap, mine_K=1)
    g = torch.Generator().manual_seed(0)
    for c in range(C):
        model.queue.push(F.normalize(torch.randn(cap, d, generator=g), dim=1),
                         torch.full((cap,), c, dtype=torch.long))
        model.memory_updated_cls[c] = True
    before = model.prototype_means.data.clone()
    model.update_GMM()
    after = model.prototype_means.data
    assert torch.isfinite(after).all()
    assert not torch.equal(after, before)          # NLL grad still moves them
    assert torch.isfinite(model.last_layer.weight).all()


def test_mine_pool_larger_than_grid():
    """mine_T > HW clamps to the grid size instead of erroring (the
    reference's topk(T) would throw on small latent grids)."""
    C, K, d = 3, 2, 8
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1),
                              num_classes=C, add_on_layers_type='regular',
                              sz_embedding=8, mem_capacity=4, mine_K=64)
    x = torch.randn(2, 3, 64, 64)
    out, _ = model(x, torch.tensor([0, 1]))
    assert out.shape == (2, C, 16)      # 4x4 grid -> 16 mining levels
    assert torch.isfinite(out).all()


def test_gradient_flow_partition():
    """Exactly the gradient-trained parameter sets get grads: backbone,
    add-on, embedding. Prototype means/covs and the mixture head are
    EM-state (reference model.py:64,264-265) and must stay grad-free."""
    torch.manual_seed(0)
    C, K, d = 3, 2, 8
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(C * K, d, 1, 1), num_classes=C,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=4, mine_K=2)
    m.train()
    x = torch.randn(2, 3, 64, 64)
    y = torch.tensor([0, 1])
    out, aux = m(x, y)
    (F.cross_entropy(out[:, :, 0], y) + aux.square().mean()).backward()

    def has_grad(params):
        return any(p.grad is not None and p.grad.abs().sum() > 0
                   for p in params)

    assert has_grad(m.features.parameters())
    assert has_grad(m.add_on_layers.parameters())
    assert has_grad(m.embedding.parameters())
    assert m.prototype_means.grad is None
    assert m.prototype_covs.grad is None
    assert m.last_layer.weight.grad is None
    assert m.iteration_counter.grad is None
