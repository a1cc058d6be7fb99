"""hipGraph-captured production training step == eager.

Two identically-seeded models train for one epoch on identical synthetic
batches — one through engine/graphstep.py's captured whole-step graph
(the path train.py now runs), one eager — and must produce matching loss
statistics and weights. Run on MI355X (@gpu)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip('needs a GPU', allow_module_level=True)


def _make_model(device):
    from mgproto_amd.model import construct_MGProto
    torch.manual_seed(0)
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(20, 16, 1, 1), num_classes=5,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=64, mine_K=4).to(device)
    m.features = m.features.to(memory_format=torch.channels_last)
    return m


def _batches(n, bs, device):
    g = torch.Generator().manual_seed(7)
    return [(torch.randn(bs, 3, 64, 64, generator=g).to(device),
             torch.randint(0, 5, (bs,), generator=g).to(device))
            for _ in range(n)]


def _prefill(model, device):
    import torch.nn.functional as F
    with torch.no_grad():
        mem = F.normalize(torch.randn(5 * 64, 16, device=device,
                                      generator=torch.Generator(
                                          device=device).manual_seed(3)),
                          dim=1)
        labels = torch.arange(5, device=device).repeat_interleave(64)
        model.queue.push(mem, labels)
        model.memory_updated_cls[:] = True


def _run(device, use_graph, n_batches=8, bs=8):
    import torch.nn.functional as F
    from mgproto_amd.engine.graphstep import GraphedStep
    from mgproto_amd.losses import build_aux_loss

    model = _make_model(device)
    _prefill(model, device)
    torch.manual_seed(11)
    aux = build_aux_loss('Proxy_Anchor', nb_classes=5, sz_embed=8).to(device)
    opt = torch.optim.Adam(
        [{'params': model.add_on_layers.parameters(), 'lr': 1e-3},
         {'params': model.features.parameters(), 'lr': 1e-4},
         {'params': aux.parameters(), 'lr': 1e-3}],
        capturable=use_graph)
    coefs = {'crs_ent': 1.0, 'mine': 0.2, 'aux': 0.5}
    gs = (GraphedStep(model, aux, coefs, device, bs, 64, amp_dtype='bf16')
          if use_graph else None)

    model.train()
    losses = []
    for image, target in _batches(n_batches, bs, device):
        if gs is not None and gs.matches(image, opt, True, True):
            st = gs.step(image, target, opt, em_active=True, use_mine=True)
            losses.append(float(st['ce']))
        else:
            with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
                output, x_aux = model(image, target)
            output = output.float()
            mine = sum(F.cross_entropy(output[:, :, k], target)
                       for k in range(1, output.shape[2])) / (output.shape[2] - 1)
            ce = F.cross_entropy(output[:, :, 0], target)
            al = aux(x_aux.float(), target)
            loss = coefs['crs_ent'] * ce + coefs['mine'] * mine + coefs['aux'] * al
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
            model.update_GMM()
            losses.append(float(ce))
    torch.cuda.synchronize()
    if gs is not None:
        assert gs.graph is not None, 'capture must succeed on GPU'
    return losses, model


def test_graphed_step_matches_eager():
    device = torch.device('cuda', 0)
    l_eager, m_eager = _run(device, use_graph=False)
    l_graph, m_graph = _run(device, use_graph=True)

    # per-step CE trajectories agree (bf16 backbone -> loose elementwise tol;
    # run-to-run bf16 noise compounds through Adam+EM over the 8 steps, so
    # these bounds are calibrated against same-program eager-vs-eager reruns)
    for i, (a, b) in enumerate(zip(l_eager, l_graph)):
        assert abs(a - b) < 0.15 + 0.05 * abs(a), (i, a, b)
    # end-of-epoch weights agree
    for (n, p), (_, q) in zip(m_eager.named_parameters(),
                              m_graph.named_parameters()):
        assert torch.allclose(p, q, atol=5e-2, rtol=5e-2), \
            (n, (p - q).abs().max().item())
    # prototype means (EM-updated inside the graph) agree
    assert torch.allclose(m_eager.prototype_means, m_graph.prototype_means,
                          atol=5e-2), \
        (m_eager.prototype_means - m_graph.prototype_means).abs().max().item()


def test_graphed_step_recaptures_on_phase_flip():
    """warm->joint flips requires_grad; the graph must re-capture, not
    replay a stale program."""
    device = torch.device('cuda', 0)
    import torch.nn.functional as F  # noqa: F401
    from mgproto_amd.engine.graphstep import GraphedStep
    from mgproto_amd.losses import build_aux_loss

    model = _make_model(device)
    _prefill(model, device)
    aux = build_aux_loss('Proxy_Anchor', nb_classes=5, sz_embed=8).to(device)
    opt = torch.optim.Adam(
        [{'params': model.add_on_layers.parameters(), 'lr': 1e-3},
         {'params': model.features.parameters(), 'lr': 1e-4},
         {'params': aux.parameters(), 'lr': 1e-3}], capturable=True)
    coefs = {'crs_ent': 1.0, 'mine': 0.2, 'aux': 0.5}
    gs = GraphedStep(model, aux, coefs, device, 8, 64, amp_dtype='bf16',
                     warmup_steps=1)
    batches = _batches(8, 8, device)

    # warm phase: backbone frozen
    for p in model.features.parameters():
        p.requires_grad = False
    for image, target in batches[:3]:
        assert gs.matches(image, opt, True, True)
        gs.step(image, target, opt, em_active=True, use_mine=True)
    torch.cuda.synchronize()
    g_warm = gs.graph
    assert g_warm is not None
    feat_before = [p.detach().clone() for p in model.features.parameters()]

    # joint phase: unfreeze -> signature flip -> recapture
    for p in model.features.parameters():
        p.requires_grad = True
    for image, target in batches[3:]:
        assert gs.matches(image, opt, True, True)
        gs.step(image, target, opt, em_active=True, use_mine=True)
    torch.cuda.synchronize()
    assert gs.graph is not None and gs.graph is not g_warm
    # backbone actually trained in the joint phase
    changed = any(not torch.equal(p.detach(), b)
                  for p, b in zip(model.features.parameters(), feat_before))
    assert changed
