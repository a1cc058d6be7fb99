"""End-to-end CPU plumbing test — BASELINE.json config 1:
ResNet-34 backbone, 10-class synthetic 224x224, CPU-only forward + push.

Uses a reduced image size to keep runtime sane; the full-size path is
exercised on GPU (test_gpu_e2e.py).
"""

import os

import torch
from torch.utils.data import DataLoader

from mgproto_amd.model import construct_MGProto
from mgproto_amd.data import SyntheticImages
from mgproto_amd.engine import push_prototypes, warm_only, joint
from mgproto_amd.engine import train as tnt_train
from mgproto_amd.engine import test as tnt_test
from mgproto_amd.losses import build_aux_loss


def _collate(batch):
    imgs = torch.stack([b[0] for b in batch])
    labels = torch.tensor([b[1] for b in batch])
    idx = torch.tensor([b[2] for b in batch])
    return imgs, labels, idx


def test_train_test_push_cycle(tmp_path):
    torch.manual_seed(0)
    C, K, d = 10, 3, 32
    model = construct_MGProto('resnet34', pretrained=False, img_size=96,
                              prototype_shape=(C * K, d, 1, 1), num_classes=C,
                              add_on_layers_type='regular', sz_embedding=16,
                              mem_capacity=8, mine_K=4)
    aux = build_aux_loss('Proxy_Anchor', nb_classes=C, sz_embed=16)
    opt = torch.optim.Adam([
        {'params': model.features.parameters(), 'lr': 1e-4},
        {'params': model.add_on_layers.parameters(), 'lr': 3e-3},
        {'params': aux.parameters(), 'lr': 1e-2},
    ])

    ds = SyntheticImages(n=24, num_classes=C, img_size=96)
    loader = DataLoader(ds, batch_size=8, collate_fn=_collate)

    joint(model)
    coefs = {'crs_ent': 1, 'mine': 0.2, 'aux': 0.5}
    acc, results = tnt_train(model, loader, opt, aux_criterion=aux, use_mine=True,
                         update_GMM=False, coefs=coefs, log=lambda *a: None,
                         amp_dtype='off', print_every=0)
    assert 0.0 <= acc <= 1.0
    assert results['cross_entropy'] > 0

    # fill memory then run an EM epoch
    for c in range(C):
        model.queue.push(torch.nn.functional.normalize(torch.randn(8, d), dim=1),
                         torch.full((8,), c, dtype=torch.long))
        model.memory_updated_cls[c] = True
    acc, _ = tnt_train(model, loader, opt, aux_criterion=aux, use_mine=True,
                   update_GMM=True, coefs=coefs, log=lambda *a: None,
                   amp_dtype='off', print_every=0)

    # eval
    acc, results = tnt_test(model, (loader,), log=lambda *a: None, amp_dtype='off')
    assert 0.0 <= acc <= 1.0

    # push (with artifact rendering into tmp_path)
    push_ds = SyntheticImages(n=24, num_classes=C, img_size=96, normalize=False)
    push_loader = DataLoader(push_ds, batch_size=8, collate_fn=_collate)
    means_before = model.prototype_means.data.clone()
    chosen = push_prototypes(push_loader, model,
                             root_dir_for_saving_prototypes=str(tmp_path),
                             epoch_number=0, log=lambda *a: None)
    assert len(chosen) > 0
    assert not torch.allclose(model.prototype_means.data, means_before)
    # every pushed mean equals an actual (normalized) patch feature: unit norm
    P = model.num_prototypes
    pushed_j = torch.tensor([c[0] for c in chosen])
    norms = model.prototype_means.data.view(P, -1)[pushed_j].norm(dim=1)
    assert torch.allclose(norms, torch.ones_like(norms), atol=1e-4)
    # each image claimed at most once
    imgs = [c[1] for c in chosen]
    assert len(imgs) == len(set(imgs))

    # OoD scoring path
    from mgproto_amd.engine import _testing_with_OoD
    ood_ds = SyntheticImages(n=16, num_classes=C, img_size=96, seed=7)
    ood_loader = DataLoader(ood_ds, batch_size=8, collate_fn=_collate)
    acc, results = _testing_with_OoD(model, (loader, ood_loader),
                                     log=lambda *a: None, amp_dtype='off')
    assert 'FPR95_1' in results

    # prune
    model.prune_prototypes_topM(top_M=2)
    acc, _ = tnt_test(model, (loader,), log=lambda *a: None, amp_dtype='off')


def test_push_deterministic_reforward():
    """Pass-2 re-forward of a chosen image must reproduce pass-1 features
    (requires a deterministic dataset + eval-mode model)."""
    torch.manual_seed(0)
    C, K, d = 4, 2, 16
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1), num_classes=C,
                              add_on_layers_type='regular', sz_embedding=8,
                              mem_capacity=4, mine_K=2)
    model.eval()
    ds = SyntheticImages(n=8, num_classes=C, img_size=64, normalize=False)
    loader = DataLoader(ds, batch_size=4, collate_fn=_collate)
    chosen = push_prototypes(loader, model, log=lambda *a: None)
    # re-check: forwarding the chosen image again gives the stored mean
    for (j, img_idx, h, w) in chosen[:3]:
        img = ds[img_idx][0].unsqueeze(0)
        with torch.no_grad():
            feats, _ = model.push_forward(img)
        want = feats[0, :, h, w]
        got = model.prototype_means.data.view(model.num_prototypes, -1)[j]
        assert torch.allclose(got, want, atol=1e-6)


def test_density_auroc():
    from mgproto_amd.engine.trainer import _density_auroc
    ident = torch.tensor([5.0, 4.0, 6.0, 5.5])
    ood = torch.tensor([1.0, 0.5, 2.0])
    assert _density_auroc(ident, ood) == 1.0        # perfectly separated
    assert _density_auroc(ident, ident) == 0.5 or \
        abs(_density_auroc(ident, ident) - 0.5) < 0.2
    assert _density_auroc(torch.zeros(0), ood) is None


def test_push_renders_artifacts(tmp_path):
    """Push writes the reference's three JPEG artifacts per pushed
    prototype (original+bbox, heatmap overlay, cropped patch —
    reference push.py:203-226)."""
    torch.manual_seed(1)
    C, K, d = 4, 2, 16
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1), num_classes=C,
                              add_on_layers_type='regular', sz_embedding=8,
                              mem_capacity=8, mine_K=2)
    ds = SyntheticImages(n=12, num_classes=C, img_size=64, normalize=False)
    loader = DataLoader(ds, batch_size=4, collate_fn=_collate)
    chosen = push_prototypes(loader, model,
                             root_dir_for_saving_prototypes=str(tmp_path),
                             epoch_number=3, log=lambda *a: None)
    assert chosen
    out = tmp_path / 'epoch-3'
    files = sorted(os.listdir(out))
    j = chosen[0][0]
    for suffix in ('-original.jpg', '-original_with_self_act.jpg', '.jpg'):
        assert f'{j}prototype-img{suffix}' in files, (suffix, files[:6])
    # artifacts decode as images
    from PIL import Image
    with Image.open(out / f'{j}prototype-img-original.jpg') as im:
        assert im.size == (64, 64)
