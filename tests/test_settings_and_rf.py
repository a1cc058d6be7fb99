"""Settings config surface + receptive-field math + losses coverage."""

import math

import pytest
import torch

from mgproto_amd import settings
from mgproto_amd.utils.receptive_field import (compute_layer_rf_info,
                                               compute_proto_layer_rf_info_v2,
                                               compute_rf_prototype)


def test_settings_surface_matches_reference():
    # module constants the reference exposes (settings.py:1-52)
    assert settings.img_size == 224
    assert settings.num_classes == 200
    assert settings.prototype_shape == (2000, 64, 1, 1)
    assert settings.prototype_activation_function == 'log'
    assert settings.add_on_layers_type == 'regular'
    assert settings.train_batch_size == 80
    assert settings.joint_optimizer_lrs == {'features': 1e-4,
                                            'add_on_layers': 3e-3,
                                            'prototype_vectors': 3e-3}
    assert settings.warm_optimizer_lrs == {'add_on_layers': 3e-3,
                                           'prototype_vectors': 3e-3}
    assert settings.last_layer_optimizer_lr == 1e-4
    assert settings.coefs == {'crs_ent': 1, 'mine': 0.2, 'aux': 0.5}
    assert settings.num_train_epochs == 120
    assert settings.num_warm_epochs == 0
    assert settings.mine_start == 40
    assert settings.updateGMM_start == 35
    assert settings.push_start == 100
    assert settings.push_epochs == [i for i in range(120) if i % 10 == 0]
    for name in ('train_dir', 'test_dir', 'train_push_dir', 'test_dir_ood1',
                 'test_dir_ood2'):
        assert isinstance(getattr(settings, name), str)


def test_settings_dataclass_decay_tables():
    cfg = settings.Settings(base_architecture='resnet34')
    assert cfg.lr_decay_epochs() == [30, 45, 60, 75, 90]
    cfg = settings.Settings(base_architecture='resnet50')
    assert cfg.lr_decay_epochs() == [10, 15, 20, 25, 30]
    d = cfg.to_dict()
    assert d['mem_capacity'] == 800 and d['mine_K'] == 20


def test_rf_single_conv():
    # 3x3 stride-1 pad-1 conv on 224: n stays, rf grows to 3
    out = compute_layer_rf_info(3, 1, 1, [224, 1, 1, 0.5])
    assert out == [224, 1, 3, 0.5]
    # stride-2 7x7 pad-3 stem
    out = compute_layer_rf_info(7, 2, 3, [224, 1, 1, 0.5])
    assert out[0] == 112 and out[1] == 2 and out[2] == 7


def test_rf_matches_backbone_grid():
    """The RF latent grid size equals the actual backbone output size
    (this is the conv_info fix — the reference's phantom max-pool made
    n half the true grid)."""
    from mgproto_amd.models import resnet34_features
    f = resnet34_features()
    ks, ss, ps = f.conv_info()
    info = compute_proto_layer_rf_info_v2(224, ks, ss, ps, 1)
    with torch.no_grad():
        out = f(torch.randn(1, 3, 224, 224))
    assert info[0] == out.shape[-1] == 14
    # a valid latent index maps to a bbox inside the image
    box = compute_rf_prototype(224, [0, 13, 13], info)
    assert 0 <= box[1] <= box[2] <= 224 and 0 <= box[3] <= box[4] <= 224


@pytest.mark.parametrize('name', ['Proxy_Anchor', 'Proxy_NCA', 'MS',
                                  'Contrastive', 'Triplet', 'NPair'])
def test_aux_losses_forward_backward(name):
    from mgproto_amd.losses import build_aux_loss
    torch.manual_seed(0)
    crit = build_aux_loss(name, nb_classes=5, sz_embed=8)
    x = torch.randn(12, 8, requires_grad=True)
    t = torch.tensor([0, 0, 1, 1, 2, 2, 3, 3, 4, 4, 0, 1])
    loss = crit(x, t)
    assert torch.isfinite(loss)
    if loss.requires_grad:
        loss.backward()
        assert torch.isfinite(x.grad).all()


def test_unknown_aux_loss_raises():
    from mgproto_amd.losses import build_aux_loss
    with pytest.raises(ValueError):
        build_aux_loss('nope', nb_classes=2, sz_embed=4)


def test_warm_joint_freezing():
    from mgproto_amd.engine import warm_only, joint
    from mgproto_amd.model import construct_MGProto
    import torch as _t
    _t.manual_seed(0)
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(10, 16, 1, 1), num_classes=5,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=4, mine_K=2)
    warm_only(m, log=lambda *a: None)
    assert all(not p.requires_grad for p in m.features.parameters())
    assert all(p.requires_grad for p in m.add_on_layers.parameters())
    joint(m, log=lambda *a: None)
    assert all(p.requires_grad for p in m.features.parameters())


def test_loggers(tmp_path):
    from mgproto_amd.utils import create_logger, MetricsLogger
    log, close = create_logger(str(tmp_path / 'x.log'))
    for i in range(12):
        log(f'line {i}')
    close()
    assert len(open(tmp_path / 'x.log').readlines()) == 12

    ml = MetricsLogger(str(tmp_path / 'm.jsonl'), rank=0)
    ml.log({'a': 1.0, 'b': 'txt'})
    ml.log({'a': 2.0}, step=7)
    ml.close()
    import json
    recs = [json.loads(l) for l in open(tmp_path / 'm.jsonl')]
    assert recs[0]['a'] == 1.0 and recs[0]['_step'] == 0
    assert recs[1]['_step'] == 7
    # rank>0 writes nothing
    ml2 = MetricsLogger(str(tmp_path / 'n.jsonl'), rank=1)
    ml2.log({'a': 1})
    ml2.close()
    import os as _os
    assert not _os.path.exists(tmp_path / 'n.jsonl')


def test_save_model_w_condition(tmp_path):
    from mgproto_amd.utils.checkpoint import save_model_w_condition
    import torch as _t

    class Tiny(_t.nn.Module):
        def __init__(self):
            super().__init__()
            self.w = _t.nn.Parameter(_t.zeros(2))

    m = Tiny()
    save_model_w_condition(m, str(tmp_path), '10nopush', accu=0.8224,
                           target_accu=0.6, log=lambda *a: None)
    assert (tmp_path / '10nopush0.8224.pth').is_file()   # reference name fmt
    save_model_w_condition(m, str(tmp_path), '11nopush', accu=0.5,
                           target_accu=0.6, log=lambda *a: None)
    assert not (tmp_path / '11nopush0.5000.pth').exists()


def test_training_without_aux_criterion():
    import torch as _t
    from torch.utils.data import DataLoader
    from mgproto_amd.engine import trainer as tnt
    from mgproto_amd.model import construct_MGProto
    from mgproto_amd.data import SyntheticImages
    _t.manual_seed(0)
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(10, 16, 1, 1), num_classes=5,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=4, mine_K=2)
    opt = _t.optim.Adam(m.parameters(), lr=1e-4)
    ds = SyntheticImages(8, 5, 64)
    loader = DataLoader(ds, batch_size=4,
                        collate_fn=lambda b: (_t.stack([x[0] for x in b]),
                                              _t.tensor([x[1] for x in b])))
    acc, res = tnt.train(m, loader, opt, aux_criterion=None, use_mine=True,
                         coefs={'crs_ent': 1, 'mine': 0.2, 'aux': 0.5},
                         log=lambda *a: None, amp_dtype='off', print_every=0)
    assert res['aux_loss'] == 0.0


def test_rf_prototypes_plural_and_helpers():
    from mgproto_amd.utils.receptive_field import (
        compute_proto_layer_rf_info_v2, compute_rf_prototype,
        compute_rf_prototypes)
    from mgproto_amd.utils.helpers import (list_of_distances, make_one_hot,
                                           find_high_activation_crop)
    import numpy as np

    info = compute_proto_layer_rf_info_v2(
        224, [7, 3, 3, 3], [2, 2, 2, 2], [3, 1, 1, 1], 1)
    singles = [compute_rf_prototype(224, [i, 3, 5], info) for i in range(2)]
    plural = compute_rf_prototypes(224, [[0, 3, 5], [1, 3, 5]], info)
    assert [list(p) for p in plural] == [list(s) for s in singles]

    # list_of_distances == pairwise squared L2
    X, Y = torch.randn(4, 3), torch.randn(5, 3)
    D = list_of_distances(X, Y)
    assert torch.allclose(D, torch.cdist(X, Y).pow(2), atol=1e-5)

    oh = torch.zeros(3, 4)
    make_one_hot(torch.tensor([1, 0, 3]), oh)
    assert oh.argmax(1).tolist() == [1, 0, 3] and oh.sum() == 3

    # >5% of pixels must exceed the 95th percentile for a tight crop
    # (reference helpers.py:38 thresholds on the percentile of ALL pixels)
    act = np.zeros((10, 10)); act[4:7, 6:8] = 5.0
    y1, y2, x1, x2 = find_high_activation_crop(act, percentile=95)
    assert (y1, y2, x1, x2) == (4, 7, 6, 8)


@pytest.mark.parametrize('arch', ['resnet18', 'resnet50', 'densenet121',
                                  'vgg11', 'vgg16_bn'])
def test_conv_info_predicts_actual_grid(arch):
    """conv_info() must describe the real forward: the RF recurrence's
    output size equals the actual feature-map size (the reference counts
    a skipped max-pool, reference resnet_features.py:199)."""
    import torch

    from mgproto_amd.model import base_architecture_to_features
    from mgproto_amd.utils.receptive_field import compute_proto_layer_rf_info_v2

    f = base_architecture_to_features[arch](pretrained=False)
    ks, ss, ps = f.conv_info()
    info = compute_proto_layer_rf_info_v2(96, ks, ss, ps, 1)
    with torch.no_grad():
        out = f(torch.randn(1, 3, 96, 96))
    assert info[0] == out.shape[-1], (info[0], out.shape)
