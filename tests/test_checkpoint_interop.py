"""Checkpoint layout interop: the state dict must contain EXACTLY the key
families a reference-trained checkpoint has (SURVEY.md §5 checkpoint
contract), so checkpoints flow both ways without key surgery."""

import re

import torch

from mgproto_amd.model import construct_MGProto


def _model():
    torch.manual_seed(0)
    return construct_MGProto('resnet34', pretrained=False, img_size=64,
                             prototype_shape=(20, 16, 1, 1), num_classes=5,
                             add_on_layers_type='regular', sz_embedding=8,
                             mem_capacity=4, mine_K=2)


REFERENCE_KEY_PATTERNS = [
    # backbone (torchvision ResNet naming; reference models/resnet_features.py)
    r'^features\.conv1\.weight$',
    r'^features\.bn1\.(weight|bias|running_mean|running_var|num_batches_tracked)$',
    r'^features\.layer[1-4]\.\d+\.(conv[123]|bn[123])\.'
    r'(weight|bias|running_mean|running_var|num_batches_tracked)$',
    r'^features\.layer[1-4]\.\d+\.downsample\.[01]\.'
    r'(weight|bias|running_mean|running_var|num_batches_tracked)$',
    # add-on 1x1 convs (reference model.py:117-143)
    r'^add_on_layers\.\d+\.(weight|bias)$',
    # aux embedding head (reference model.py:146)
    r'^embedding\.(weight|bias)$',
    # prototype state (reference model.py:148-154)
    r'^prototype_means$',
    r'^prototype_covs$',
    r'^last_layer\.weight$',
    # memory bank (reference utils/memory.py:19-20)
    r'^queue\.cls\d+$',
    r'^queue\.mem_len$',
    # counter (reference model.py:168)
    r'^iteration_counter$',
]


def test_state_dict_keys_match_reference_families():
    sd = _model().state_dict()
    pats = [re.compile(p) for p in REFERENCE_KEY_PATTERNS]
    unexpected = [k for k in sd if not any(p.match(k) for p in pats)]
    assert not unexpected, f'keys a reference checkpoint lacks: {unexpected}'
    # every queue.cls{i} present
    for c in range(5):
        assert f'queue.cls{c}' in sd
    # the prototype-state tensors have the reference shapes
    assert sd['prototype_means'].shape == (5, 4, 16)   # [C, K, d]
    assert sd['prototype_covs'].shape == (5, 4, 16)
    assert sd['last_layer.weight'].shape == (5, 20)    # [C, P]
    assert sd['queue.cls0'].shape == (4, 16)           # [cap, d]
    assert sd['queue.mem_len'].shape == (5,)
    assert sd['iteration_counter'].shape == (1,)


def test_strict_load_of_reference_shaped_dict():
    """A dict with exactly the reference key families loads strict=True."""
    m1 = _model()
    sd = {k: v.clone() for k, v in m1.state_dict().items()}
    torch.manual_seed(9)
    m2 = _model()
    missing, unexpected = m2.load_state_dict(sd, strict=True), None
    x = torch.randn(2, 3, 64, 64)
    with torch.no_grad():
        o1, _ = m1(x, None)
        o2, _ = m2(x, None)
    assert torch.allclose(o1, o2, atol=1e-6)


def test_nonneg_linear_prune_arg_and_score():
    m = _model()
    # NonNegLinear with the pruning mask argument (reference model.py:71)
    m.last_layer.debug_asserts = True
    keep = torch.ones_like(m.last_layer.weight)
    out = m.last_layer(torch.rand(3, 20), prototypes_to_keep_with_negative=keep)
    assert out.shape == (3, 5)
    # _score: per-class mixture log-likelihood (reference model.py:403-421)
    x = torch.randn(10, 16)
    s = m._score(x, m.prototype_means[0].unsqueeze(0),
                 m.prototype_covs[0].unsqueeze(0),
                 torch.full((1, 4, 1), 0.25))
    assert torch.isfinite(s)
    per = m._score(x, m.prototype_means[0].unsqueeze(0),
                   m.prototype_covs[0].unsqueeze(0),
                   torch.full((1, 4, 1), 0.25), as_average=False)
    assert per.shape == (10,)


def test_rng_resume(tmp_path):
    """python/numpy/torch RNG streams continue identically after resume."""
    import random
    import numpy as np
    import torch
    from mgproto_amd.utils.checkpoint import (save_train_state,
                                              load_train_state)

    m = torch.nn.Linear(2, 2)
    random.seed(7); np.random.seed(7); torch.manual_seed(7)
    random.random(); np.random.random(); torch.rand(1)
    save_train_state(str(tmp_path / 'st.pt'), m, {}, {}, epoch=3)
    expect = (random.random(), float(np.random.random()),
              float(torch.rand(1)))

    random.seed(99); np.random.seed(99); torch.manual_seed(99)
    st = load_train_state(str(tmp_path / 'st.pt'), m)
    got = (random.random(), float(np.random.random()), float(torch.rand(1)))
    assert got == expect and st['epoch'] == 3


def test_inspect_checkpoint_tool(tmp_path):
    import os
    import subprocess
    import sys

    from mgproto_amd.model import construct_MGProto
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(6, 16, 1, 1), num_classes=3,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=4, mine_K=2)
    p = tmp_path / 'm.pth'
    torch.save(m.state_dict(), str(p))
    r = subprocess.run([sys.executable, 'tools/inspect_checkpoint.py',
                        str(p)], capture_output=True, text=True,
                       timeout=300, cwd=root)
    assert r.returncode == 0, r.stderr
    assert 'C=3 classes x K=2' in r.stdout
    assert 'memory bank: 3 classes x cap 4' in r.stdout
