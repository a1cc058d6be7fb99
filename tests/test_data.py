"""Data pipeline: PIL transforms, ImageFolder, preprocess."""

import os

import numpy as np
import pytest
import torch

PIL = pytest.importorskip('PIL')
from PIL import Image

from mgproto_amd.data import transforms as T
from mgproto_amd.data.folder import ImageFolder, MyImageFolder
from mgproto_amd.data.preprocess import (preprocess_input_function,
                                         undo_preprocess_input_function)


def _img(w=64, h=48, seed=0):
    rng = np.random.RandomState(seed)
    return Image.fromarray(rng.randint(0, 255, (h, w, 3), dtype=np.uint8))


def test_to_tensor_normalize_roundtrip():
    img = _img()
    t = T.ToTensor()(img)
    assert t.shape == (3, 48, 64)
    assert 0.0 <= t.min() and t.max() <= 1.0
    x = t.unsqueeze(0)
    y = preprocess_input_function(x)
    back = undo_preprocess_input_function(y)
    assert torch.allclose(back, x, atol=1e-6)


def test_resize_semantics():
    img = _img(100, 50)
    r = T.Resize(25)(img)            # shorter side -> 25
    assert r.size == (50, 25)
    r2 = T.Resize((30, 40))(img)     # exact (h, w)
    assert r2.size == (40, 30)


def test_center_crop():
    img = _img(100, 80)
    c = T.CenterCrop(60)(img)
    assert c.size == (60, 60)


def test_random_transforms_run_and_preserve_size():
    img = _img(64, 64)
    tf = T.Compose([
        T.RandomPerspective(distortion_scale=0.2, p=1.0),
        T.ColorJitter((0.6, 1.4), (0.6, 1.4), (0.6, 1.4), (-0.02, 0.02)),
        T.RandomHorizontalFlip(),
        T.RandomAffine(degrees=25, shear=(-15, 15), translate=[0.05, 0.05]),
        T.RandomResizedCrop(size=(32, 32), scale=(0.60, 1.0)),
        T.ToTensor(),
    ])
    out = tf(img)
    assert out.shape == (3, 32, 32)
    assert torch.isfinite(out).all()


def test_image_folder(tmp_path):
    for ci, cls in enumerate(['001.sparrow', '002.wren']):
        d = tmp_path / cls
        d.mkdir()
        for i in range(3):
            _img(seed=ci * 10 + i).save(d / f'img_{i}.jpg')
    ds = ImageFolder(str(tmp_path), T.Compose([T.ToTensor()]))
    assert len(ds) == 6
    assert ds.classes == ['001.sparrow', '002.wren']
    img, label, idx = ds[4]
    assert label == 1 and idx == 4
    assert img.shape[0] == 3

    mds = MyImageFolder(str(tmp_path), T.Compose([T.ToTensor()]))
    (img, label), (path, plabel) = mds[0]
    assert label == plabel == 0
    assert path.endswith('.jpg')


def test_push_accepts_reference_format(tmp_path):
    """push_prototypes consumes the reference MyImageFolder item format."""
    from torch.utils.data import DataLoader
    from mgproto_amd.engine import push_prototypes
    from mgproto_amd.model import construct_MGProto

    for ci, cls in enumerate(['a', 'b']):
        d = tmp_path / cls
        d.mkdir()
        for i in range(2):
            _img(64, 64, seed=ci * 10 + i).save(d / f'{i}.jpg')
    ds = MyImageFolder(str(tmp_path),
                       T.Compose([T.Resize((64, 64)), T.ToTensor()]))

    def collate(batch):
        imgs = torch.stack([b[0][0] for b in batch])
        labels = torch.tensor([b[0][1] for b in batch])
        paths = [b[1][0] for b in batch]
        return (imgs, labels), (paths, labels)

    loader = DataLoader(ds, batch_size=2, collate_fn=collate)
    torch.manual_seed(0)
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(4, 16, 1, 1), num_classes=2,
                              add_on_layers_type='regular', sz_embedding=8,
                              mem_capacity=4, mine_K=2)
    model.eval()
    chosen = push_prototypes(loader, model, log=lambda *a: None,
                             preprocess_input_function=preprocess_input_function)
    assert len(chosen) > 0


def test_fused_train_transform():
    from mgproto_amd.data.preprocess import mean, std
    img = _img(120, 100, seed=3)
    tf = T.FusedTrainTransform(64, normalize=T.Normalize(mean, std))
    import random as _r
    _r.seed(0)
    outs = [tf(img) for _ in range(4)]
    for o in outs:
        assert o.shape == (3, 64, 64)
        assert torch.isfinite(o).all()
    # stochastic: different draws differ
    assert not torch.allclose(outs[0], outs[1])


def test_fused_transform_identity_params_matches_resize():
    """With all randomness disabled and a square image, the fused
    homography must reduce to a plain resize."""
    img = _img(80, 80, seed=5)
    tf = T.FusedTrainTransform(64, scale=(1.0, 1.0), perspective_p=0.0,
                               jitter=None, degrees=0, shear=(0, 0),
                               translate=(0, 0))
    import random as _r
    for seed in range(3):
        _r.seed(seed)
        out = tf(img)
        want = T.ToTensor()(T.Resize((64, 64))(img))
        # im.transform point-samples (no antialias) while PIL resize
        # filters, so downscales differ slightly; bound the gap
        assert (out - want).abs().mean() < 0.06, \
            float((out - want).abs().mean())


def test_fast_color_jitter_matches_pil_enhance():
    from PIL import ImageEnhance
    img = _img(32, 32, seed=7)
    import random as _r

    # brightness only
    fj = T.FastColorJitter(brightness=(1.3, 1.3))
    _r.seed(0)
    got = np.asarray(fj(img), dtype=np.float32)
    want = np.asarray(ImageEnhance.Brightness(img).enhance(1.3),
                      dtype=np.float32)
    assert np.abs(got - want).mean() < 1.0

    # saturation only
    fj = T.FastColorJitter(saturation=(0.5, 0.5))
    _r.seed(0)
    got = np.asarray(fj(img), dtype=np.float32)
    want = np.asarray(ImageEnhance.Color(img).enhance(0.5), dtype=np.float32)
    assert np.abs(got - want).mean() < 2.0


def test_native_fastaug_matches_python_path():
    """The C++ warp+jitter+normalize core vs the PIL/numpy path with the
    SAME sampled parameters (the native path keeps float precision where
    the python path quantizes to uint8 between stages — small tolerance)."""
    import random
    from mgproto_amd.data.preprocess import mean, std
    if T._fastaug() is None:
        pytest.skip('native fastaug not built')
    img = _img(120, 100, seed=11)
    tf = T.FusedTrainTransform(64, normalize=T.Normalize(mean, std))
    for seed in (0, 1, 2):
        random.seed(seed)
        a = tf(img)
        os.environ['MGPROTO_NO_FASTAUG'] = '1'
        T._FASTAUG[:] = [None, False]
        try:
            random.seed(seed)
            b = tf(img)
        finally:
            os.environ.pop('MGPROTO_NO_FASTAUG')
            T._FASTAUG[:] = [None, False]
        d = (a - b).abs()
        assert float(d.mean()) < 0.08, float(d.mean())


def test_native_fastaug_warp_only_tight():
    """No jitter: the two paths differ only by the python path's uint8
    round-trip — must agree to ~1 gray level."""
    import random
    if T._fastaug() is None:
        pytest.skip('native fastaug not built')
    img = _img(100, 100, seed=13)
    tf = T.FusedTrainTransform(64, jitter=None)
    random.seed(5)
    a = tf(img)
    os.environ['MGPROTO_NO_FASTAUG'] = '1'
    T._FASTAUG[:] = [None, False]
    try:
        random.seed(5)
        b = tf(img)
    finally:
        os.environ.pop('MGPROTO_NO_FASTAUG')
        T._FASTAUG[:] = [None, False]
    assert float((a - b).abs().mean()) < 0.01


def test_scaled_jpeg_decode(tmp_path):
    """decode_size triggers libjpeg 1/n decode for large JPEGs, and the
    fused transform still yields the right output shape."""
    from PIL import Image as PILImage
    d = tmp_path / 'c0'
    d.mkdir()
    big = PILImage.fromarray(
        np.random.default_rng(0).integers(0, 255, (900, 1200, 3),
                                          dtype=np.uint8))
    big.save(str(d / 'a.jpg'), quality=90)

    tf = T.FusedTrainTransform(224, normalize=T.Normalize([0.5] * 3,
                                                          [0.5] * 3))
    ds = ImageFolder(str(tmp_path), tf, decode_size=448)
    path, _ = ds.samples[0]
    img = ds.loader(path)
    # draft picks the smallest 1/n scale with both dims >= 448: 1/2 here
    assert min(img.size) >= 448 and max(img.size) < 1200
    out, label, idx = ds[0]
    assert out.shape == (3, 224, 224) and label == 0 and idx == 0

    # PNGs (no draft support) decode at full size
    png = tmp_path / 'c0' / 'b.png'
    big.save(str(png))
    ds2 = ImageFolder(str(tmp_path), None, decode_size=448)
    p2 = [p for p, _ in ds2.samples if p.endswith('.png')][0]
    assert ds2.loader(p2).size == (1200, 900)


def test_device_prefetcher_cpu_passthrough():
    """CPU fallback: yields every batch unchanged, in order; and a
    gpu-marked variant checks the stream path separately."""
    from torch.utils.data import DataLoader

    from mgproto_amd.data.prefetch import DevicePrefetcher
    from mgproto_amd.data.synthetic import SyntheticImages

    ds = SyntheticImages(n=10, num_classes=3, img_size=32)
    loader = DataLoader(ds, batch_size=4)
    pf = DevicePrefetcher(loader, torch.device('cpu'))
    assert len(pf) == len(loader)
    batches = list(pf)
    ref = list(loader)
    assert len(batches) == len(ref) == 3
    for (a, b, c), (x, y, z) in zip(batches, ref):
        assert torch.equal(a, x) and torch.equal(b, y) and torch.equal(c, z)
