"""preprocess_data CLI: crop-cub, augment, pets-restructure on tiny fixtures."""

import os
import subprocess
import sys

import numpy as np
import pytest

PIL = pytest.importorskip('PIL')
from PIL import Image

TOOL = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                    'tools', 'preprocess_data.py')


def _run(*args):
    r = subprocess.run([sys.executable, TOOL, *args], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    return r.stdout


def test_crop_cub(tmp_path):
    root = tmp_path / 'cub'
    (root / 'images' / '001.A').mkdir(parents=True)
    rng = np.random.RandomState(0)
    Image.fromarray(rng.randint(0, 255, (50, 70, 3), dtype=np.uint8)) \
        .save(root / 'images' / '001.A' / 'x.jpg')
    (root / 'images.txt').write_text('1 001.A/x.jpg\n')
    (root / 'bounding_boxes.txt').write_text('1 10.0 5.0 30.0 20.0\n')
    (root / 'train_test_split.txt').write_text('1 1\n')
    out = tmp_path / 'out'
    _run('crop-cub', '--root', str(root), '--out', str(out))
    cropped = Image.open(out / 'train_cropped' / '001.A' / 'x.jpg')
    assert cropped.size == (30, 20)


def test_augment(tmp_path):
    src = tmp_path / 'train' / '001.A'
    src.mkdir(parents=True)
    Image.fromarray(np.zeros((32, 32, 3), dtype=np.uint8)).save(src / 'a.jpg')
    out = tmp_path / 'aug'
    _run('augment', '--root', str(tmp_path / 'train'), '--out', str(out),
         '--repeats', '2')
    files = os.listdir(out / '001.A')
    assert len(files) == 4 * 2   # 4 pipelines x 2 repeats


def test_pets_restructure(tmp_path):
    imgs = tmp_path / 'imgs'
    imgs.mkdir()
    Image.fromarray(np.zeros((8, 8, 3), dtype=np.uint8)).save(imgs / 'cat_1.jpg')
    labels = tmp_path / 'trainval.txt'
    labels.write_text('cat_1 3 1 1\n')
    out = tmp_path / 'pets'
    _run('pets-restructure', '--root', str(imgs), '--labels', str(labels),
         '--out', str(out))
    assert (out / '3' / 'cat_1.jpg').is_file()


def test_crop_cars(tmp_path):
    scipy_io = pytest.importorskip('scipy.io')
    root = tmp_path / 'cars'
    (root / 'car_ims').mkdir(parents=True)
    rng = np.random.RandomState(1)
    Image.fromarray(rng.randint(0, 255, (60, 90, 3), dtype=np.uint8)) \
        .save(root / 'car_ims' / '000001.jpg')
    ann = np.zeros((1,), dtype=[('relative_im_path', 'O'),
                                ('bbox_x1', 'O'), ('bbox_y1', 'O'),
                                ('bbox_x2', 'O'), ('bbox_y2', 'O'),
                                ('class', 'O'), ('test', 'O')])
    ann[0] = ('car_ims/000001.jpg', 10, 5, 50, 45, 3, 0)
    mat = tmp_path / 'cars_annos.mat'
    scipy_io.savemat(str(mat), {'annotations': ann.reshape(1, -1)})
    out = tmp_path / 'out'
    _run('crop-cars', '--root', str(root), '--annos', str(mat),
         '--out', str(out))
    cropped = Image.open(out / 'train_cropped' / '003' / '000001.jpg')
    assert cropped.size == (40, 40)


def test_binarize_masks(tmp_path):
    root = tmp_path / 'segs'
    (root / '001.A').mkdir(parents=True)
    # three gray levels: the two darkest are background
    m = np.zeros((20, 20), dtype=np.uint8)
    m[5:15, 5:15] = 128
    m[8:12, 8:12] = 255
    Image.fromarray(m).save(root / '001.A' / 'x.png')
    out = tmp_path / 'out'
    _run('binarize-masks', '--root', str(root), '--out', str(out))
    fg = np.array(Image.open(out / '001.A' / 'x.png'))
    assert set(np.unique(fg)) <= {0, 255}
    assert fg[10, 10] == 255 and fg[0, 0] == 0 and fg[6, 6] == 0


def test_crop_masks(tmp_path):
    cub = tmp_path / 'cub'
    cub.mkdir()
    (cub / 'images.txt').write_text('1 001.A/x.jpg\n')
    (cub / 'bounding_boxes.txt').write_text('1 4.0 2.0 10.0 8.0\n')
    masks = tmp_path / 'masks'
    (masks / '001.A').mkdir(parents=True)
    Image.fromarray(np.full((30, 30), 255, dtype=np.uint8)) \
        .save(masks / '001.A' / 'x.png')
    out = tmp_path / 'out'
    _run('crop-masks', '--root', str(masks), '--cub-root', str(cub),
         '--out', str(out))
    cropped = Image.open(out / '001.A' / 'x.png')
    assert cropped.size == (10, 8)
