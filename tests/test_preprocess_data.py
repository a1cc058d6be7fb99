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
