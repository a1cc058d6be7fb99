"""End-to-end runs of the eval_{consistency,stability,purity}.py drivers on
a fabricated mini CUB-200-2011 tree (2 classes, 6 images) with a tiny
checkpoint — exercises Cub2011Eval, CubPartAnnotations and both purity
variants through the real CLIs."""

import os
import subprocess
import sys

import numpy as np
import pytest
import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope='module')
def mini_cub(tmp_path_factory):
    root = tmp_path_factory.mktemp('cub')
    from PIL import Image
    rng = np.random.default_rng(0)
    classes = ['001.First', '002.Second']
    lines_img, lines_lab, lines_split, lines_bbox, lines_parts = \
        [], [], [], [], []
    img_id = 0
    for ci, cls in enumerate(classes):
        d = root / 'images' / cls
        d.mkdir(parents=True)
        for j in range(3):
            img_id += 1
            name = f'im{j}.jpg'
            Image.fromarray(rng.integers(0, 255, (80, 100, 3),
                                         dtype=np.uint8)).save(str(d / name))
            lines_img.append(f'{img_id} {cls}/{name}')
            lines_lab.append(f'{img_id} {ci + 1}')
            lines_split.append(f'{img_id} 0')           # all test images
            lines_bbox.append(f'{img_id} 10.0 10.0 60.0 50.0')
            for pid in (1, 2, 3):
                vis = 1 if (pid + j) % 3 else 0
                lines_parts.append(
                    f'{img_id} {pid} {20.0 + 10 * pid} {15.0 + 5 * pid} {vis}')
    (root / 'parts').mkdir()
    (root / 'images.txt').write_text('\n'.join(lines_img) + '\n')
    (root / 'image_class_labels.txt').write_text('\n'.join(lines_lab) + '\n')
    (root / 'train_test_split.txt').write_text('\n'.join(lines_split) + '\n')
    (root / 'bounding_boxes.txt').write_text('\n'.join(lines_bbox) + '\n')
    (root / 'parts' / 'parts.txt').write_text('1 beak\n2 tail\n3 head\n')
    (root / 'parts' / 'part_locs.txt').write_text('\n'.join(lines_parts) + '\n')

    from mgproto_amd.model import construct_MGProto
    torch.manual_seed(0)
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(8, 16, 1, 1), num_classes=2,
                              add_on_layers_type='regular', sz_embedding=8,
                              mem_capacity=8, mine_K=2)
    ckpt = root / 'model.pth'
    torch.save(model.state_dict(), str(ckpt))
    return root, ckpt


def _run(script, mini_cub, extra=()):
    root, ckpt = mini_cub
    cmd = [sys.executable, os.path.join(ROOT, script),
           '--data_path', str(root), '--resume', str(ckpt),
           '--base_architecture', 'resnet18', '--prototype_shape', '8', '16',
           '1', '--nb_classes', '2', '--img_size', '64',
           '--test_batch_size', '4', *extra]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                       cwd=ROOT)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    return r.stdout


def test_eval_consistency_driver(mini_cub):
    out = _run('eval_consistency.py', mini_cub, ('--half_size', '18'))
    assert 'Consistency Score' in out


def test_eval_stability_driver(mini_cub):
    out = _run('eval_stability.py', mini_cub, ('--half_size', '18'))
    assert 'Stability Score' in out


def test_eval_purity_region_driver(mini_cub):
    out = _run('eval_purity.py', mini_cub, ('--topK', '2'))
    assert 'Purity Score' in out


def test_eval_purity_csv_driver(mini_cub, tmp_path):
    out = _run('eval_purity.py', mini_cub,
               ('--topK', '2', '--csv', '--log_dir', str(tmp_path / 'logs')))
    assert 'CSV Purity' in out


def test_eval_consumes_train_driver_checkpoint(mini_cub, tmp_path):
    """Full hand-off: a train.py-produced checkpoint (latest.pth with the
    {'model': ...} wrapper + optimizer state) loads in the eval drivers."""
    env = dict(os.environ)
    env['MGPROTO_TINY_TEST'] = '1'
    cmd = [sys.executable, os.path.join(ROOT, 'train.py'),
           '-arch', 'resnet18', '-mem_sz', '8', '-mine_level', '3',
           '-aux_emb_sz', '8', '--epochs', '1',
           '--out', str(tmp_path / 'run')]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=900,
                       env=env, cwd=ROOT)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    ckpt = tmp_path / 'run' / 'latest.pth'

    root, _ = mini_cub
    cmd = [sys.executable, os.path.join(ROOT, 'eval_consistency.py'),
           '--data_path', str(root), '--resume', str(ckpt),
           '--base_architecture', 'resnet18', '--prototype_shape', '30',
           '32', '1', '--nb_classes', '10', '--img_size', '64',
           '--test_batch_size', '4', '--half_size', '18']
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                       cwd=ROOT)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert 'Consistency Score' in r.stdout
