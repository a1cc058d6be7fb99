"""Interpretability metrics: pure scoring math + end-to-end on a synthetic
CUB-layout fixture (no real dataset in this environment)."""

import os

import numpy as np
import pytest
import torch

from mgproto_amd.utils.interpretability import (
    region_from_map, correspondence_from_maps, consistency_score,
    stability_score, purity_score, perturb_img)


def test_region_from_map():
    m = np.zeros((224, 224))
    m[100, 50] = 5.0
    assert region_from_map(m, 36, 224) == (64, 136, 14, 86)
    m2 = np.zeros((224, 224))
    m2[2, 220] = 1.0
    assert region_from_map(m2, 36, 224) == (0, 38, 184, 224)


def test_correspondence():
    maps = np.zeros((2, 224, 224))
    maps[0, 100, 100] = 1.0
    maps[1, 10, 10] = 1.0
    # image 0: part 0 at (x=110, y=95) inside box; part 1 far away
    labels = [[[0, 110, 95], [1, 200, 200]],
              [[0, 12, 12]]]
    corr = correspondence_from_maps(maps, labels, part_num=3, half_size=36,
                                    img_size=224)
    assert corr[0].tolist() == [1, 0, 0]
    assert corr[1].tolist() == [1, 0, 0]


def test_consistency_score():
    # prototype A matches part 0 on all images -> consistent;
    # prototype B matches nothing -> inconsistent
    p2p_a = np.array([[1, 0], [1, 0], [1, 0]])
    p2p_b = np.zeros((3, 2))
    mask = np.ones((3, 2))
    score = consistency_score([p2p_a, p2p_b], [mask, mask], part_thresh=0.8)
    assert score == 50.0


def test_consistency_respects_visibility_mask():
    # part 0 visible on 2/3 images, matched on both visible -> 2/2 >= 0.8
    p2p = np.array([[1, 0], [1, 0], [0, 0]])
    mask = np.array([[1, 1], [1, 1], [0, 1]])
    assert consistency_score([p2p], [mask]) == 100.0


def test_stability_score():
    a = np.array([[1, 0], [0, 1]])
    b = np.array([[1, 0], [1, 1]])   # second image changed
    assert stability_score([a], [b]) == 50.0
    assert stability_score([a], [a]) == 100.0


def test_purity_score():
    p = np.array([[1, 0], [1, 0], [1, 1], [0, 0]])  # part 0: 3/4
    mean_p, std_p = purity_score([p])
    assert mean_p == 75.0


def test_perturb_img_bounded():
    x = torch.zeros(2, 3, 8, 8)
    y = perturb_img(x, std=0.5, eps=0.25)
    assert (y.abs() <= 0.25 + 1e-6).all()


@pytest.fixture
def synthetic_cub(tmp_path):
    """Minimal CUB-200-2011 directory layout with 2 classes x 3 images."""
    PIL = pytest.importorskip('PIL')
    from PIL import Image
    root = tmp_path / 'CUB'
    (root / 'parts').mkdir(parents=True)
    (root / 'images').mkdir()
    rng = np.random.RandomState(0)
    lines_img, lines_cls, lines_bbox, lines_split, lines_ploc = [], [], [], [], []
    img_id = 1
    for ci, cls in enumerate(['001.A', '002.B']):
        (root / 'images' / cls).mkdir()
        for i in range(3):
            name = f'{cls}/img_{i}.jpg'
            Image.fromarray(rng.randint(0, 255, (60, 80, 3), dtype=np.uint8)) \
                .save(root / 'images' / name)
            lines_img.append(f'{img_id} {name}')
            lines_cls.append(f'{img_id} {ci + 1}')
            lines_bbox.append(f'{img_id} 5.0 5.0 60.0 45.0')
            lines_split.append(f'{img_id} 0')       # all test
            lines_ploc.append(f'{img_id} 1 40.0 30.0 1')
            lines_ploc.append(f'{img_id} 2 10.0 10.0 1')
            lines_ploc.append(f'{img_id} 3 0.0 0.0 0')  # invisible
            img_id += 1
    (root / 'images.txt').write_text('\n'.join(lines_img) + '\n')
    (root / 'image_class_labels.txt').write_text('\n'.join(lines_cls) + '\n')
    (root / 'bounding_boxes.txt').write_text('\n'.join(lines_bbox) + '\n')
    (root / 'train_test_split.txt').write_text('\n'.join(lines_split) + '\n')
    (root / 'parts' / 'parts.txt').write_text(
        '1 head\n2 left leg\n3 right leg\n')
    (root / 'parts' / 'part_locs.txt').write_text('\n'.join(lines_ploc) + '\n')
    return root


def test_local_parts_loader(synthetic_cub):
    from mgproto_amd.utils.local_parts import CubPartAnnotations
    ann = CubPartAnnotations(str(synthetic_cub))
    assert ann.part_num == 3
    assert ann.id_to_path[1] == ('001.A', 'img_0.jpg')
    assert ann.id_to_bbox[1] == (5, 5, 65, 50)
    assert len(ann.id_to_part_loc[1]) == 2      # invisible part excluded
    assert ann.cls_to_id[0] == [1, 2, 3]


def test_cub_eval_dataset(synthetic_cub):
    from mgproto_amd.utils.datasets import Cub2011Eval
    from mgproto_amd.data import transforms as T
    ds = Cub2011Eval(str(synthetic_cub), train=False,
                     transform=T.Compose([T.Resize((64, 64)), T.ToTensor()]))
    assert len(ds) == 6
    img, target, img_id = ds[0]
    assert img.shape == (3, 64, 64) and target == 0 and img_id == 1


def test_end_to_end_consistency_on_synthetic(synthetic_cub):
    from mgproto_amd.model import construct_MGProto
    from mgproto_amd.utils.datasets import Cub2011Eval
    from mgproto_amd.utils.local_parts import CubPartAnnotations
    from mgproto_amd.utils.interpretability import (evaluate_consistency,
                                                    evaluate_purity)
    from mgproto_amd.data import transforms as T
    from mgproto_amd.data.preprocess import mean, std
    from torch.utils.data import DataLoader

    torch.manual_seed(0)
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(8, 16, 1, 1), num_classes=2,
                              add_on_layers_type='regular', sz_embedding=8,
                              mem_capacity=4, mine_K=2)
    tf = T.Compose([T.Resize((64, 64)), T.ToTensor(),
                    T.Normalize(mean=mean, std=std)])
    ds = Cub2011Eval(str(synthetic_cub), train=False, transform=tf)
    loader = DataLoader(ds, batch_size=3)
    ann = CubPartAnnotations(str(synthetic_cub))
    score = evaluate_consistency(model, loader, ann, str(synthetic_cub),
                                 half_size=16, device=torch.device('cpu'))
    assert 0.0 <= score <= 100.0
    mean_p, std_p = evaluate_purity(model, loader, ann, str(synthetic_cub),
                                    half_size=8, topK=2,
                                    device=torch.device('cpu'))
    assert 0.0 <= mean_p <= 100.0


def test_cub_csv_pipeline(synthetic_cub, tmp_path):
    from mgproto_amd.model import construct_MGProto
    from mgproto_amd.utils.datasets import Cub2011Eval
    from mgproto_amd.utils import cub_csv
    from mgproto_amd.data import transforms as T
    from torch.utils.data import DataLoader

    torch.manual_seed(0)
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(8, 16, 1, 1), num_classes=2,
                              add_on_layers_type='regular', sz_embedding=8,
                              mem_capacity=4, mine_K=2)
    tf = T.Compose([T.Resize((64, 64)), T.ToTensor()])
    ds = Cub2011Eval(str(synthetic_cub), train=False, transform=tf)
    # cub_csv expects dataset.imgs with absolute paths
    ds.imgs = [(os.path.join(str(synthetic_cub), 'images', row.filepath),
                row.target - 1) for _, row in ds.data.iterrows()]
    loader = DataLoader(ds, batch_size=3)
    log_dir = str(tmp_path / 'logs')
    csvpath = cub_csv.get_topk_cub(model, loader, k=2, epoch=0,
                                   device=torch.device('cpu'),
                                   log_dir=log_dir, img_size=64)
    assert os.path.isfile(csvpath)
    mean_p, std_p, related = cub_csv.eval_prototypes_cub_parts_csv(
        csvpath,
        os.path.join(str(synthetic_cub), 'parts', 'part_locs.txt'),
        os.path.join(str(synthetic_cub), 'parts', 'parts.txt'),
        os.path.join(str(synthetic_cub), 'images.txt'),
        epoch=0, img_size=64, wshape=4, log=lambda *a: None)
    assert 0.0 <= mean_p <= 1.0


def test_cub_csv_threshold_variant(synthetic_cub, tmp_path):
    """get_proto_patches_cub (threshold CSV, reference cub_csv.py:226-265)
    writes scoreable rows for every patch above threshold."""
    import csv as _csv

    from torch.utils.data import DataLoader

    from mgproto_amd.data import transforms as T
    from mgproto_amd.model import construct_MGProto
    from mgproto_amd.utils import cub_csv
    from mgproto_amd.utils.datasets import Cub2011Eval

    torch.manual_seed(0)
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(8, 16, 1, 1), num_classes=2,
                              add_on_layers_type='regular', sz_embedding=8,
                              mem_capacity=4, mine_K=2)
    tf = T.Compose([T.Resize((64, 64)), T.ToTensor()])
    ds = Cub2011Eval(str(synthetic_cub), train=False, transform=tf)
    loader = DataLoader(ds, batch_size=3)
    csvpath = cub_csv.get_proto_patches_cub(
        model, loader, epoch=0, device=torch.device('cpu'),
        log_dir=str(tmp_path / 'logs'), threshold=0.0, img_size=64)
    with open(csvpath, newline='') as f:
        rows = list(_csv.reader(f))
    assert rows[0][:2] == ['prototype', 'img name']
    assert len(rows) > 1           # threshold 0: probs are positive
    mean_p, _std, _rel = cub_csv.eval_prototypes_cub_parts_csv(
        csvpath,
        os.path.join(str(synthetic_cub), 'parts', 'part_locs.txt'),
        os.path.join(str(synthetic_cub), 'parts', 'parts.txt'),
        os.path.join(str(synthetic_cub), 'images.txt'),
        epoch=0, img_size=64, wshape=4, log=lambda *a: None)
    assert 0.0 <= mean_p <= 1.0
