#!/usr/bin/env python3
"""Offline dataset preparation — the reference's ``preprocess_data/``
scripts consolidated into one CLI with path arguments (the originals
hardcode personal paths and need the Augmentor package, absent here;
augmentation uses mgproto_amd.data.transforms instead).

Subcommands:
  crop-cub          crop CUB images by bounding box + train/test split
                    (reference preprocess_data/cropimages.py)
  crop-cars         crop Stanford Cars by cars_annos.mat boxes + split
                    (reference preprocess_data/cropimages_cars.py)
  binarize-masks    foreground/background masks from segmentations
                    (reference preprocess_data/preprocess_mask.py)
  crop-masks        crop masks by CUB bounding boxes
                    (reference preprocess_data/cropmasks.py)
  augment           offline x40 augmentation (rotate/skew/shear/distort
                    pipelines, reference preprocess_data/img_aug.py)
  pets-restructure  Oxford-IIIT Pets into class folders
                    (reference preprocess_data/img_pets.py)
"""

import argparse
import os
import random
import sys
from shutil import copyfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np


def makedir(path):
    os.makedirs(path, exist_ok=True)


def crop_cub(args):
    import pandas as pd
    from PIL import Image
    root = args.root
    names = pd.read_table(os.path.join(root, 'images.txt'), delimiter=' ',
                          names=['id', 'name']).to_numpy()
    boxs = pd.read_table(os.path.join(root, 'bounding_boxes.txt'),
                         delimiter=' ',
                         names=['id', 'x', 'y', 'width', 'height']).to_numpy()
    labels = pd.read_table(os.path.join(root, 'train_test_split.txt'),
                           delimiter=' ', names=['id', 'label']).to_numpy()
    trainpath = os.path.join(args.out, 'train_cropped')
    testpath = os.path.join(args.out, 'test_cropped')
    for i in range(len(names)):
        src = os.path.join(root, 'images', names[i][1])
        im = Image.open(src).convert('RGB')
        b = boxs[i]
        im = im.crop((b[1], b[2], b[1] + b[3], b[2] + b[4]))
        dst_root = trainpath if labels[i][1] == 1 else testpath
        dst = os.path.join(dst_root, names[i][1])
        makedir(os.path.dirname(dst))
        im.save(dst, quality=95)
    print(f'cropped {len(names)} images into {args.out}')


def crop_cars(args):
    import scipy.io
    from PIL import Image
    mat = scipy.io.loadmat(args.annos)['annotations'][0]
    test_flag = np.array([int(info[-1]) for info in mat])
    boxs = np.array([[int(info[1]), int(info[2]), int(info[3]), int(info[4])]
                     for info in mat])
    classes = np.array([int(info[-2]) for info in mat])
    names = np.array([str(info[0][0]) for info in mat])
    for i in range(len(names)):
        im = Image.open(os.path.join(args.root, names[i])).convert('RGB')
        b = boxs[i]
        im = im.crop((b[0], b[1], b[2], b[3]))
        cls = f'{classes[i]:03d}'
        split = 'test_cropped' if test_flag[i] else 'train_cropped'
        dst = os.path.join(args.out, split, cls, os.path.basename(names[i]))
        makedir(os.path.dirname(dst))
        im.save(dst, quality=95)
    print(f'cropped {len(names)} car images into {args.out}')


def binarize_masks(args):
    from glob import glob
    from PIL import Image
    paths = glob(os.path.join(args.root, '*', '*.png'))
    for p in paths:
        mask = np.array(Image.open(p).convert('L'))
        lv = np.sort(np.unique(mask))
        # background = two darkest levels (reference preprocess_mask.py:28-30)
        bg = np.logical_or(mask == lv[0], mask == lv[min(1, len(lv) - 1)])
        fg = np.logical_not(bg).astype(np.uint8) * 255
        dst = p.replace(args.root.rstrip('/'), args.out.rstrip('/'))
        makedir(os.path.dirname(dst))
        Image.fromarray(fg).save(dst)
    print(f'binarized {len(paths)} masks into {args.out}')


def crop_masks(args):
    import pandas as pd
    from PIL import Image
    names = pd.read_table(os.path.join(args.cub_root, 'images.txt'),
                          delimiter=' ', names=['id', 'name']).to_numpy()
    boxs = pd.read_table(os.path.join(args.cub_root, 'bounding_boxes.txt'),
                         delimiter=' ',
                         names=['id', 'x', 'y', 'width', 'height']).to_numpy()
    n = 0
    for i in range(len(names)):
        src = os.path.join(args.root, names[i][1]).rsplit('.', 1)[0] + '.png'
        if not os.path.isfile(src):
            continue
        im = Image.open(src)
        b = boxs[i]
        im = im.crop((b[1], b[2], b[1] + b[3], b[2] + b[4]))
        dst = os.path.join(args.out, names[i][1]).rsplit('.', 1)[0] + '.png'
        makedir(os.path.dirname(dst))
        im.save(dst)
        n += 1
    print(f'cropped {n} masks into {args.out}')


def augment(args):
    """Offline x(4 pipelines x repeats) augmentation with flip, matching the
    reference's Augmentor recipes: rotate +-15, skew 0.2, shear +-10,
    random distortion."""
    from PIL import Image
    from mgproto_amd.data import transforms as T

    pipelines = {
        'rot': T.RandomAffine(degrees=15),
        'skew': T.RandomPerspective(distortion_scale=0.2, p=1.0),
        'shear': T.RandomAffine(degrees=0, shear=(-10, 10)),
        'distort': T.RandomPerspective(distortion_scale=0.15, p=1.0),
    }
    flip = T.RandomHorizontalFlip(0.5)
    rng = random.Random(args.seed)
    classes = sorted(next(os.walk(args.root))[1])
    total = 0
    for cls in classes:
        src_dir = os.path.join(args.root, cls)
        dst_dir = os.path.join(args.out, cls)
        makedir(dst_dir)
        for fname in sorted(os.listdir(src_dir)):
            if not fname.lower().endswith(('.jpg', '.jpeg', '.png')):
                continue
            img = Image.open(os.path.join(src_dir, fname)).convert('RGB')
            stem = fname.rsplit('.', 1)[0]
            for pname, pipe in pipelines.items():
                for r in range(args.repeats):
                    random.seed(rng.random())
                    out = flip(pipe(img))
                    out.save(os.path.join(dst_dir,
                                          f'{stem}_{pname}{r}.jpg'),
                             quality=95)
                    total += 1
    print(f'wrote {total} augmented images into {args.out}')


def pets_restructure(args):
    img_list = open(args.labels).readlines()
    for line in img_list:
        info = line.strip().split(' ')
        src = os.path.join(args.root, info[0] + '.jpg')
        dst = os.path.join(args.out, info[1], info[0] + '.jpg')
        makedir(os.path.dirname(dst))
        copyfile(src, dst)
    print(f'restructured {len(img_list)} pet images into {args.out}')


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    sub = ap.add_subparsers(dest='cmd', required=True)

    p = sub.add_parser('crop-cub')
    p.add_argument('--root', required=True)
    p.add_argument('--out', required=True)
    p.set_defaults(fn=crop_cub)

    p = sub.add_parser('crop-cars')
    p.add_argument('--root', required=True)
    p.add_argument('--annos', required=True)
    p.add_argument('--out', required=True)
    p.set_defaults(fn=crop_cars)

    p = sub.add_parser('binarize-masks')
    p.add_argument('--root', required=True)
    p.add_argument('--out', required=True)
    p.set_defaults(fn=binarize_masks)

    p = sub.add_parser('crop-masks')
    p.add_argument('--root', required=True)
    p.add_argument('--cub-root', required=True)
    p.add_argument('--out', required=True)
    p.set_defaults(fn=crop_masks)

    p = sub.add_parser('augment')
    p.add_argument('--root', required=True)
    p.add_argument('--out', required=True)
    p.add_argument('--repeats', type=int, default=10)
    p.add_argument('--seed', type=int, default=0)
    p.set_defaults(fn=augment)

    p = sub.add_parser('pets-restructure')
    p.add_argument('--root', required=True)
    p.add_argument('--labels', required=True)
    p.add_argument('--out', required=True)
    p.set_defaults(fn=pets_restructure)

    args = ap.parse_args()
    args.fn(args)


if __name__ == '__main__':
    main()
