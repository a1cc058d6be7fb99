"""PIP-Net-style CUB part-purity evaluation via patch-coordinate CSVs.

Reference ``utils/cub_csv.py``: write per-prototype top-k / thresholded
image-patch coordinate CSVs (coordinates on the 224-resized image), then
score them against CUB part locations with left/right part merging.
Re-implemented with the same CSV schema ("prototype, img name, h_min_224,
h_max_224, w_min_224, w_max_224") and scoring semantics; the model sweep is
batched (the reference forwards one image at a time, cub_csv.py:245-253).
"""

import csv
import os
from typing import Dict, List

import numpy as np
import torch


def get_img_coordinates(img_size, softmaxes_shape, patchsize, skip,
                        h_idx, w_idx):
    """Latent (h, w) index -> pixel patch box (reference cub_csv.py:14-46,
    minus the 26x26-convnext special case which no backbone here hits)."""
    h_min = h_idx * skip
    h_max = min(img_size, h_idx * skip + patchsize)
    w_min = w_idx * skip
    w_max = min(img_size, w_idx * skip + patchsize)
    if h_idx == softmaxes_shape[1] - 1:
        h_max = img_size
    if w_idx == softmaxes_shape[2] - 1:
        w_max = img_size
    if h_max == img_size:
        h_min = img_size - patchsize
    if w_max == img_size:
        w_min = img_size - patchsize
    return h_min, h_max, w_min, w_max


def get_patch_size(img_size: int, wshape: int, patchsize: int = 32):
    skip = round((img_size - patchsize) / (wshape - 1))
    return patchsize, skip


@torch.no_grad()
def _sweep(net, loader, device):
    """Batched push_forward sweep: per image (pooled [P], argmax h, w)."""
    m = net.module if hasattr(net, 'module') else net
    m.eval()
    pooled_all, hw_all = [], []
    wshape = None
    for batch in loader:
        xs = batch[0].to(device)
        _, dist = m.push_forward(xs)
        acts = -dist                                   # [B, P, h, w]
        wshape = acts.shape[-1]
        B, P, H, W = acts.shape
        flat = acts.view(B, P, H * W)
        pooled, arg = flat.max(dim=2)
        pooled_all.append(pooled.cpu())
        hw_all.append(torch.stack([arg // W, arg % W], dim=2).cpu())
    return torch.cat(pooled_all), torch.cat(hw_all), wshape


def _dataset_imgs(ds):
    """(path, target) list for ImageFolder-style datasets (.imgs, as the
    reference's project loaders provide) or Cub2011Eval (metadata frame)."""
    if hasattr(ds, 'imgs'):
        return ds.imgs
    if hasattr(ds, 'data') and hasattr(ds, 'root'):
        return [(os.path.join(ds.root, 'images', r.filepath), r.target - 1)
                for _, r in ds.data.iterrows()]
    raise TypeError(f'unsupported dataset type {type(ds).__name__}: '
                    'need .imgs or CUB metadata')


@torch.no_grad()
def get_topk_cub(net, projectloader, k, epoch, device, log_dir, img_size=224):
    """Write the top-k patch-coordinate CSV per prototype
    (reference cub_csv.py:267-349)."""
    m = net.module if hasattr(net, 'module') else net
    imgs = _dataset_imgs(projectloader.dataset)
    weights = m.last_layer.weight
    pooled, hw, wshape = _sweep(net, projectloader, device)
    patchsize, skip = get_patch_size(img_size, wshape)

    relevant = (weights.max(dim=0).values > 1e-5).cpu().numpy()
    csvfilepath = os.path.join(log_dir, f'{epoch}_pipnet_prototypes_cub_topk.csv')
    os.makedirs(log_dir, exist_ok=True)
    rows = []
    P = pooled.shape[1]
    for p in range(P):
        if not relevant[p]:
            continue
        scores = pooled[:, p].numpy()
        top = np.argsort(-scores, kind='stable')[:k]
        for imgid in top:
            h_idx, w_idx = int(hw[imgid, p, 0]), int(hw[imgid, p, 1])
            box = get_img_coordinates(img_size, (P, wshape, wshape),
                                      patchsize, skip, h_idx, w_idx)
            rows.append([p, imgs[imgid][0], *box])
    with open(csvfilepath, 'w', newline='') as f:
        w = csv.writer(f, delimiter=',')
        w.writerow(['prototype', 'img name', 'h_min_224', 'h_max_224',
                    'w_min_224', 'w_max_224'])
        w.writerows(rows)
    return csvfilepath


@torch.no_grad()
def get_proto_patches_cub(net, projectloader, epoch, device, log_dir,
                          threshold=0.5, img_size=224):
    """Write all patches whose similarity exceeds threshold
    (reference cub_csv.py:226-265)."""
    m = net.module if hasattr(net, 'module') else net
    imgs = _dataset_imgs(projectloader.dataset)
    weights = m.last_layer.weight
    pooled, hw, wshape = _sweep(net, projectloader, device)
    patchsize, skip = get_patch_size(img_size, wshape)
    relevant = (weights.max(dim=0).values > 1e-5).cpu().numpy()

    csvfilepath = os.path.join(log_dir, f'{epoch}_pipnet_prototypes_cub_all.csv')
    os.makedirs(log_dir, exist_ok=True)
    rows = []
    N, P = pooled.shape
    for imgid in range(N):
        for p in range(P):
            if relevant[p] and float(pooled[imgid, p]) > threshold:
                h_idx, w_idx = int(hw[imgid, p, 0]), int(hw[imgid, p, 1])
                box = get_img_coordinates(img_size, (P, wshape, wshape),
                                          patchsize, skip, h_idx, w_idx)
                rows.append([p, imgs[imgid][0], *box])
    with open(csvfilepath, 'w', newline='') as f:
        w = csv.writer(f, delimiter=',')
        w.writerow(['prototype', 'img name', 'h_min_224', 'h_max_224',
                    'w_min_224', 'w_max_224'])
        w.writerows(rows)
    return csvfilepath


def eval_prototypes_cub_parts_csv(csvfile, parts_loc_path, parts_name_path,
                                  imgs_id_path, epoch, img_size=224,
                                  wshape=28, log=print):
    """Score a patch CSV against CUB part locations
    (reference cub_csv.py:57-225): per (prototype, part), the fraction of
    the prototype's patches containing the part; left/right parts merged;
    reports mean/std purity of each prototype's purest part."""
    from PIL import Image

    patchsize, _ = get_patch_size(img_size, wshape)
    path_to_id = {}
    with open(imgs_id_path) as f:
        for line in f:
            iid, path = line.strip().split(' ')
            path_to_id[path] = iid

    img_to_part_xy = {}
    with open(parts_loc_path) as f:
        for line in f:
            img, partid, x, y, vis = line.strip().split(' ')
            img_to_part_xy.setdefault(img, {})
            if vis == '1':
                img_to_part_xy[img][partid] = (float(x), float(y))

    parts_id_to_name, parts_name_to_id = {}, {}
    with open(parts_name_path) as f:
        for line in f:
            iid, name = line.strip().split(' ', 1)
            parts_id_to_name[iid] = name
            parts_name_to_id[name] = iid
    duplicate_part_ids = [(iid, parts_name_to_id[name.replace('left', 'right')])
                          for iid, name in parts_id_to_name.items()
                          if 'left' in name]

    presences: Dict[str, Dict[str, List[int]]] = {}
    with open(csvfile, newline='') as f:
        reader = csv.reader(f, delimiter=',')
        next(reader)
        for (proto, imgname, h_min, h_max, w_min, w_max) in reader:
            presences.setdefault(proto, {})
            with Image.open(imgname) as im:
                ow, oh = im.size
            key = '/'.join(imgname.replace('\\', '/').split('/')[-2:])
            if 'normal_' in key:
                key = key.split('normal_')[-1]
            img_id = path_to_id[key]
            h_min, h_max = float(h_min), float(h_max)
            w_min, w_max = float(w_min), float(w_max)
            # clamp oversized patches to their center (reference :120-127)
            if h_max - h_min > patchsize:
                corr = (h_max - h_min) - patchsize
                h_min += corr // 2.
                h_max -= corr // 2.
            if w_max - w_min > patchsize:
                corr = (w_max - w_min) - patchsize
                w_min += corr // 2.
                w_max -= corr // 2.
            oh_min = (oh / img_size) * h_min
            oh_max = (oh / img_size) * h_max
            ow_min = (ow / img_size) * w_min
            ow_max = (ow / img_size) * w_max

            part_dict = img_to_part_xy.get(img_id, {})
            for part, (x, y) in part_dict.items():
                inside = int(oh_min <= y <= oh_max and ow_min <= x <= ow_max)
                presences[proto].setdefault(part, []).append(inside)
            # merge left into right (reference :145-159)
            for left, right in duplicate_part_ids:
                if left in part_dict:
                    if right in part_dict:
                        p0 = presences[proto][left][-1]
                        if p0 > presences[proto][right][-1]:
                            presences[proto][right][-1] = p0
                        del presences[proto][left]
                    else:
                        presences[proto].setdefault(right, []).append(
                            presences[proto][left][-1])
                        del presences[proto][left]

    log(f'\n Eval CUB Parts - Epoch: \t{epoch}')
    log(f'Number of prototypes in parts_presences: \t{len(presences)}')
    max_purity, best_part, related = {}, {}, 0
    for proto, parts in presences.items():
        max_purity[proto] = 0.
        best_sum = -1
        for part, pres in parts.items():
            purity = float(np.mean(pres))
            ssum = int(np.sum(pres))
            if purity > max_purity[proto] or (purity == max_purity[proto]
                                              and ssum > best_sum):
                max_purity[proto] = purity
                best_part[proto] = parts_id_to_name[part]
                best_sum = ssum
        if max_purity[proto] > 0.5:
            related += 1
    vals = list(max_purity.values())
    log(f'Number of part-related prototypes (purity>0.5): \t{related}')
    log('Mean purity of prototypes: \t{0}  std: \t{1}'.format(
        np.mean(vals) if vals else 0.0, np.std(vals) if vals else 0.0))
    return (float(np.mean(vals)) if vals else 0.0,
            float(np.std(vals)) if vals else 0.0, related)
