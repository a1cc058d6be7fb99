"""Prototype projection ("push") — nearest-patch re-anchoring.

Reference behaviour (``/root/reference/push.py``): pass 1 sweeps the push
set recording, for every prototype j and every image of j's class, the
spatial argmin of the distance map; pass 2 sorts each prototype's
candidates by distance, greedily claims images (an image feeds at most one
prototype), re-forwards the chosen image and copies its winning patch
feature into ``prototype_means``; optionally renders bbox/heatmap JPEGs.

MI355X re-design:
* pass-1 argmin runs on device (the reference copies the full [B,P,H,W]
  map to host and np.argmins in Python, push.py:109-151);
* pass 2 batches the re-forwards (the reference re-loads and forwards one
  image at a time, push.py:181-198);
* distributed: each rank sweeps its shard, candidates are all-gathered and
  merged with a deterministic greedy (ties broken by (dist, img, h, w)), so
  results are identical for any GPU count (SURVEY.md hard part #4);
  re-forwards are sharded round-robin and combined with one all-reduce.
* artifact rendering uses PIL/torch (cv2/matplotlib are not available).

``prototype_self_act_filename_prefix`` / ``proto_bound_boxes_filename_prefix``
are accepted for signature parity but have no effect — same as the
reference, where the corresponding arrays are allocated (push.py:63-67)
but never written or saved (SURVEY.md §2.1 dead spots).
"""

import os
import time
from typing import Optional

import numpy as np
import torch
import torch.nn.functional as F

from ..utils.helpers import makedir, find_high_activation_crop


def _unwrap(model):
    return model.module if hasattr(model, 'module') else model


def _batch_fields(item):
    """Accept (img, label, idx) datasets or the reference's
    ((img, label), (path, _)) MyImageFolder format."""
    if isinstance(item[0], (list, tuple)):          # reference format
        (img, label), meta = item[0], item[1]
        return img, label, None, meta[0]
    if len(item) >= 3:
        return item[0], item[1], item[2], None
    return item[0], item[1], None, None


def _item_image(item):
    """Image tensor of one dataset item, both formats."""
    if isinstance(item[0], (list, tuple)):          # reference format
        return item[0][0]
    return item[0]


@torch.no_grad()
def push_prototypes(dataloader,
                    prototype_network_parallel,
                    class_specific: bool = True,
                    preprocess_input_function=None,
                    prototype_layer_stride: int = 1,
                    root_dir_for_saving_prototypes: Optional[str] = None,
                    epoch_number: Optional[int] = None,
                    prototype_img_filename_prefix=None,
                    prototype_self_act_filename_prefix=None,
                    proto_bound_boxes_filename_prefix=None,
                    save_prototype_class_identity: bool = True,
                    log=print,
                    comm=None,
                    device=None):
    model = _unwrap(prototype_network_parallel)
    model.eval()
    log('\tpush')
    start = time.time()

    device = device or next(model.parameters()).device
    P = model.num_prototypes
    K = model.num_prototypes_per_class
    dataset = dataloader.dataset
    # pass-2 re-loads by GLOBAL image index; unwrap rank shards (Subset)
    lookup = dataset
    while isinstance(lookup, torch.utils.data.Subset):
        lookup = lookup.dataset

    proto_epoch_dir = None
    if root_dir_for_saving_prototypes is not None:
        if epoch_number is not None:
            proto_epoch_dir = os.path.join(root_dir_for_saving_prototypes,
                                           'epoch-' + str(epoch_number))
        else:
            proto_epoch_dir = root_dir_for_saving_prototypes
        if comm is None or comm.rank == 0:
            makedir(proto_epoch_dir)

    # ---------------- pass 1: per-(image, own-class prototype) argmin -----
    cand_dists = []   # [M] distances
    cand_meta = []    # [M, 4] (prototype j, global image idx, h, w)
    batch_offset = 0
    for item in dataloader:
        image, label, idx, _path = _batch_fields(item)
        B = image.shape[0]
        if idx is None:
            if comm is not None and comm.is_distributed:
                raise ValueError(
                    'distributed push needs datasets that yield a GLOBAL '
                    'image index per item ((img, label, idx) format) — '
                    'rank-local offsets cannot identify images across ranks')
            idx = torch.arange(batch_offset, batch_offset + B)
        batch_offset += B
        search_batch = (preprocess_input_function(image)
                        if preprocess_input_function is not None else image)
        search_batch = search_batch.to(device, non_blocking=True)
        _, dist = model.push_forward(search_batch)      # [B, P, H, W]
        Bd, Pd, H, W = dist.shape
        mind, argd = dist.view(Bd, Pd, H * W).min(dim=2)  # [B, P]

        gt = label.to(device)
        ar = torch.arange(Bd, device=device)
        own = mind.view(Bd, -1, K)[ar, gt]               # [B, K]
        own_arg = argd.view(Bd, -1, K)[ar, gt]           # [B, K]
        kk = torch.arange(K, device=device)
        j = (gt.unsqueeze(1) * K + kk.unsqueeze(0))      # [B, K] prototype ids
        h = own_arg // W
        w = own_arg % W
        cand_dists.append(own.reshape(-1).float().cpu())
        meta = torch.stack([j.reshape(-1).cpu(),
                            idx.repeat_interleave(K).cpu().long(),
                            h.reshape(-1).cpu(), w.reshape(-1).cpu()], dim=1)
        cand_meta.append(meta)

    dists = torch.cat(cand_dists) if cand_dists else torch.zeros(0)
    meta = (torch.cat(cand_meta) if cand_meta
            else torch.zeros(0, 4, dtype=torch.int64))

    if comm is not None and comm.is_distributed:
        from ..parallel.state_sync import gather_push_candidates
        dists, meta = gather_push_candidates(comm, dists, meta)

    # ---------------- pass 2: deterministic global greedy -----------------
    log('\tExecuting push ...')
    d_np = dists.numpy()
    m_np = meta.numpy()
    # order candidates per prototype by (dist, img, h, w): stable across
    # rank counts and rank orderings
    order = np.lexsort((m_np[:, 3], m_np[:, 2], m_np[:, 1], d_np))
    per_proto = {j: [] for j in range(P)}
    for oi in order:
        per_proto[int(m_np[oi, 0])].append(oi)

    claimed = set()
    chosen = []       # (j, img_idx, h, w)
    for j in range(P):
        for oi in per_proto[j]:
            img_idx = int(m_np[oi, 1])
            if img_idx in claimed:
                continue
            claimed.add(img_idx)
            chosen.append((j, img_idx, int(m_np[oi, 2]), int(m_np[oi, 3])))
            break

    # ---------------- batched re-forward of chosen images -----------------
    d_feat = model.prototype_shape[1]
    updates = torch.zeros(P, d_feat, device=device)
    have = torch.zeros(P, device=device)
    rank = comm.rank if comm is not None else 0
    world = comm.world_size if comm is not None else 1
    my = [c for i, c in enumerate(chosen) if i % world == rank]
    bs = getattr(dataloader, 'batch_size', None) or 32

    for s in range(0, len(my), bs):
        blk = my[s:s + bs]
        imgs = [_item_image(lookup[img_idx]) for (_j, img_idx, _h, _w) in blk]
        batch = torch.stack(imgs)
        if preprocess_input_function is not None:
            batch = preprocess_input_function(batch)
        batch = batch.to(device, non_blocking=True)
        feats, dist = model.push_forward(batch)          # [b, d, H, W]
        for bi, (j, img_idx, h, w) in enumerate(blk):
            updates[j] = feats[bi, :, h, w]
            have[j] = 1.0
            if proto_epoch_dir is not None:
                _render_artifacts(lookup, img_idx, j, dist[bi, j],
                                  proto_epoch_dir,
                                  prototype_img_filename_prefix or 'prototype-img',
                                  preprocess_input_function)

    if comm is not None and comm.is_distributed:
        comm.all_reduce_sum(updates)
        comm.all_reduce_sum(have)

    # apply the projection
    sel = have > 0
    means = model.prototype_means.data.view(P, d_feat)
    means[sel] = updates[sel]

    log('\tpush time: \t{0}'.format(time.time() - start))
    return chosen


def _render_artifacts(dataset, img_idx, j, dist_map, out_dir, prefix,
                      preprocess_input_function):
    """Save original-with-bbox, heatmap overlay, and cropped patch JPEGs
    (reference push.py:203-226), using PIL instead of cv2/matplotlib."""
    try:
        from PIL import Image, ImageDraw
    except ImportError:  # pragma: no cover
        return
    item = dataset[img_idx]
    img = item[0] if not isinstance(item[0], (list, tuple)) else item[0][0]
    img_np = img.permute(1, 2, 0).numpy()
    img_np = np.clip(img_np, 0.0, 1.0)
    Hs, Ws = img_np.shape[0], img_np.shape[1]

    act = (-dist_map).float().unsqueeze(0).unsqueeze(0)
    up = F.interpolate(act, size=(Hs, Ws), mode='bicubic',
                       align_corners=False)[0, 0].cpu().numpy()
    bbox = find_high_activation_crop(up, percentile=95)

    def _save(arr, name, box=None):
        im = Image.fromarray((np.clip(arr, 0, 1) * 255).astype(np.uint8))
        if box is not None:
            dr = ImageDraw.Draw(im)
            dr.rectangle([box[2], box[0], box[3] - 1, box[1] - 1],
                         outline=(0, 255, 255), width=2)
        im.save(os.path.join(out_dir, name), quality=95)

    _save(img_np, f'{j}{prefix}-original.jpg', bbox)

    rescaled = (up - up.min()) / max(up.max() - up.min(), 1e-8)
    heat = _jet(rescaled)
    overlay = 0.5 * img_np + 0.3 * heat
    _save(overlay, f'{j}{prefix}-original_with_self_act.jpg', bbox)

    patch = img_np[bbox[0]:bbox[1], bbox[2]:bbox[3], :]
    if patch.size:
        _save(patch, f'{j}{prefix}.jpg')


def _jet(x: np.ndarray) -> np.ndarray:
    """Minimal jet colormap (cv2.COLORMAP_JET stand-in), x in [0,1]."""
    r = np.clip(1.5 - np.abs(4 * x - 3), 0, 1)
    g = np.clip(1.5 - np.abs(4 * x - 2), 0, 1)
    b = np.clip(1.5 - np.abs(4 * x - 1), 0, 1)
    return np.stack([r, g, b], axis=-1)
