"""Interpretability metrics: consistency / stability / purity.

Reference ``utils/interpretability.py``: run ``push_forward`` over the CUB
test set, take each own-class prototype's max-activation region and match
it against CUB part annotations. Re-designed so that

* the test-set sweep and the activation-map upsampling run batched on the
  GPU (the reference upsamples one map at a time with cv2 on the CPU);
* the scoring math lives in pure-numpy functions
  (``correspondence_from_maps``, ``consistency_score``, ``stability_score``,
  ``purity_score``) that are unit-testable on synthetic data without CUB.
"""

from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch
import torch.nn.functional as F

from .local_parts import CubPartAnnotations, in_bbox


def perturb_img(norm_img: torch.Tensor, std: float = 0.2, eps: float = 0.25):
    """Clipped Gaussian input noise (reference interpretability.py:14-18)."""
    noise = torch.zeros_like(norm_img).normal_(mean=0, std=std)
    noise = torch.clip(noise, min=-eps, max=eps)
    return norm_img + noise


# ---------------------------------------------------------------------------
# pure scoring math (testable without CUB)
# ---------------------------------------------------------------------------

def region_from_map(upsampled: np.ndarray, half_size: int,
                    img_size: int) -> Tuple[int, int, int, int]:
    """(y1, y2, x1, x2) box of half_size around the activation max
    (first max in row-major order, matching np.where semantics)."""
    idx = np.unravel_index(np.argmax(upsampled), upsampled.shape)
    return (max(0, idx[0] - half_size), min(img_size, idx[0] + half_size),
            max(0, idx[1] - half_size), min(img_size, idx[1] + half_size))


def correspondence_from_maps(maps: np.ndarray,
                             part_labels: Sequence[Sequence[Sequence[int]]],
                             part_num: int, half_size: int,
                             img_size: int) -> np.ndarray:
    """[n_img, H', W'] UPSAMPLED activation maps + per-image part labels
    ([part_id, x, y] lists) -> binary [n_img, part_num] correspondence.
    Reference interpretability.py:109-125."""
    n = maps.shape[0]
    out = np.zeros((n, part_num))
    for i in range(n):
        region = region_from_map(maps[i], half_size, img_size)
        for part_id, loc_x, loc_y in part_labels[i]:
            if in_bbox((loc_y, loc_x), region):
                out[i, part_id] = 1
    return out


def consistency_score(all_proto_to_part: List[np.ndarray],
                      all_proto_part_mask: List[np.ndarray],
                      part_thresh: float = 0.8) -> float:
    """A prototype is consistent iff some part is matched in >= part_thresh
    of the images where that part is visible (reference :134-161)."""
    consis = []
    for proto_to_part, mask in zip(all_proto_to_part, all_proto_part_mask):
        assert ((1. - mask) * proto_to_part).sum() == 0
        num = proto_to_part.sum(axis=0)
        den = mask.sum(axis=0)
        den = np.where(den == 0, den + 1, den)
        mean_part = (num / den >= part_thresh)
        consis.append(1 if mean_part.sum() > 0 else 0)
    return float(np.mean(consis) * 100)


def stability_score(all_proto_to_part: List[np.ndarray],
                    all_proto_to_part_noise: List[np.ndarray]) -> float:
    """Fraction of images whose part correspondence is unchanged under
    input noise, averaged over prototypes (reference :163-181)."""
    stab = []
    for a, b in zip(all_proto_to_part, all_proto_to_part_noise):
        is_equal = (np.abs(a - b).sum(axis=-1) == 0).astype(np.float32)
        stab.append(is_equal.mean())
    return float(np.mean(stab) * 100)


def purity_score(all_proto_to_part: List[np.ndarray]) -> Tuple[float, float]:
    """Mean/std over prototypes of the best part's match rate over the
    top-K activating images (reference :298-315)."""
    purity = [float(p.mean(0).max()) for p in all_proto_to_part]
    return float(np.mean(purity) * 100), float(np.std(purity) * 100)


# ---------------------------------------------------------------------------
# data collection (GPU sweep)
# ---------------------------------------------------------------------------

@torch.no_grad()
def _sweep_own_class_maps(ppnet, loader, device, use_noise=False):
    """Run push_forward over the loader, keep each image's own-class
    prototype activation maps. Returns (acts [N, K, h, w] cpu fp32,
    targets [N], img_ids [N])."""
    ppnet.eval()
    m = ppnet.module if hasattr(ppnet, 'module') else ppnet
    K = m.num_prototypes_per_class
    all_acts, all_targets, all_ids = [], [], []
    for batch in loader:
        data, targets, img_ids = batch[0], batch[1], batch[2]
        data = data.to(device)
        targets_d = targets.to(device)
        if use_noise:
            data = perturb_img(data)
        _, proto_dist = m.push_forward(data)
        proto_acts = -proto_dist                              # [B, P, h, w]
        fea = proto_acts.shape[-1]
        idx = (targets_d * K).unsqueeze(-1) + torch.arange(K, device=device)
        idx = idx[:, :, None, None].expand(-1, -1, proto_acts.shape[2], fea)
        own = torch.gather(proto_acts, 1, idx)                # [B, K, h, w]
        all_acts.append(own.float().cpu())
        all_targets.append(torch.as_tensor(targets).cpu())
        all_ids.append(torch.as_tensor(img_ids).cpu())
    return (torch.cat(all_acts), torch.cat(all_targets).numpy(),
            torch.cat(all_ids).numpy())


def _upsample_maps(acts: torch.Tensor, img_size: int,
                   device) -> np.ndarray:
    """Bicubic-upsample [n, K, h, w] -> numpy [n, K, img, img] in batches."""
    out = []
    for i in range(0, acts.shape[0], 64):
        blk = acts[i:i + 64].to(device)
        up = F.interpolate(blk, size=(img_size, img_size), mode='bicubic',
                           align_corners=False)
        out.append(up.cpu().numpy())
    return np.concatenate(out, axis=0)


def _image_part_labels(ann: CubPartAnnotations, img_id: int, img_size: int,
                       image_sizes: Dict[int, Tuple[int, int]]):
    """Part labels of one image rescaled to the resized image
    ([part_id0, x, y] with ids starting at 0), plus the part mask."""
    ow, oh = image_sizes[img_id]
    labels, mask = [], np.zeros(ann.part_num)
    for part_id, loc_x, loc_y in ann.id_to_part_loc.get(img_id, []):
        pid0 = part_id - 1
        mask[pid0] = 1
        labels.append([pid0, int(img_size * loc_x / ow),
                       int(img_size * loc_y / oh)])
    return labels, mask


def _original_sizes(ann: CubPartAnnotations, img_ids, data_path):
    """(width, height) per image id via PIL header reads (no full decode)."""
    import os
    from PIL import Image
    sizes = {}
    for img_id in np.unique(img_ids):
        folder, name = ann.id_to_path[int(img_id)]
        with Image.open(os.path.join(data_path, 'images', folder, name)) as im:
            sizes[int(img_id)] = im.size
    return sizes


def get_corresponding_object_parts(ppnet, loader, ann: CubPartAnnotations,
                                   data_path: str, half_size: int,
                                   device=None, use_noise=False,
                                   topK: Optional[int] = None):
    """Per prototype: binary image x part correspondence (+ part masks).

    ``topK`` selects, per prototype, only its topK max-activating images of
    its class (the purity variant, reference :183-296); None keeps all
    class images (consistency/stability variant, reference :22-131).
    """
    m = ppnet.module if hasattr(ppnet, 'module') else ppnet
    device = device or next(m.parameters()).device
    img_size = m.img_size
    K = m.num_prototypes_per_class

    acts, targets, img_ids = _sweep_own_class_maps(ppnet, loader, device,
                                                   use_noise)
    sizes = _original_sizes(ann, img_ids, data_path)

    all_proto_to_part, all_proto_part_mask = [], []
    for cls in range(m.num_classes):
        sel = np.nonzero(targets == cls)[0]
        if len(sel) == 0:
            for _ in range(K):
                all_proto_to_part.append(np.zeros((0, ann.part_num)))
                all_proto_part_mask.append(np.zeros((0, ann.part_num)))
            continue
        cls_acts = acts[sel]                                 # [n, K, h, w]
        cls_ids = img_ids[sel]
        up = _upsample_maps(cls_acts, img_size, device)      # [n, K, I, I]

        labels, masks = [], []
        for img_id in cls_ids:
            lab, mask = _image_part_labels(ann, int(img_id), img_size, sizes)
            labels.append(lab)
            masks.append(mask)
        masks = np.stack(masks, axis=0)

        if topK is not None:
            flat_max = up.max(axis=(2, 3))                   # [n, K]
            order = np.argsort(flat_max, axis=0)[::-1][:topK]  # [topK, K]

        for k in range(K):
            if topK is None:
                corr = correspondence_from_maps(up[:, k], labels,
                                                ann.part_num, half_size,
                                                img_size)
                all_proto_part_mask.append(masks)
            else:
                pick = order[:, k]
                corr = correspondence_from_maps(
                    up[pick, k], [labels[i] for i in pick],
                    ann.part_num, half_size, img_size)
                all_proto_part_mask.append(masks[pick])
            all_proto_to_part.append(corr)
    return all_proto_to_part, all_proto_part_mask


# ---------------------------------------------------------------------------
# public entry points (reference signatures)
# ---------------------------------------------------------------------------

def evaluate_consistency(ppnet, loader, ann, data_path, half_size=36,
                         part_thresh=0.8, device=None):
    p2p, mask = get_corresponding_object_parts(ppnet, loader, ann, data_path,
                                               half_size, device=device)
    return consistency_score(p2p, mask, part_thresh)


def evaluate_stability(ppnet, loader, ann, data_path, half_size=36,
                       device=None):
    p2p, _ = get_corresponding_object_parts(ppnet, loader, ann, data_path,
                                            half_size, device=device)
    p2p_noise, _ = get_corresponding_object_parts(ppnet, loader, ann,
                                                  data_path, half_size,
                                                  device=device,
                                                  use_noise=True)
    return stability_score(p2p, p2p_noise)


def evaluate_purity(ppnet, loader, ann, data_path, half_size=16, topK=10,
                    device=None):
    p2p, _ = get_corresponding_object_parts(ppnet, loader, ann, data_path,
                                            half_size, device=device,
                                            topK=topK)
    return purity_score(p2p)
