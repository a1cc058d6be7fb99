#!/usr/bin/env python3
"""Push-projection equivalence A/B: reference push.py vs engine/push.py.

The push is where this framework's re-design diverges most from the
reference implementation (device argmin + deterministic global greedy +
batched re-forwards vs the reference's host numpy argmin + per-image
re-loads, reference push.py:80-198). This harness runs BOTH on identical
models (ours loads the reference's state_dict) over the same on-disk
image set and compares

  * the prototype -> image assignment each push chose, and
  * the updated ``prototype_means`` tensors.

Ties in min-distances are broken differently by design (ours:
(dist, img, h, w) lexsort — rank-count invariant; reference: stable sort
by dist with batch-order ties), so exact assignment equality is expected
whenever distances are distinct — which real features give.

    python tools/ab_push.py --out profiles/ab_push.md
"""

import argparse
import os
import sys
import tempfile

import numpy as np
import torch
from torch.utils.data import DataLoader, Dataset

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tools.ab_reference import load_reference, load_reference_push  # noqa: E402


class _Wrap:
    """Minimal DataParallel-shaped wrapper (reference accesses .module)."""

    def __init__(self, m):
        self.module = m

    def eval(self):
        self.module.eval()


class RefFormatFolder(Dataset):
    """((img, label), (path, label)) items — the reference MyImageFolder
    format (utils/helpers.py:8) — with a .transform attr pass 2 re-uses."""

    def __init__(self, samples, transform):
        self.samples = samples            # [(path, label)]
        self.transform = transform

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, i):
        from PIL import Image
        path, label = self.samples[i]
        img = self.transform(Image.open(path).convert('RGB'))
        return (img, label), (path, label)


def make_image_tree(root, C=3, per_class=4, size=48, seed=0):
    from PIL import Image
    rng = np.random.default_rng(seed)
    samples = []
    for c in range(C):
        d = os.path.join(root, f'class_{c:02d}')
        os.makedirs(d, exist_ok=True)
        for i in range(per_class):
            # class-dependent structure so distances are well-separated
            base = rng.normal(loc=(c + 1) * 60, scale=40, size=(size, size, 3))
            arr = np.clip(base + rng.normal(0, 25, (size, size, 3)), 0, 255)
            p = os.path.join(d, f'im{i}.png')
            Image.fromarray(arr.astype(np.uint8)).save(p)
            samples.append((p, c))
    return samples


def run_ab_push(img=64, C=3, K=2, d=16, per_class=4, device='cpu'):
    # img 64 -> 4x4 latent grid: the reference's blocked compute_log_prob
    # asserts N %% 4 == 0 even for the B=1 pass-2 re-forwards (model.py:260)
    from mgproto_amd.data import transforms as T
    from mgproto_amd.data.preprocess import preprocess_input_function
    from mgproto_amd.engine.push import push_prototypes

    ref_model_mod, _ = load_reference()
    ref_push = load_reference_push()

    # the reference renders artifacts UNCONDITIONALLY in pass 2
    # (push.py:203-227): give its cv2/plt stubs just enough behavior —
    # the rendering outputs are not part of the comparison
    import types as _t  # noqa: F401
    cv2 = sys.modules['cv2']
    from PIL import Image as _PILImage

    def _cv2_resize(arr, dsize=None, interpolation=None):
        im = _PILImage.fromarray(np.asarray(arr, dtype=np.float32), mode='F')
        return np.asarray(im.resize(dsize, _PILImage.BICUBIC),
                          dtype=np.float32)

    cv2.resize = _cv2_resize
    cv2.INTER_CUBIC = 2
    cv2.applyColorMap = lambda a, m: np.zeros(a.shape + (3,), np.uint8)
    cv2.COLORMAP_JET = 2
    cv2.cvtColor = lambda a, code: a[..., ::-1].copy()
    cv2.COLOR_RGB2BGR = 4
    cv2.rectangle = lambda img, p1, p2, color, thickness=1: img

    def _ccws(mask, connectivity=8, ltype=None):
        # scipy-backed stand-in for cv2.connectedComponentsWithStats; the
        # reference only consumes (n_labels, labeled_img) (helpers.py:44-48)
        from scipy import ndimage
        structure = np.ones((3, 3)) if connectivity == 8 else None
        labeled, n = ndimage.label(mask, structure=structure)
        return n + 1, labeled, None, None

    cv2.connectedComponentsWithStats = _ccws
    cv2.CV_32S = 4
    sys.modules['matplotlib.pyplot'].imsave = lambda *a, **k: None

    with tempfile.TemporaryDirectory() as root:
        samples = make_image_tree(root, C=C, per_class=per_class, size=img)
        tf = T.Compose([T.Resize(size=(img, img)), T.ToTensor()])
        ds = RefFormatFolder(samples, tf)
        loader = DataLoader(ds, batch_size=4, shuffle=False)

        torch.manual_seed(0)
        ref_net = ref_model_mod.construct_MGProto(
            'resnet18', pretrained=False, img_size=img,
            prototype_shape=(C * K, d, 1, 1), num_classes=C,
            add_on_layers_type='regular', sz_embedding=8,
            mem_capacity=8, mine_K=2)
        from mgproto_amd.model import construct_MGProto
        torch.manual_seed(1)
        our_net = construct_MGProto(
            'resnet18', pretrained=False, img_size=img,
            prototype_shape=(C * K, d, 1, 1), num_classes=C,
            add_on_layers_type='regular', sz_embedding=8,
            mem_capacity=8, mine_K=2)
        our_net.load_state_dict(ref_net.state_dict())   # identical weights
        if device != 'cpu':
            ref_net = ref_net.to(device)
            our_net = our_net.to(device)

        ref_art = os.path.join(root, '_ref_art')
        os.makedirs(ref_art, exist_ok=True)
        ref_push.push_prototypes(
            loader, _Wrap(ref_net), class_specific=True,
            preprocess_input_function=preprocess_input_function,
            root_dir_for_saving_prototypes=ref_art,
            prototype_img_filename_prefix='p', log=lambda *a: None)

        chosen = push_prototypes(
            loader, our_net, class_specific=True,
            preprocess_input_function=preprocess_input_function,
            root_dir_for_saving_prototypes=None, log=lambda *a: None)

        ours_by_proto = {j: samples[idx][0]
                         for (j, idx, _h, _w) in chosen}
        # the decisive comparison: both pushes copied their chosen patch
        # feature into prototype_means — identical weights + identical
        # images mean identical assignments leave identical means
        d_means = (ref_net.prototype_means.data
                   - our_net.prototype_means.data).abs().max().item()
        return d_means, chosen, ours_by_proto


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--out', type=str, default='profiles/ab_push.md')
    ap.add_argument('--device', type=str, default='cpu')
    args = ap.parse_args()
    d_means, chosen, by_proto = run_ab_push(device=args.device)
    lines = ['# Push-projection equivalence A/B', '',
             'Reference push.py vs engine/push.py on identical models '
             '(reference state_dict loaded strict), same on-disk images. '
             f'device={args.device}.', '',
             f'- max |Δ prototype_means| after push: **{d_means:.2e}**',
             f'- prototypes re-anchored by our push: {len(chosen)} '
             f'of {max(j for j, *_ in chosen) + 1 if chosen else 0}',
             '',
             'Assignments (our push):']
    for (j, idx, h, w), path in zip(chosen, by_proto.values()):
        lines.append(f'- prototype {j} <- image {idx} patch ({h},{w})')
    with open(args.out, 'w') as f:
        f.write('\n'.join(lines) + '\n')
    print('\n'.join(lines[:8]))


if __name__ == '__main__':
    main()
