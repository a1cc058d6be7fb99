#!/usr/bin/env python3
"""Offline MIOpen auto-tune (SEARCH) for the flagship conv shapes.

The shipped find-db (mgproto_amd/miopen_db) records which solver wins per
conv shape under the DEFAULT tuning parameters; MIOPEN_FIND_ENFORCE=4
(SEARCH_DB_UPDATE) additionally tunes each solver's parameter space and
writes the winning config to the user perf-db. The 3x3 convs are 13.4 ms
of the 48 ms step and the 1x1 CK picks 11.2 ms (profiles/README.md) —
this is library tuning, the biggest remaining lever.

Run ON a GPU box with a wall budget; shapes are tuned heaviest-first and
the db is written incrementally, so a timeout keeps everything tuned so
far:

    gpurun --timeout 2700 -- \
        'python tools/tune_miopen.py --budget 1800 --db gpurun_out/miopen_tuned'

then commit the db files into mgproto_amd/miopen_db/.
"""

import argparse
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)


def collect_conv_shapes(arch='resnet50', img=224, batch=80,
                        addon='regular_upsample'):
    """Unique (Cin, Cout, kh, kw, stride, pad, H, W, groups) of one
    forward pass, via shape hooks on a CPU meta run."""
    import torch
    from mgproto_amd.model import construct_MGProto
    model = construct_MGProto(arch, pretrained=False, img_size=img,
                              prototype_shape=(2000, 64, 1, 1),
                              num_classes=200, add_on_layers_type=addon,
                              sz_embedding=32, mem_capacity=8, mine_K=2)
    shapes = []

    def hook(m, inp, out):
        x = inp[0]
        shapes.append((x.shape[1], m.out_channels, m.kernel_size[0],
                       m.kernel_size[1], m.stride[0], m.padding[0],
                       x.shape[2], x.shape[3], m.groups))

    hs = [m.register_forward_hook(hook) for m in model.modules()
          if isinstance(m, torch.nn.Conv2d)]
    with torch.no_grad():
        model.conv_features(torch.zeros(1, 3, img, img))
    for h in hs:
        h.remove()
    # dedup, keep an estimated-cost order (heaviest first)
    uniq = {}
    for s in shapes:
        cin, cout, kh, kw, st, pad, H, W, g = s
        cost = (H // st) * (W // st) * cin * cout * kh * kw
        uniq[s] = cost
    return sorted(uniq, key=lambda s: -uniq[s]), batch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--budget', type=int, default=1800,
                    help='wall seconds for the tuning loop')
    ap.add_argument('--db', type=str, default='gpurun_out/miopen_tuned')
    ap.add_argument('--arch', type=str, default='resnet50')
    ap.add_argument('--batch', type=int, default=80)
    ap.add_argument('--seed-from-repo', action='store_true',
                    help='start from the shipped find-db instead of empty')
    args = ap.parse_args()

    os.makedirs(args.db, exist_ok=True)
    if args.seed_from_repo:
        import shutil
        src = os.path.join(ROOT, 'mgproto_amd', 'miopen_db')
        for f in os.listdir(src):
            shutil.copy2(os.path.join(src, f), os.path.join(args.db, f))
    os.environ['MIOPEN_USER_DB_PATH'] = os.path.abspath(args.db)
    os.environ['MIOPEN_FIND_ENFORCE'] = '4'        # SEARCH_DB_UPDATE

    import torch
    assert torch.cuda.is_available()
    torch.backends.cudnn.benchmark = True
    dev = torch.device('cuda', 0)

    shapes, _ = collect_conv_shapes(args.arch, batch=args.batch)
    print(f'{len(shapes)} unique conv shapes; budget {args.budget}s',
          flush=True)
    t0 = time.time()
    done = 0
    for (cin, cout, kh, kw, st, pad, H, W, g) in shapes:
        if time.time() - t0 > args.budget:
            print(f'budget exhausted after {done}/{len(shapes)} shapes',
                  flush=True)
            break
        te = time.time()
        x = torch.randn(args.batch, cin, H, W, device=dev,
                        dtype=torch.bfloat16) \
            .to(memory_format=torch.channels_last).requires_grad_(True)
        w = torch.randn(cout, cin // g, kh, kw, device=dev,
                        dtype=torch.bfloat16) \
            .to(memory_format=torch.channels_last).requires_grad_(True)
        y = torch.nn.functional.conv2d(x, w, stride=st, padding=pad, groups=g)
        y.sum().backward()        # fwd + bwd-data + bwd-weight searches
        torch.cuda.synchronize()
        done += 1
        print(f'tuned {cin}x{H}x{W} -> {cout} k{kh} s{st} '
              f'({time.time() - te:.0f}s; total {time.time() - t0:.0f}s)',
              flush=True)
    print(f'db files in {args.db}:', os.listdir(args.db), flush=True)


if __name__ == '__main__':
    main()
