#!/usr/bin/env python3
"""Sectioned timing of the flagship step: isolates backbone vs prototype path
vs losses vs optimizer. GPU-only diagnostic tool."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from mgproto_amd.model import construct_MGProto
from mgproto_amd.losses import build_aux_loss


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--batch', type=int, default=80)
    ap.add_argument('--arch', type=str, default='resnet50')
    ap.add_argument('--addon', type=str, default='regular_upsample')
    ap.add_argument('--no-channels-last', action='store_true')
    ap.add_argument('--no-amp', action='store_true')
    ap.add_argument('--no-find', action='store_true')
    args = ap.parse_args()

    from mgproto_amd.utils.helpers import setup_miopen_db
    setup_miopen_db()
    torch.backends.cudnn.benchmark = not args.no_find
    dev = torch.device('cuda', 0)
    C, K, d = 200, 10, 64
    torch.manual_seed(0)
    model = construct_MGProto(args.arch, pretrained=False, img_size=224,
                              prototype_shape=(C * K, d, 1, 1), num_classes=C,
                              add_on_layers_type=args.addon, sz_embedding=32,
                              mem_capacity=800, mine_K=20).to(dev)
    if not args.no_channels_last:
        model.features = model.features.to(memory_format=torch.channels_last)
    aux = build_aux_loss('Proxy_Anchor', nb_classes=C, sz_embed=32).to(dev)
    model.train()

    x = torch.randn(args.batch, 3, 224, 224, device=dev)
    if not args.no_channels_last:
        x = x.contiguous(memory_format=torch.channels_last)
    gt = torch.randint(0, C, (args.batch,), device=dev)

    amp = (lambda: torch.autocast('cuda', dtype=torch.bfloat16)) \
        if not args.no_amp else (lambda: torch.enable_grad())

    # 1. backbone fwd only
    def bb_fwd():
        with torch.no_grad(), amp():
            model.features(x)
    print(f'backbone fwd: {timeit(bb_fwd):8.2f} ms')

    # 2. backbone fwd+bwd
    def bb_fwdbwd():
        with amp():
            y = model.features(x)
        y.float().sum().backward()
    print(f'backbone fwd+bwd: {timeit(bb_fwdbwd):8.2f} ms')

    # 3. conv_features (backbone + addons + embed)
    def cf():
        with torch.no_grad(), amp():
            model.conv_features(x)
    print(f'conv_features fwd: {timeit(cf):8.2f} ms')

    # 4. full forward no grad
    def fwd():
        with torch.no_grad(), amp():
            model(x, gt)
    print(f'full fwd (train, enqueue): {timeit(fwd):8.2f} ms')

    # 5. full forward+backward+losses
    def full():
        with amp():
            out, emb = model(x, gt)
        out = out.float()
        mine = sum(F.cross_entropy(out[:, :, k], gt)
                   for k in range(1, out.shape[2])) / (out.shape[2] - 1)
        loss = F.cross_entropy(out[:, :, 0], gt) + 0.2 * mine \
            + 0.5 * aux(emb.float(), gt)
        model.zero_grad(set_to_none=True)
        loss.backward()
    print(f'full fwd+bwd: {timeit(full):8.2f} ms')

    # 6. prototype section alone (on cached features)
    with torch.no_grad(), amp():
        bf, _ = model.conv_features(x)
    bf = bf.float()
    bfl = bf.detach().requires_grad_(True)

    def proto():
        emb = torch.zeros(args.batch, 32, device=dev)
        out, _ = model._prototype_forward(bfl, emb, gt)
        out.float().sum().backward()
    print(f'prototype fwd+bwd: {timeit(proto):8.2f} ms')

    # 7. EM
    with torch.no_grad():
        mem = F.normalize(torch.randn(C * 800, d, device=dev), dim=1)
        labels = torch.arange(C, device=dev).repeat_interleave(800)
        model.queue.push(mem, labels)

    def em():
        model.memory_updated_cls[:] = True
        model.update_GMM()
    print(f'EM (all 200 classes): {timeit(em):8.2f} ms')


if __name__ == '__main__':
    main()
