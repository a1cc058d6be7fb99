#!/usr/bin/env python3
"""Serving benchmark: InferenceEngine latency/throughput, eager vs
hipGraph-captured forward.

    python tools/serve_bench.py --batch 8 --iters 50 [--resume ckpt.pth]

Prints one JSON line per mode: p50/p95 latency (ms) and images/s.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench_mode(engine, x, iters, warmup):
    times = []
    for i in range(warmup + iters):
        if x.is_cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        engine.predict(x, topk_classes=5, explain_topk=3)
        if x.is_cuda:
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) * 1e3
        if i >= warmup:
            times.append(dt)
    times.sort()
    p50 = times[len(times) // 2]
    p95 = times[int(len(times) * 0.95)]
    return p50, p95, x.shape[0] / (sum(times) / len(times)) * 1e3


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--arch', type=str, default='resnet50')
    ap.add_argument('--addon', type=str, default='regular_upsample')
    ap.add_argument('--classes', type=int, default=200)
    ap.add_argument('--proto-dim', type=int, default=64)
    ap.add_argument('--proto-per-class', type=int, default=10)
    ap.add_argument('--img', type=int, default=224)
    ap.add_argument('--batch', type=int, default=8)
    ap.add_argument('--iters', type=int, default=50)
    ap.add_argument('--warmup', type=int, default=10)
    ap.add_argument('--resume', type=str, default=None)
    args = ap.parse_args()

    device = torch.device('cuda', 0) if torch.cuda.is_available() \
        else torch.device('cpu')
    if device.type == 'cuda':
        from mgproto_amd.utils.helpers import setup_miopen_db
        setup_miopen_db()
        torch.backends.cudnn.benchmark = True

    from mgproto_amd.model import construct_MGProto
    from mgproto_amd.serving import InferenceEngine

    kw = {}
    sd = None
    if args.resume:
        from mgproto_amd.utils.checkpoint import infer_ctor_kwargs_from_state
        sd = torch.load(args.resume, map_location=device, weights_only=False)
        sd = sd.get('model', sd)
        kw = infer_ctor_kwargs_from_state(sd)
    C, K, d = args.classes, args.proto_per_class, args.proto_dim
    torch.manual_seed(0)
    model = construct_MGProto(args.arch, pretrained=False, img_size=args.img,
                              prototype_shape=(C * K, d, 1, 1), num_classes=C,
                              add_on_layers_type=args.addon, **kw).to(device)
    if sd is not None:
        model.load_state_dict(sd, strict=False)
    if device.type == 'cuda':
        model.features = model.features.to(memory_format=torch.channels_last)

    x = torch.randn(args.batch, 3, args.img, args.img, device=device)
    engine = InferenceEngine(model, device)
    for mode in (['eager', 'graph'] if device.type == 'cuda' else ['eager']):
        if mode == 'graph':
            try:
                engine.capture(args.batch, args.img)
            except Exception as e:  # noqa: BLE001
                print(f'# capture failed: {e}', flush=True)
                continue
        p50, p95, ips = bench_mode(engine, x, args.iters, args.warmup)
        print(json.dumps({'mode': mode, 'batch': args.batch,
                          'p50_ms': round(p50, 3), 'p95_ms': round(p95, 3),
                          'images_per_sec': round(ips, 1),
                          'arch': args.arch, 'device': device.type}))


if __name__ == '__main__':
    main()
