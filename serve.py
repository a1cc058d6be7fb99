#!/usr/bin/env python3
"""Model-serving driver: classification + OoD score + prototype explanations
over HTTP (FastAPI/uvicorn). No counterpart in the reference repo.

    python serve.py --resume saved_models/.../latest.pth --arch resnet50 \
        --addon regular_upsample --port 8000 [--capture-batch 8]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

from mgproto_amd.model import construct_MGProto
from mgproto_amd.serving import InferenceEngine, create_app


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--resume', type=str, default=None)
    ap.add_argument('--arch', type=str, default='resnet50')
    ap.add_argument('--addon', type=str, default='regular_upsample')
    ap.add_argument('--classes', type=int, default=200)
    ap.add_argument('--proto-dim', type=int, default=64)
    ap.add_argument('--proto-per-class', type=int, default=10)
    ap.add_argument('--img', type=int, default=224)
    ap.add_argument('--host', type=str, default='127.0.0.1')
    ap.add_argument('--port', type=int, default=8000)
    ap.add_argument('--capture-batch', type=int, default=0,
                    help='hipGraph-capture the forward at this batch size')
    args = ap.parse_args()

    device = torch.device('cuda', 0) if torch.cuda.is_available() \
        else torch.device('cpu')
    if device.type == 'cuda':
        from mgproto_amd.utils.helpers import setup_miopen_db
        setup_miopen_db()
        torch.backends.cudnn.benchmark = True

    kw = {}
    sd = None
    if args.resume:
        from mgproto_amd.utils.checkpoint import infer_ctor_kwargs_from_state
        sd = torch.load(args.resume, map_location=device, weights_only=False)
        sd = sd.get('model', sd)
        kw = infer_ctor_kwargs_from_state(sd)
    model = construct_MGProto(
        args.arch, pretrained=False, img_size=args.img,
        prototype_shape=(args.classes * args.proto_per_class,
                         args.proto_dim, 1, 1),
        num_classes=args.classes, add_on_layers_type=args.addon,
        **kw).to(device)
    if sd is not None:
        model.load_state_dict(sd, strict=False)
    if device.type == 'cuda':
        model.features = model.features.to(memory_format=torch.channels_last)

    engine = InferenceEngine(model, device)
    if args.capture_batch and device.type == 'cuda':
        engine.capture(args.capture_batch)

    import uvicorn
    uvicorn.run(create_app(engine), host=args.host, port=args.port)


if __name__ == '__main__':
    main()
