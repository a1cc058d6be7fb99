#!/usr/bin/env python3
"""Prototype consistency evaluation (reference ``eval_consistency.py``)."""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch
from torch.utils.data import DataLoader

from mgproto_amd.model import construct_MGProto
from mgproto_amd.utils.datasets import Cub2011Eval
from mgproto_amd.utils.local_parts import CubPartAnnotations
from mgproto_amd.utils.interpretability import evaluate_consistency
from mgproto_amd.data import transforms as T
from mgproto_amd.data.preprocess import mean, std


def build_argparser():
    p = argparse.ArgumentParser()
    p.add_argument('--data_path', type=str, required=True,
                   help='CUB_200_2011 root (with images/, parts/, *.txt)')
    p.add_argument('--test_batch_size', type=int, default=64)
    p.add_argument('--nb_classes', type=int, default=200)
    p.add_argument('--base_architecture', type=str, default='vgg19')
    p.add_argument('--prototype_shape', nargs=3, type=int, default=[2000, 64, 1])
    p.add_argument('--addon', type=str, default='regular')
    p.add_argument('--resume', type=str, required=True)
    p.add_argument('--img_size', type=int, default=224)
    p.add_argument('--half_size', type=int, default=36)
    p.add_argument('--mem_sz', type=int, default=None,
                   help='memory-bank capacity per class; default: inferred '
                        'from the checkpoint')
    return p


def load_model(args, device):
    P, d, ks = args.prototype_shape
    sd = torch.load(args.resume, map_location='cpu', weights_only=False)
    if 'model' in sd:
        sd = sd['model']
    # shape-bearing hyperparams not on the reference CLI: infer from the
    # checkpoint so any training config loads
    from mgproto_amd.utils.checkpoint import infer_ctor_kwargs_from_state
    kw = infer_ctor_kwargs_from_state(sd)
    if args.mem_sz is not None:
        kw['mem_capacity'] = args.mem_sz
    model = construct_MGProto(args.base_architecture, pretrained=False,
                              img_size=args.img_size,
                              prototype_shape=(P, d, ks, ks),
                              num_classes=args.nb_classes,
                              add_on_layers_type=args.addon, **kw)
    model.load_state_dict(sd, strict=False)
    return model.to(device).eval()


def build_loader(args):
    tf = T.Compose([T.Resize((args.img_size, args.img_size)), T.ToTensor(),
                    T.Normalize(mean=mean, std=std)])
    ds = Cub2011Eval(args.data_path, train=False, transform=tf)
    return DataLoader(ds, batch_size=args.test_batch_size, num_workers=4)


def main():
    args = build_argparser().parse_args()
    device = torch.device('cuda' if torch.cuda.is_available() else 'cpu')
    model = load_model(args, device)
    loader = build_loader(args)
    ann = CubPartAnnotations(args.data_path)
    score = evaluate_consistency(model, loader, ann, args.data_path,
                                 half_size=args.half_size, device=device)
    print(f'Consistency Score : {score:.2f}%')


if __name__ == '__main__':
    main()
