#!/usr/bin/env python3
"""Prototype stability evaluation (reference ``eval_stability.py``)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

from eval_consistency import build_argparser, load_model, build_loader
from mgproto_amd.utils.local_parts import CubPartAnnotations
from mgproto_amd.utils.interpretability import evaluate_stability


def main():
    args = build_argparser().parse_args()
    device = torch.device('cuda' if torch.cuda.is_available() else 'cpu')
    model = load_model(args, device)
    loader = build_loader(args)
    ann = CubPartAnnotations(args.data_path)
    score = evaluate_stability(model, loader, ann, args.data_path,
                               half_size=args.half_size, device=device)
    print(f'Stability Score : {score:.2f}%')


if __name__ == '__main__':
    main()
