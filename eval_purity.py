#!/usr/bin/env python3
"""Prototype purity evaluation (reference ``eval_purity.py``).

Two purity variants, as in the reference:
* region-based (interpretability.evaluate_purity): top-K activating images
  per prototype, max-region vs part annotations;
* PIP-Net CSV-based (utils/cub_csv): patch-coordinate CSVs scored against
  part locations (``--csv`` mode).
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch
from torch.utils.data import DataLoader

from eval_consistency import build_argparser, load_model, build_loader
from mgproto_amd.utils.local_parts import CubPartAnnotations
from mgproto_amd.utils.interpretability import evaluate_purity
from mgproto_amd.utils import cub_csv


def main():
    parser = build_argparser()
    parser.add_argument('--topK', type=int, default=10)
    parser.add_argument('--csv', action='store_true',
                        help='PIP-Net CSV purity instead of region purity')
    parser.add_argument('--log_dir', type=str, default='./purity_logs')
    args = parser.parse_args()
    device = torch.device('cuda' if torch.cuda.is_available() else 'cpu')
    model = load_model(args, device)
    loader = build_loader(args)

    if args.csv:
        csvpath = cub_csv.get_topk_cub(model, loader, k=args.topK, epoch=0,
                                       device=device, log_dir=args.log_dir,
                                       img_size=args.img_size)
        with torch.no_grad():
            x = loader.dataset[0][0].unsqueeze(0).to(device)
            _, dist = model.push_forward(x)
        mean_p, std_p, related = cub_csv.eval_prototypes_cub_parts_csv(
            csvpath,
            os.path.join(args.data_path, 'parts', 'part_locs.txt'),
            os.path.join(args.data_path, 'parts', 'parts.txt'),
            os.path.join(args.data_path, 'images.txt'),
            epoch=0, img_size=args.img_size, wshape=dist.shape[-1])
        print(f'CSV Purity : {mean_p * 100:.2f}% (std {std_p * 100:.2f}), '
              f'part-related prototypes: {related}')
    else:
        ann = CubPartAnnotations(args.data_path)
        mean_p, std_p = evaluate_purity(model, loader, ann, args.data_path,
                                        half_size=16, topK=args.topK,
                                        device=device)
        print(f'Purity Score : {mean_p:.2f}% (std {std_p:.2f})')


if __name__ == '__main__':
    main()
