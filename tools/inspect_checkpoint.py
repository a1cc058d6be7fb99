#!/usr/bin/env python3
"""Checkpoint inspector: prints what a .pth holds — reference-layout model
keys, shapes/hyperparams inferred from them, memory-bank fill, mixture
prior stats, plus resume metadata (epoch/optimizers/RNG) when present.

    python tools/inspect_checkpoint.py saved_models/.../latest.pth
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('path')
    args = ap.parse_args()

    state = torch.load(args.path, map_location='cpu', weights_only=False)
    is_train_state = isinstance(state, dict) and 'model' in state
    sd = state['model'] if is_train_state else state

    print(f'file: {args.path} ({os.path.getsize(args.path) / 1e6:.1f} MB)')
    if is_train_state:
        print(f"train state: epoch={state.get('epoch')} "
              f"optimizers={sorted(state.get('optimizers', {}))} "
              f"schedulers={sorted(state.get('schedulers', {}))} "
              f"rng={'yes' if state.get('torch_rng') is not None else 'no'}")

    n_params = sum(v.numel() for v in sd.values() if torch.is_tensor(v))
    print(f'model: {len(sd)} tensors, {n_params / 1e6:.2f} M elements')

    if 'prototype_means' in sd:
        m = sd['prototype_means']
        C, K, d = m.shape
        print(f'prototypes: C={C} classes x K={K} components, d={d} '
              f'(mean norm {m.view(-1, d).norm(dim=1).mean():.4f})')
    if 'last_layer.weight' in sd:
        w = sd['last_layer.weight']
        C = w.shape[0]
        K = w.shape[1] // C
        diag = torch.arange(C)
        pi = w.view(C, C, K)[diag, diag]
        nz = int((pi > 1e-6).sum())
        print(f'mixture priors: per-class sum {pi.sum(1).mean():.4f}, '
              f'{nz}/{C * K} active (pruned: {C * K - nz})')
    cls_keys = [k for k in sd if k.startswith('queue.cls')]
    if cls_keys and 'queue.mem_len' in sd:
        lens = sd['queue.mem_len']
        cap = sd[cls_keys[0]].shape[0]
        full = int((lens == cap).sum())
        print(f'memory bank: {len(cls_keys)} classes x cap {cap}, '
              f'{full} full, fill {float(lens.float().mean()) / cap * 100:.1f}%')
    if 'iteration_counter' in sd:
        print(f"iterations: {int(sd['iteration_counter'].item())}")

    backbones = sorted({k.split('.')[0] for k in sd})
    print(f'top-level groups: {backbones}')


if __name__ == '__main__':
    main()
