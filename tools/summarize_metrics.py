#!/usr/bin/env python3
"""Summarize a run's metrics.jsonl: epoch table (loss/acc/gates) and the
best test accuracy. Companion to MetricsLogger (no plotting deps here).

    python tools/summarize_metrics.py saved_models/.../metrics.jsonl
"""

import argparse
import json


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('path')
    ap.add_argument('--last', type=int, default=20,
                    help='show at most the last N epochs')
    args = ap.parse_args()

    recs = [json.loads(l) for l in open(args.path) if l.strip()]
    epochs = {}
    cur = {}
    for r in recs:
        cur.update(r)
        if 'epoch' in r:                     # epoch-end marker record
            epochs[int(r['epoch'])] = dict(cur)

    if not epochs:
        print(f'{len(recs)} records, no epoch markers yet')
        if recs:
            last = recs[-1]
            print('latest:', {k: v for k, v in last.items()
                              if not k.startswith('_')})
        return

    keys = ['train/loss', 'train/acc', 'test/acc', 'lr', 'use_mining',
            'update_GMM']
    print(f"{'epoch':>5}  " + '  '.join(f'{k.split("/")[-1]:>10}'
                                        for k in keys))
    shown = sorted(epochs)[-args.last:]
    for e in shown:
        row = epochs[e]
        print(f'{e:>5}  ' + '  '.join(
            f'{row.get(k, float("nan")):>10.4f}' if isinstance(
                row.get(k), (int, float)) else f'{"-":>10}'
            for k in keys))
    best = max((v.get('test/acc', float('-inf')), k)
               for k, v in epochs.items())
    if best[0] > float('-inf'):
        print(f'best test acc: {best[0]:.4f} (epoch {best[1]})')


if __name__ == '__main__':
    main()
