#!/usr/bin/env python3
"""Per-shape fused-BN bandwidth microbench (GPU).

Times ext.bn_fwd / ext.bn_bwd on each R50-flagship BN shape and reports
achieved GB/s against the analytic traffic bound. Used to A/B the
reduction-kernel MLP unroll (round-2: stats measured 2.1 TB/s, bwd-reduce
2.6 TB/s — one 16B load in flight per thread per trip).

    gpurun -- 'python tools/bench_bn.py'
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    from mgproto_amd.ops import hip_loader
    ext = hip_loader.load()
    dev = torch.device('cuda', 0)
    B = 80
    # (C, H): the R50-noMaxpool stage shapes (plus conv1)
    shapes = [(64, 112), (256, 112), (128, 56), (512, 56),
              (256, 28), (1024, 28), (512, 14), (2048, 14)]
    rows = []
    for C, H in shapes:
        M = B * H * H
        x = torch.randn(M, C, device=dev).bfloat16()
        w = torch.ones(C, device=dev)
        b = torch.zeros(C, device=dev)
        rm = torch.zeros(C, device=dev)
        rv = torch.ones(C, device=dev)

        def fwd():
            return ext.bn_fwd(x, w, b, rm, rv, True, 0.1, 1e-5, True, None,
                              True)
        y, mean, rstd, mask = fwd()
        dy = torch.randn_like(x)

        def bwd():
            # mask path (the default): y_or_mask = mask, use_mask=True
            return ext.bn_bwd(dy, mask, x, w, mean, rstd, True, True,
                              False, True)

        tf = timeit(fwd)
        tb = timeit(bwd)
        gb = M * C * 2 / 1e9
        # fwd: stats read x, apply read x write y (+mask/16) ~= 3 passes
        # bwd(mask): reduce read dy,x; apply read dy,x write dx ~= 5 passes
        f_bw = gb * 3 / (tf / 1e3)
        b_bw = gb * 5 / (tb / 1e3)
        rows.append((C, H, tf, f_bw, tb, b_bw))
        print(f'C={C:5d} H={H:4d}  fwd {tf:7.3f} ms ({f_bw:6.0f} GB/s)   '
              f'bwd {tb:7.3f} ms ({b_bw:6.0f} GB/s)', flush=True)
    tot_f = sum(r[2] for r in rows)
    tot_b = sum(r[4] for r in rows)
    print(f'stage-total fwd {tot_f:.3f} ms, bwd {tot_b:.3f} ms '
          f'(one call per unique shape; the model runs each 2-8x)')


if __name__ == '__main__':
    main()
